"""Core reporter: trace-event ingestion, labeling, batching, flush loop.

The arrowReporter analog (reference: reporter/parca_reporter.go):
- `report_trace_event` is the hot path called for every CPU sample, GPU
  kernel timing, GPU PC sample batch, probe fire and OOM profile
  (parca_reporter.go:321-457).
- Per-PID labels come from metadata providers through an LRU with TTL
  (parca_reporter.go:762-798); per-sample cpu/thread_id patching happens
  outside the cache (812-847); Prometheus relabeling may drop the PID.
- A flush loop swaps the pending batch every `batch_write_interval`
  (default 10 s, parca_reporter.go:1440-1525) and hands it to the
  configured destinations: remote ProfileStore (Arrow), local store
  directory (pprof files), offline-mode framed log.
"""

from __future__ import annotations

import logging
import os
import random
import threading
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Protocol, Sequence, Tuple

from ..lru import LRU
from ..model import (
    SampleType,
    Trace,
    TraceEventMeta,
    TraceOrigin,
    sample_type_for,
)
from ..relabel import RelabelConfig, relabel, strip_meta_labels

log = logging.getLogger("parca_agent_amd.reporter")

# Longest label value shipped (fork support.CustomLabelMaxValLen analog).
MAX_LABEL_VALUE_LEN = 255


def sanitize_label_value(value: str, max_len: int = MAX_LABEL_VALUE_LEN) -> str:
    """Clamp to max_len WITHOUT splitting a multi-byte codepoint, and
    repair invalid surrogates (reference test behaviour: UTF-8
    truncation repair, parca_reporter_test.go:44-101)."""
    if not isinstance(value, str):
        value = str(value)
    # Round-trip drops lone surrogates from bad decodes.
    value = value.encode("utf-8", "replace").decode("utf-8", "replace")
    encoded = value.encode("utf-8")
    if len(encoded) <= max_len:
        return value
    cut = encoded[:max_len]
    # Back off to a codepoint boundary.
    while cut and (cut[-1] & 0xC0) == 0x80:
        cut = cut[:-1]
    if cut and cut[-1] >= 0xC0:
        cut = cut[:-1]
    return cut.decode("utf-8", "ignore")


@dataclass
class PendingSample:
    trace: Trace
    labels: Dict[str, str]
    value: int
    timestamp_ns: int
    sample_type: SampleType
    period: int
    duration_ns: int = 0


class Destination(Protocol):
    def write_batch(self, samples: List[PendingSample]) -> None: ...

    def close(self) -> None: ...


class MetadataProviderLike(Protocol):
    name: str

    def add_metadata(self, pid: int, labels: Dict[str, str]) -> bool: ...


@dataclass
class ReporterMetrics:
    samples_ingested: int = 0
    samples_dropped_relabel: int = 0
    samples_by_origin: Dict[str, int] = field(default_factory=dict)
    batches_written: int = 0
    batch_errors: int = 0
    last_flush_duration_s: float = 0.0

    def observe(self, origin: TraceOrigin) -> None:
        self.samples_ingested += 1
        key = origin.value
        self.samples_by_origin[key] = self.samples_by_origin.get(key, 0) + 1


class Reporter:
    def __init__(
        self,
        destinations: Sequence[Destination],
        metadata_providers: Sequence[MetadataProviderLike] = (),
        relabel_configs: Sequence[RelabelConfig] = (),
        cpu_sampling_frequency: int = 19,
        batch_write_interval: float = 10.0,
        merge_gpu_profiles: bool = False,
        label_ttl_seconds: float = 600.0,
        disable_cpu_label: bool = False,
        disable_thread_id_label: bool = False,
        disable_thread_comm_label: bool = False,
        label_cache_size: int = 8192,
        stack_cache_size: int = 65536,
        clock_ns: Callable[[], int] = time.time_ns,
        aggregate_batches: bool = True,
    ) -> None:
        self.destinations = list(destinations)
        self.metadata_providers = list(metadata_providers)
        self.relabel_configs = list(relabel_configs)
        self.cpu_sampling_frequency = cpu_sampling_frequency
        self.batch_write_interval = batch_write_interval
        self.merge_gpu_profiles = merge_gpu_profiles
        self.aggregate_batches = aggregate_batches
        self.disable_cpu_label = disable_cpu_label
        self.disable_thread_id_label = disable_thread_id_label
        self.disable_thread_comm_label = disable_thread_comm_label
        self.metrics = ReporterMetrics()
        self._clock_ns = clock_ns

        # pid -> Optional[labels]; None caches a relabel-drop decision
        # (reference keeps dropped PIDs cached too, parca_reporter.go:776).
        self._pid_labels: LRU[int, Optional[Dict[str, str]]] = LRU(
            label_cache_size, ttl_seconds=label_ttl_seconds)
        # trace hash -> Trace (dedup retained stacks between flushes).
        self._stacks: LRU[bytes, Trace] = LRU(stack_cache_size)
        # GpuConfig per (pid, gpu): ns per PC sample (parca_reporter.go:89-102)
        self._gpu_ns_per_sample: Dict[Tuple[int, int], float] = {}

        self._mu = threading.Lock()
        self._pending: List[PendingSample] = []
        # custom-labels tuple -> sanitized dict (see _sanitized_custom)
        self._custom_label_memo: Dict[tuple, Dict[str, str]] = {}
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    # -- labels ------------------------------------------------------------

    def labels_for_pid(self, pid: int) -> Optional[Dict[str, str]]:
        cached = self._pid_labels.get(pid, default="MISS")
        if cached != "MISS":
            return cached
        labels: Dict[str, str] = {}
        for provider in self.metadata_providers:
            try:
                provider.add_metadata(pid, labels)
            except Exception:
                log.debug("metadata provider %s failed for pid %d",
                          provider.name, pid, exc_info=True)
        if self.relabel_configs:
            relabeled = relabel(labels, self.relabel_configs)
            if relabeled is None:
                self._pid_labels.put(pid, None)
                return None
            labels = relabeled
        labels = strip_meta_labels(labels)
        labels = {k: sanitize_label_value(v) for k, v in labels.items()}
        self._pid_labels.put(pid, labels)
        return labels

    def set_gpu_config(self, pid: int, gpu_id: int, ns_per_sample: float) -> None:
        self._gpu_ns_per_sample[(pid, gpu_id)] = ns_per_sample

    def _sanitized_custom(self, custom: tuple) -> Dict[str, str]:
        cached = self._custom_label_memo.get(custom)
        if cached is not None:
            return cached
        out: Dict[str, str] = {}
        for k, v in custom:
            try:
                k.encode("utf-8")
            except UnicodeEncodeError:
                continue  # non-UTF8 key: drop the label entirely
            if not k:
                continue
            out[k] = sanitize_label_value(v)
        if len(self._custom_label_memo) > 4096:
            self._custom_label_memo.clear()
        self._custom_label_memo[custom] = out
        return out

    # -- hot path ----------------------------------------------------------

    def report_trace_event(self, trace: Trace, meta: TraceEventMeta) -> None:
        base_labels = self.labels_for_pid(meta.pid)
        if base_labels is None:
            self.metrics.samples_dropped_relabel += 1
            return

        st = meta.sample_type if meta.sample_type is not None else \
            sample_type_for(meta.origin, self.merge_gpu_profiles)
        labels = dict(base_labels)
        # Per-sample custom labels (probe names, Go labels, ...) — the
        # reference's per-sample relabel pass input
        # (parca_reporter.go:800-841). Keys that don't encode as UTF-8
        # are dropped and values get the truncation repair, like the
        # write-time pass at parca_reporter.go:362-374; memoized because
        # label sets are low-cardinality but per-sample.
        if trace.custom_labels:
            labels.update(self._sanitized_custom(trace.custom_labels))
        # Per-sample patching (parca_reporter.go:812-847), each label
        # individually defeatable (flags.go:338-340).
        if not self.disable_thread_comm_label and meta.comm and \
                labels.get("comm") != meta.comm:
            labels["thread_comm"] = meta.comm
        if not self.disable_thread_id_label and meta.tid:
            labels["thread_id"] = str(meta.tid)
        if not self.disable_cpu_label and meta.cpu >= 0:
            labels["cpu"] = str(meta.cpu)
        if meta.gpu_id >= 0:
            labels["gpu"] = str(meta.gpu_id)
        if self.merge_gpu_profiles and meta.origin in (
                TraceOrigin.GPU_KERNEL, TraceOrigin.GPU_PC):
            labels["gpu_view"] = (
                "kernel_time" if meta.origin == TraceOrigin.GPU_KERNEL
                else "pc_sample")

        # Probe samples additionally run a per-sample relabel pass so
        # rules can act on (or drop by) per-sample fields like
        # thread_id/probe; other origins keep the cheap patch-and-ship
        # path (reference gate, parca_reporter.go:805-841).
        if meta.origin == TraceOrigin.PROBE and self.relabel_configs:
            relabeled = relabel(labels, self.relabel_configs)
            if relabeled is None:
                self.metrics.samples_dropped_relabel += 1
                return
            labels = strip_meta_labels(relabeled)

        value = meta.value
        if self.merge_gpu_profiles and meta.origin == TraceOrigin.GPU_PC:
            # Merged view folds PC samples into gpu_time/nanoseconds, so
            # the raw sample COUNT must be scaled to nanoseconds here or
            # it would be summed together with kernel-time ns (reference
            # value *= nsPerSample, parca_reporter.go TraceOriginGpuPC).
            nsps = self._gpu_ns_per_sample.get((meta.pid, meta.gpu_id))
            if nsps is None:
                nsps = self._gpu_ns_per_sample.get((meta.pid, -1))
            if nsps:
                value = int(value * nsps)

        period = self._period_for(meta)
        th = trace.trace_hash()
        cached_trace = self._stacks.get(th)
        if cached_trace is None:
            self._stacks.put(th, trace)
            cached_trace = trace

        sample = PendingSample(
            trace=cached_trace,
            labels=labels,
            value=value,
            timestamp_ns=meta.timestamp_ns or self._clock_ns(),
            sample_type=st,
            period=period,
        )
        with self._mu:
            self._pending.append(sample)
        self.metrics.observe(meta.origin)

    def _period_for(self, meta: TraceEventMeta) -> int:
        if meta.period:
            return meta.period
        if meta.origin == TraceOrigin.SAMPLING:
            return int(1e9 / self.cpu_sampling_frequency)
        if meta.origin == TraceOrigin.GPU_PC:
            nsps = self._gpu_ns_per_sample.get((meta.pid, meta.gpu_id))
            if nsps is None:
                nsps = self._gpu_ns_per_sample.get((meta.pid, -1), 0.0)
            return int(nsps)
        return 0

    # -- flush loop --------------------------------------------------------

    def start(self) -> None:
        if self._thread is not None:
            return
        self._stop.clear()
        self._thread = threading.Thread(
            target=self._run, name="reporter-flush", daemon=True)
        self._thread.start()

    @staticmethod
    def _malloc_trim() -> None:
        """Return freed glibc arenas to the kernel. Flush batches are
        tens of MB of transient encode buffers every interval; without
        the trim the allocator retains the high-water mark forever and
        agent RSS creeps by hundreds of MB on many-core nodes."""
        try:
            import ctypes

            ctypes.CDLL("libc.so.6").malloc_trim(0)
        except Exception:
            pass

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=30)
            self._thread = None
        self.flush()
        for dest in self.destinations:
            try:
                dest.close()
            except Exception:
                log.warning("destination close failed", exc_info=True)

    def _run(self) -> None:
        while not self._stop.is_set():
            # +-20% jitter like the reference's report interval
            # (parca_reporter.go:1486).
            interval = self.batch_write_interval * random.uniform(0.8, 1.2)
            if self._stop.wait(interval):
                break
            try:
                self.flush()
            except Exception:
                log.error("flush failed", exc_info=True)
                self.metrics.batch_errors += 1
            self._malloc_trim()

    @staticmethod
    def aggregate_batch(batch: List[PendingSample]) -> List[PendingSample]:
        """Collapse rows identical in (stack, labels, type, period) by
        summing values (counts and nanoseconds both sum correctly);
        timestamp becomes the window's latest. At 19 Hz x hundreds of
        cores a 10 s window repeats hot stacks heavily, cutting encode
        and wire cost several-fold with unchanged pprof totals."""
        merged: Dict[tuple, PendingSample] = {}
        out: List[PendingSample] = []
        for s in batch:
            if s.sample_type.sample_type == "probe":
                # Probe fires are individual span-like events; summing
                # their durations would destroy per-invocation data.
                out.append(s)
                continue
            key = (s.trace.trace_hash(), tuple(sorted(s.labels.items())),
                   s.sample_type, s.period)
            found = merged.get(key)
            if found is None:
                s = PendingSample(trace=s.trace, labels=s.labels,
                                  value=s.value,
                                  timestamp_ns=s.timestamp_ns,
                                  sample_type=s.sample_type,
                                  period=s.period,
                                  duration_ns=s.duration_ns)
                merged[key] = s
                out.append(s)
            else:
                found.value += s.value
                if s.timestamp_ns > found.timestamp_ns:
                    found.timestamp_ns = s.timestamp_ns
        return out

    def flush(self) -> int:
        with self._mu:
            batch, self._pending = self._pending, []
        if not batch:
            return 0
        n_ingested = len(batch)
        if self.aggregate_batches:
            batch = self.aggregate_batch(batch)
        t0 = time.monotonic()
        for dest in self.destinations:
            try:
                dest.write_batch(batch)
            except Exception:
                # Data loss accepted; next batch retries the connection
                # (reference behaviour, parca_reporter.go:1482-1484).
                log.error("destination %s write failed",
                          type(dest).__name__, exc_info=True)
                self.metrics.batch_errors += 1
        self.metrics.batches_written += 1
        self.metrics.last_flush_duration_s = time.monotonic() - t0
        return n_ingested

    @property
    def pending_count(self) -> int:
        with self._mu:
            return len(self._pending)


def build_arrow_record(samples: List[PendingSample]):
    """Assemble one Arrow v2 record batch from a flush batch."""
    from .arrow_v2 import SampleWriterV2

    writer = SampleWriterV2()
    for s in samples:
        writer.append_sample(
            s.trace, s.labels, s.value, s.timestamp_ns,
            sample_type=s.sample_type.sample_type,
            sample_unit=s.sample_type.sample_unit,
            period_type=s.sample_type.period_type,
            period_unit=s.sample_type.period_unit,
            period=s.period,
            duration_ns=s.duration_ns,
            temporality=s.sample_type.temporality,
        )
    return writer.build_record()
