"""GPU profiler service: drains per-process shm rings, dispatches events,
reports GPU kernel-time and PC-sample traces.

The orchestration analog of parcagpu.Start (reference:
parcagpu/parcagpu.go:69-216): a drain thread reads every live ring,
dispatches on the u32 event tag, batches kernel timings (×100,
parcagpu.go:96), runs the correlation fixer, accumulates PC samples into
bucket histograms (device kernel when a GPU is visible to the agent, numpy
otherwise), and reports completed traces via the shared Reporter. Fixer GC
runs every 2 s (parcagpu.go:126-130); drop/error counters surface as
rocm.* metrics.
"""

from __future__ import annotations

import glob
import logging
import os
import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import numpy as np

from ..model import (
    Frame,
    FrameType,
    Trace,
    TraceEventMeta,
    TraceOrigin,
)
from ..procmaps import ExecutableCache, ProcessTable
from ..symbolize import FrameResolver
from . import events as ev
from .codeobj import CodeObjectRegistry
from .fixer import CompletedKernel, GpuTraceFixer
from .pcbuckets import BucketLayout, DeviceAccumulator, HostAccumulator

log = logging.getLogger("parca_agent_amd.gpu")

KERNEL_BATCH_SIZE = 100  # parcagpu.go:96


@dataclass
class GpuServiceMetrics:
    rings_open: int = 0
    ring_dropped: int = 0          # rocm.errors.ringbuf_full analog
    events_by_type: Dict[int, int] = field(default_factory=dict)
    pc_samples: int = 0
    kernels_reported: int = 0
    pc_buckets_reported: int = 0
    tool_errors: int = 0


class _RingState:
    def __init__(self, consumer, pid: int) -> None:
        self.consumer = consumer
        self.pid = pid
        self.fixer = GpuTraceFixer()
        self.kernel_batch: List[ev.KernelDispatch] = []
        self.kernel_names: Dict[int, str] = {}  # kernel_id -> name
        self.kernel_code_objects: Dict[int, int] = {}  # kernel_id -> co_id
        self.gpu_config: Dict[int, ev.GpuConfig] = {}
        self.comm: str = ""
        self.last_dropped = 0
        # Duty-cycled tracing scale factor ((on+off)/on), 1.0 = always on.
        self.duty_scale = 1.0
        # kernel_id -> (Frame, kernel-only Trace): dispatch rates reach
        # tens of kHz, so per-dispatch frame construction is cached.
        self.kernel_traces: Dict[int, tuple] = {}
        # correlation_id -> resolved host-frame tuple (launch stacks
        # repeat across graph replays).
        self.stack_frames: Dict[int, tuple] = {}


class GPUProfilerService:
    def __init__(
        self,
        reporter,
        shm_dir: str = "/dev/shm",
        poll_interval: float = 0.1,
        pc_flush_interval: float = 5.0,
        bucket_shift: int = 6,
        use_device_bucketize: Optional[bool] = None,
        processes: Optional[ProcessTable] = None,
        executables: Optional[ExecutableCache] = None,
        on_executable=None,
        on_code_object=None,
        clock_offset_ns: Optional[int] = None,
    ) -> None:
        from ..native import gpu as native_gpu

        self._native = native_gpu()
        self.reporter = reporter
        self.shm_dir = shm_dir
        self.poll_interval = poll_interval
        self.pc_flush_interval = pc_flush_interval
        self.metrics = GpuServiceMetrics()

        self.processes = processes or ProcessTable()
        self.executables = executables or ExecutableCache()
        self.resolver = FrameResolver(self.processes, self.executables,
                                      on_executable=on_executable)
        # Native custom labels joined onto GPU samples by launcher tid
        # (include/parca_custom_labels.h + nativelabels.py): labels are
        # read at drain time, so a label changed between launch and
        # drain may attribute a few trailing kernels to the new value —
        # acceptable skew for scope-style labels.
        from ..nativelabels import NativeLabelReader

        self.native_labels = NativeLabelReader()
        self.code_objects = CodeObjectRegistry(on_executable=on_code_object)
        self.layout = BucketLayout(bucket_shift=bucket_shift)
        # Device detection via /dev/kfd, NOT hipGetDeviceCount: the
        # first HIP call initializes the whole ROCm runtime in the
        # AGENT (several hundred MB of RSS). The DeviceAccumulator is
        # therefore constructed LAZILY on the first PC-sample batch —
        # exactly when there is device work to do — and the agent stays
        # HIP-free while the driver exposes no PC sampling.
        if use_device_bucketize is None:
            use_device_bucketize = os.path.exists("/dev/kfd")
        self.accumulator = HostAccumulator(self.layout)
        self.device_bucketize = use_device_bucketize
        self._device_pending = use_device_bucketize

        # rocprofiler timestamps are CLOCK_BOOTTIME-domain nanoseconds;
        # convert to walltime for report timestamps.
        if clock_offset_ns is None:
            clock_offset_ns = time.time_ns() - time.clock_gettime_ns(
                time.CLOCK_BOOTTIME)
        self.clock_offset_ns = clock_offset_ns

        # Hardware metrics (utilization/VRAM/power), only where a GPU and
        # rocm-smi exist; scrape-driven via metrics.AgentCollector.
        # (rocm-smi subprocess — no in-agent HIP init.)
        self.hw_metrics = None
        if os.path.exists("/dev/kfd"):
            try:
                from .hwmetrics import GpuHwMetrics

                self.hw_metrics = GpuHwMetrics()
            except Exception:
                pass

        self._rings: Dict[int, _RingState] = {}
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._last_gc = 0.0
        self._last_pc_flush = time.monotonic()

    # -- lifecycle ---------------------------------------------------------

    def start(self) -> None:
        self._stop.clear()
        self._thread = threading.Thread(
            target=self._run, name="gpu-drain", daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=10)
            self._thread = None
        self.drain_once()
        # Finalize: emit every pending kernel timing (stack or not) so
        # short-lived sessions lose nothing.
        for state in self._rings.values():
            state.fixer.max_age = 0.0
            for done in state.fixer.clear_stale():
                self._report_kernel(state, done)
        self.flush_pc()

    def _run(self) -> None:
        while not self._stop.wait(self.poll_interval):
            try:
                self.drain_once()
            except Exception:
                log.error("gpu drain failed", exc_info=True)
            now = time.monotonic()
            if now - self._last_gc > 2.0:
                self._last_gc = now
                self._gc()
            if now - self._last_pc_flush > self.pc_flush_interval:
                self._last_pc_flush = now
                try:
                    self.flush_pc()
                except Exception:
                    log.error("pc flush failed", exc_info=True)

    # -- ring discovery ----------------------------------------------------

    def scan_rings(self) -> None:
        for path in glob.glob(os.path.join(self.shm_dir, "parca_gpu_*.ring")):
            try:
                pid = int(os.path.basename(path)[len("parca_gpu_"):-len(".ring")])
            except ValueError:
                continue
            if pid in self._rings:
                continue
            try:
                consumer = self._native.RingConsumer(path)
            except RuntimeError:
                continue  # producer still initializing
            self._rings[pid] = _RingState(consumer, pid)
            log.info("gpu ring attached: pid=%d", pid)
        # Reap rings of dead processes after one final drain.
        for pid in list(self._rings):
            if not os.path.exists(f"/proc/{pid}"):
                self._drain_ring(self._rings[pid])
                path = os.path.join(self.shm_dir, f"parca_gpu_{pid}.ring")
                try:
                    os.unlink(path)
                except OSError:
                    pass
                self._finalize_ring(self._rings.pop(pid))
        self.metrics.rings_open = len(self._rings)

    def _finalize_ring(self, state: _RingState) -> None:
        for done in state.fixer.add_times(state.kernel_batch):
            self._report_kernel(state, done)
        state.kernel_batch.clear()
        for done in state.fixer.clear_stale():
            self._report_kernel(state, done)
        self.code_objects.drop_process(state.pid)
        if hasattr(self.accumulator, "drop_process"):
            self.accumulator.drop_process(state.pid)

    # -- drain -------------------------------------------------------------

    def drain_once(self) -> int:
        self.scan_rings()
        n = 0
        for state in list(self._rings.values()):
            n += self._drain_ring(state)
        return n

    def _drain_ring(self, state: _RingState) -> int:
        total = 0
        # Bounded per visit so one firehose ring cannot starve the rest;
        # the poll loop returns to it immediately. Stackless kernel
        # dispatches (the hot path) are pre-aggregated in C++ — Python
        # sees O(unique kernels) rows, not O(dispatches) records.
        for _ in range(64):
            (others, k_ids, gpus, tids, totals, counts,
             last_ends) = state.consumer.drain_batched(8192)
            n_agg = int(counts.sum()) if len(counts) else 0
            if not others and n_agg == 0:
                break
            total += len(others) + n_agg
            for rtype, payload in others:
                self._dispatch(state, rtype, payload)
            if len(k_ids):
                self._report_agg_rows(state, k_ids, gpus, tids, totals,
                                      counts, last_ends)
        dropped = state.consumer.dropped
        if dropped > state.last_dropped:
            self.metrics.ring_dropped += dropped - state.last_dropped
            state.last_dropped = dropped
        if state.kernel_batch:
            self._report_kernels(state,
                                 state.fixer.add_times(state.kernel_batch))
            state.kernel_batch.clear()
        return total

    def _dispatch(self, state: _RingState, rtype: int, payload: bytes) -> None:
        m = self.metrics.events_by_type
        m[rtype] = m.get(rtype, 0) + 1
        if rtype == ev.EV_KERNEL_DISPATCH:
            d = ev.decode_kernel_dispatch(payload)
            state.kernel_batch.append(d)
            if len(state.kernel_batch) >= KERNEL_BATCH_SIZE:
                self._report_kernels(
                    state, state.fixer.add_times(state.kernel_batch))
                state.kernel_batch.clear()
        elif rtype == ev.EV_LAUNCH_STACK:
            s = ev.decode_launch_stack(payload)
            for done in state.fixer.add_stack(s):
                self._report_kernel(state, done)
        elif rtype == ev.EV_PC_SAMPLE_BATCH:
            gpu_index, samples = ev.decode_pc_sample_batch(payload)
            self.metrics.pc_samples += len(samples)
            if self._device_pending:
                self._device_pending = False
                try:
                    dev = DeviceAccumulator(self.layout)
                    # fold anything the host accumulator gathered in the
                    # race window into the device pending arrays
                    for g, (h, l) in self.accumulator.read(True).items():
                        dp = dev._pending_arrays(g)
                        dp[0][: len(h)] += h
                        dp[1][: len(l)] += l
                    self.accumulator = dev
                    log.info("device bucketize initialized (first PC "
                             "samples arrived)")
                except Exception:
                    log.warning("device bucketize unavailable; staying "
                                "on host accumulator", exc_info=True)
            self.accumulator.accumulate(state.pid, samples, gpu=gpu_index)
        elif rtype == ev.EV_CODE_OBJECT_LOAD:
            load = ev.decode_code_object_load(payload)
            info = self.code_objects.load(state.pid, load)
            if self.layout.add(state.pid, load.code_object_id,
                               load.load_size):
                if isinstance(self.accumulator, DeviceAccumulator):
                    self.accumulator.layout_changed()
        elif rtype == ev.EV_CODE_OBJECT_UNLOAD:
            co_id = ev.decode_code_object_unload(payload)
            self.code_objects.unload(state.pid, co_id)
        elif rtype == ev.EV_KERNEL_SYMBOL:
            sym = ev.decode_kernel_symbol(payload)
            state.kernel_names[sym.kernel_id] = demangle_kernel(sym.name)
            state.kernel_code_objects[sym.kernel_id] = sym.code_object_id
            state.kernel_traces.pop(sym.kernel_id, None)
        elif rtype == ev.EV_GPU_CONFIG:
            cfg = ev.decode_gpu_config(payload)
            if cfg.method == 100:
                # Duty-cycle advertisement: scale observed kernel time by
                # (on+off)/on so pprof totals estimate full wall coverage.
                state.duty_scale = max(cfg.ns_per_sample, 1.0)
                return
            state.gpu_config[cfg.gpu_index] = cfg
            self.reporter.set_gpu_config(state.pid, cfg.gpu_index,
                                         cfg.ns_per_sample)
            self.reporter.set_gpu_config(state.pid, -1, cfg.ns_per_sample)
        elif rtype == ev.EV_ERROR:
            err = ev.decode_error(payload)
            self.metrics.tool_errors += 1
            log.warning("gpu tool error (pid %d, code %d): %s",
                        state.pid, err.code, err.message)

    # -- GC ----------------------------------------------------------------

    def _gc(self) -> None:
        for state in self._rings.values():
            for done in state.fixer.clear_stale():
                self._report_kernel(state, done)

    # -- reporting ---------------------------------------------------------

    def _kernel_frame(self, state: _RingState, kernel_id: int) -> Frame:
        name = state.kernel_names.get(kernel_id, f"kernel_{kernel_id}")
        co_id = state.kernel_code_objects.get(kernel_id)
        mapping = None
        if co_id is not None:
            info = self.code_objects.get(state.pid, co_id)
            if info is not None:
                mapping = info.mapping_file
        return Frame(kind=FrameType.GPU_KERNEL, address=0, mapping=mapping,
                     function_name=name)

    def _report_agg_rows(self, state: _RingState, k_ids, gpus, tids,
                         totals, counts, last_ends) -> None:
        """Report C++-pre-aggregated stackless dispatches: one sample
        per (kernel, gpu, tid) row carrying the summed duration."""
        n = int(counts.sum())
        m = self.metrics.events_by_type
        m[ev.EV_KERNEL_DISPATCH] = m.get(ev.EV_KERNEL_DISPATCH, 0) + n
        for i in range(len(k_ids)):
            end = int(last_ends[i])
            d = ev.KernelDispatch(
                correlation_id=0, dispatch_id=0, kernel_id=int(k_ids[i]),
                start_ns=end, end_ns=end, tid=int(tids[i]),
                gpu_index=int(gpus[i]), pid=state.pid,
                grid=(0, 0, 0), workgroup=(0, 0, 0),
                private_segment_size=0, group_segment_size=0)
            self._report_kernel(state, CompletedKernel(dispatch=d,
                                                       stack=None),
                                total_ns=int(totals[i]))

    def _report_kernels(self, state: _RingState,
                        completed: List[CompletedKernel]) -> None:
        """Batch-level pre-aggregation: eager-mode workloads dispatch the
        same kernel from the same launch site thousands of times per
        second; summing durations per (kernel, launch-stack, gpu) within
        one processing batch collapses per-event reporter work by the
        repeat factor while the pprof sum stays identical."""
        if len(completed) == 1:
            self._report_kernel(state, completed[0])
            return
        groups: Dict[tuple, CompletedKernel] = {}
        sums: Dict[tuple, int] = {}
        for done in completed:
            d = done.dispatch
            key = (d.kernel_id,
                   done.stack.correlation_id if done.stack else 0,
                   d.gpu_index, d.tid)
            if key in sums:
                sums[key] += d.duration_ns
            else:
                sums[key] = d.duration_ns
                groups[key] = done
        for key, done in groups.items():
            self._report_kernel(state, done, total_ns=sums[key])

    def _report_kernel(self, state: _RingState, done: CompletedKernel,
                       total_ns: Optional[int] = None) -> None:
        d = done.dispatch
        cached = state.kernel_traces.get(d.kernel_id)
        if cached is None:
            kframe = self._kernel_frame(state, d.kernel_id)
            cached = (kframe, Trace(frames=(kframe,)))
            state.kernel_traces[d.kernel_id] = cached
        kframe, kernel_only_trace = cached

        if done.stack is not None:
            host = state.stack_frames.get(done.stack.correlation_id)
            if host is None:
                host = tuple(self.resolver.resolve(state.pid, ip)
                             for ip in done.stack.ips)
                if len(state.stack_frames) > 65536:
                    state.stack_frames.clear()
                state.stack_frames[done.stack.correlation_id] = host
            trace = Trace(frames=(kframe,) + host)
        else:
            trace = kernel_only_trace
        value = total_ns if total_ns is not None else d.duration_ns
        if state.duty_scale != 1.0:
            value = int(value * state.duty_scale)
        custom = self.native_labels.labels_for(state.pid, d.tid)
        if custom:
            trace = Trace(frames=trace.frames, custom_labels=custom)
        meta = TraceEventMeta(
            timestamp_ns=d.end_ns + self.clock_offset_ns,
            pid=state.pid,
            tid=d.tid,
            origin=TraceOrigin.GPU_KERNEL,
            value=value,
            gpu_id=d.gpu_index,
            kernel_name=state.kernel_names.get(d.kernel_id, ""),
        )
        self.reporter.report_trace_event(trace, meta)
        self.metrics.kernels_reported += 1

    def flush_pc(self) -> None:
        """Emit accumulated PC buckets as gpu_pcsample traces, one
        stream per source GPU (per-GPU fan-out for the daemon shape)."""
        if self.layout.total_buckets == 0:
            return
        for gpu, (hist, lane_hist) in sorted(
                self.accumulator.read(True).items()):
            self._flush_pc_gpu(gpu, hist, lane_hist)

    def _flush_pc_gpu(self, gpu: int, hist: np.ndarray,
                      lane_hist: np.ndarray) -> None:
        nonzero = np.nonzero(hist)[0]
        if len(nonzero) == 0:
            return
        offsets = self.layout.offsets()
        now_ns = time.time_ns()
        for bucket in nonzero:
            slot = int(np.searchsorted(offsets, bucket, side="right")) - 1
            pid, co_id = self.layout.key_of_slot(slot)
            info = self.code_objects.get(pid, co_id)
            first, _ = self.layout.bucket_range(slot)
            co_offset = (int(bucket) - first) << self.layout.bucket_shift
            if info is not None:
                mapping = info.mapping_file
                address = info.vaddr_for_offset(co_offset)
                func = info.symbolize(co_offset)
            else:
                from ..model import MappingFile

                mapping = MappingFile(path=f"codeobj-{co_id}")
                address = co_offset
                func = ""
            frames = (Frame(kind=FrameType.GPU_PC, address=address,
                            mapping=mapping, function_name=func),)
            meta = TraceEventMeta(
                timestamp_ns=now_ns,
                pid=pid,
                origin=TraceOrigin.GPU_PC,
                value=int(hist[bucket]),
                gpu_id=gpu,
            )
            # Wave-occupancy view: mean active lanes (of 64) at this PC,
            # from the exec-mask popcounts the bucketize kernel sums —
            # divergence shows up as avg_active_lanes << 64.
            labels = ()
            if lane_hist[bucket] > 0:
                avg_lanes = int(round(
                    int(lane_hist[bucket]) / int(hist[bucket])))
                labels = (("avg_active_lanes", str(avg_lanes)),)
            self.reporter.report_trace_event(
                Trace(frames=frames, custom_labels=labels), meta)
            self.metrics.pc_buckets_reported += 1


def demangle_kernel(name: str) -> str:
    """Light-touch demangling of kernel symbol names: strip the .kd
    suffix; full C++ demangling is left to the Parca server."""
    if name.endswith(".kd"):
        name = name[:-3]
    return name
