"""User-defined paired uprobes: YAML-declared entry/exit timing.

The reference attaches eBPF uprobe pairs per YAML spec, tracks per-tid
outermost-scope depth in the kernel, and emits backdated OTel spans
(reference: probes/config.go:17-41, probes/bpf/probe.bpf.c:85-154,
probes/service.go:146-215). This implementation uses the perf "uprobe"
PMU (no BPF needed): entry and return probes fire timestamped events,
and the depth/min-duration logic of probe.bpf.c runs in the agent.
Completed outer-scope invocations are reported as `probe/nanoseconds`
trace events carrying the probe name as the leaf frame — queryable in
Parca alongside the sampling profiles (span export can layer on top).

YAML format (probes.yaml):
    probes:
      - name: my_op
        binary: ".*/myservice"     # regex on executable path
        path: /usr/bin/myservice   # or an explicit path
        symbol: process_request    # symbol to instrument
        offset: 0x1234             # alternative to symbol
        min_duration: 1ms          # suppress shorter invocations
        main_thread_only: false
"""

from __future__ import annotations

import logging
import re
import threading
import time
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

import yaml

from ..elf import ELFFile
from ..flags import _parse_duration
from ..model import Frame, FrameType, MappingFile, Trace, TraceEventMeta, TraceOrigin

log = logging.getLogger("parca_agent_amd.probes")


@dataclass
class ProbeSpec:
    name: str
    path: str = ""
    binary: str = ""  # regex matched against discovered executables
    symbol: str = ""
    offset: int = 0
    min_duration_ns: int = 0
    main_thread_only: bool = False
    # Kernel-side flood protection: above this sustained fire rate the
    # probe pair's perf events are disabled for a 1 s cooldown (the
    # reference filters floods in-kernel, probe.bpf.c:114-154; we gate
    # the events themselves). 0 disables the limit.
    max_events_per_sec: int = 5000
    spec_id: int = 0

    def resolve_offset(self, path: str) -> Optional[int]:
        """Symbol -> file offset of the probe point (uprobe offsets are
        file offsets, not vaddrs)."""
        if self.offset:
            return self.offset
        try:
            with ELFFile.open(path) as elf:
                for sym in elf.symbols():
                    if sym.name == self.symbol and sym.is_function:
                        off = elf.file_offset_for_vaddr(sym.value)
                        if off is not None:
                            return off
        except (OSError, ValueError):
            return None
        return None


def load_probe_config(path: str) -> List[ProbeSpec]:
    with open(path) as fh:
        doc = yaml.safe_load(fh) or {}
    return parse_probe_config(doc)


def parse_probe_config(doc: dict) -> List[ProbeSpec]:
    out = []
    for i, raw in enumerate(doc.get("probes") or []):
        if not raw.get("name"):
            raise ValueError("probe entry missing name")
        if not raw.get("symbol") and not raw.get("offset"):
            raise ValueError(f"probe {raw['name']}: symbol or offset required")
        if not raw.get("path") and not raw.get("binary"):
            raise ValueError(f"probe {raw['name']}: path or binary required")
        min_dur = raw.get("min_duration", 0)
        if isinstance(min_dur, str):
            min_dur_ns = int(_parse_duration(min_dur) * 1e9)
        else:
            min_dur_ns = int(min_dur)
        out.append(ProbeSpec(
            name=raw["name"],
            path=raw.get("path", ""),
            binary=raw.get("binary", ""),
            symbol=raw.get("symbol", ""),
            offset=int(raw.get("offset", 0)),
            min_duration_ns=min_dur_ns,
            main_thread_only=bool(raw.get("main_thread_only", False)),
            max_events_per_sec=int(raw.get("max_events_per_sec", 5000)),
            spec_id=i,
        ))
    return out


@dataclass
class _TidState:
    depth: int = 0
    entry_ns: int = 0


class ProbesService:
    def __init__(self, specs: List[ProbeSpec], reporter,
                 poll_interval: float = 0.1, span_exporter=None) -> None:
        self.specs = specs
        self.reporter = reporter
        self.poll_interval = poll_interval
        # Optional OTLP span export (the reference streams probe fires as
        # backdated spans, probes/service.go:180-199); profile samples
        # are emitted either way.
        self.span_exporter = span_exporter
        from ..nativelabels import NativeLabelReader

        self.native_labels = NativeLabelReader()
        self._group = None
        self._attached: List[Tuple[ProbeSpec, str]] = []
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        # (spec_id, tid) -> depth state (probe.bpf.c scope_stacks analog)
        self._tids: Dict[Tuple[int, int], _TidState] = {}
        self.fires_seen = 0
        self.spans_emitted = 0
        self.clock_offset_ns = time.time_ns() - time.monotonic_ns()

    @classmethod
    def from_config_file(cls, path: str, reporter) -> "ProbesService":
        return cls(load_probe_config(path), reporter)

    def start(self) -> None:
        from ..native import sampler as native_sampler

        native_specs = []
        for spec in self.specs:
            if not spec.path:
                continue  # regex-only specs attach via on_executable
            off = spec.resolve_offset(spec.path)
            if off is None:
                log.warning("probe %s: cannot resolve %s in %s",
                            spec.name, spec.symbol or hex(spec.offset),
                            spec.path)
                continue
            native_specs.append((spec.path, off, False, spec.spec_id * 2,
                                 spec.max_events_per_sec))
            native_specs.append((spec.path, off, True, spec.spec_id * 2 + 1,
                                 spec.max_events_per_sec))
            self._attached.append((spec, spec.path))
        if not native_specs:
            log.info("probes: nothing to attach")
            return
        self._group = native_sampler().UprobeGroup(native_specs)
        self._group.start()
        log.info("probes: attached %d uprobe pairs (%d perf events)",
                 len(self._attached), self._group.n_events)
        self._stop.clear()
        self._thread = threading.Thread(target=self._run, name="probes",
                                        daemon=True)
        self._thread.start()

    def on_executable(self, path: str) -> None:
        """Executable-discovery hook: attach regex probes to newly seen
        binaries (reference: probes OnExecutable, attach.go:51-93)."""
        for spec in self.specs:
            if spec.path or not spec.binary:
                continue
            if re.search(spec.binary, path):
                spec.path = path
        # Attachment of late-bound specs requires a restart of the group;
        # keep it simple: restart when new paths were bound.
        if self._group is not None and any(
                s.path and (s, s.path) not in self._attached
                for s in self.specs):
            self.stop()
            self.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=5)
            self._thread = None
        if self._group is not None:
            self._group.stop()
            self._group = None

    # -- event pairing -----------------------------------------------------

    def _run(self) -> None:
        while not self._stop.wait(self.poll_interval):
            try:
                self.drain_once()
            except Exception:
                log.error("probe drain failed", exc_info=True)

    def drain_once(self) -> int:
        if self._group is None:
            return 0
        fires = self._group.take()
        if not fires:
            return 0
        # Per-CPU streams are time-ordered but the merged list is not:
        # order globally by timestamp before pairing. Caveat vs the
        # reference's in-kernel depth counter (probe.bpf.c): perf clocks
        # have small cross-CPU skew, so an exit and the next entry that
        # are only microseconds apart can occasionally be observed
        # reordered when the thread migrated between them; the pairing
        # then merges the two invocations into one longer span. Sampled
        # durations remain correct to within the skew.
        fires.sort(key=lambda f: f.time_ns)
        for f in fires:
            self._handle(f)
        self.fires_seen += len(fires)
        return len(fires)

    def _handle(self, f) -> None:
        spec = self.specs[f.probe_id // 2]
        if spec.main_thread_only and f.pid != f.tid:
            return
        key = (spec.spec_id, f.tid)
        st = self._tids.get(key)
        if not f.is_return:
            if st is None:
                st = _TidState()
                self._tids[key] = st
            st.depth += 1
            if st.depth == 1:
                st.entry_ns = f.time_ns
        else:
            if st is None or st.depth == 0:
                return  # return without entry (attached mid-call)
            st.depth -= 1
            if st.depth == 0:
                duration = f.time_ns - st.entry_ns
                del self._tids[key]
                if duration >= spec.min_duration_ns:
                    self._emit(spec, f, duration)

    def _emit(self, spec: ProbeSpec, f, duration_ns: int) -> None:
        frames = (Frame(
            kind=FrameType.NATIVE, address=0,
            mapping=MappingFile(path=spec.path),
            function_name=spec.symbol or spec.name),)
        meta = TraceEventMeta(
            # Backdate to the entry like the reference's span start
            # (service.go:173-199); perf time is CLOCK_MONOTONIC-domain.
            timestamp_ns=f.time_ns - duration_ns + self.clock_offset_ns,
            pid=f.pid, tid=f.tid, cpu=f.cpu,
            origin=TraceOrigin.PROBE, value=duration_ns)
        custom = (("probe", spec.name),) + \
            self.native_labels.labels_for(f.pid, f.tid)
        self.reporter.report_trace_event(
            Trace(frames=frames, custom_labels=custom), meta)
        if self.span_exporter is not None:
            from ..otlp import Span

            start = f.time_ns - duration_ns + self.clock_offset_ns
            self.span_exporter.add(Span(
                name=spec.name, start_ns=start, end_ns=start + duration_ns,
                attributes={"pid": str(f.pid), "tid": str(f.tid),
                            "symbol": spec.symbol}))
        self.spans_emitted += 1
