"""Off-CPU profiling service: wallclock/nanoseconds samples of blocked
time.

Reference semantics (fork StartOffCPUProfiling + arrowReporter off-cpu
origin, parca_reporter.go:394-397; threshold flag main.go:534-539):
scheduling-out events are sampled with probability `threshold`, the
blocking stack is captured at switch-out, and the reported value is the
blocked duration in nanoseconds.

Mechanics here: the native OffCpuSampler samples every Nth context
switch (N = 1/threshold) WITH a callchain, and records every switch
in/out cheaply. This service pairs OUT->IN per tid for durations and
attaches the sampled stack captured at that OUT instant.
"""

from __future__ import annotations

import logging
import threading
import time
from typing import Dict, Optional, Tuple

from ..model import (
    Frame,
    FrameType,
    Trace,
    TraceEventMeta,
    TraceOrigin,
)
from ..procmaps import ExecutableCache, ProcessTable
from ..symbolize import FrameResolver
from ..kallsyms import Kallsyms
from .service import KERNEL_MAPPING, ClockSync

log = logging.getLogger("parca_agent_amd.offcpu")

# A sampled switch-out stack is married to the OUT record whose timestamp
# is within this window (they describe the same event).
MATCH_WINDOW_NS = 1_000_000


class OffCPUService:
    def __init__(self, reporter, threshold: float,
                 poll_interval: float = 0.25,
                 processes: Optional[ProcessTable] = None,
                 executables: Optional[ExecutableCache] = None,
                 min_block_ns: int = 1_000) -> None:
        if not 0.0 < threshold <= 1.0:
            raise ValueError("off-cpu threshold must be in (0, 1]")
        from ..native import sampler as native_sampler

        period = max(1, int(round(1.0 / threshold)))
        self._sampler = native_sampler().OffCpuSampler(
            sample_period=period, ring_pages=64)
        self.reporter = reporter
        self.poll_interval = poll_interval
        self.min_block_ns = min_block_ns
        self.processes = processes or ProcessTable()
        self.executables = executables or ExecutableCache()
        self.resolver = FrameResolver(self.processes, self.executables)
        # Blocked-time profiles get the same per-thread custom labels
        # as on-CPU samples (include/parca_custom_labels.h).
        from ..nativelabels import NativeLabelReader

        self.native_labels = NativeLabelReader()
        self.kallsyms = Kallsyms()
        self.clock = ClockSync()
        # tid -> (out_time, sampled stack or None)
        self._blocked: Dict[int, Tuple[int, Optional[object]]] = {}
        # recently sampled stacks keyed by tid
        self._pending_stacks: Dict[int, object] = {}
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.samples_emitted = 0

    def start(self) -> None:
        self._sampler.start()
        self._stop.clear()
        self._thread = threading.Thread(target=self._run, name="offcpu",
                                        daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=5)
            self._thread = None
        self._sampler.stop()
        self.drain_once()

    @property
    def lost(self) -> int:
        return self._sampler.lost

    def _run(self) -> None:
        while not self._stop.wait(self.poll_interval):
            try:
                self.drain_once()
            except Exception:
                log.error("off-cpu drain failed", exc_info=True)

    def drain_once(self) -> int:
        samples, switches = self._sampler.take()
        # One time-ordered stream: a batch can contain MANY block/unblock
        # cycles per tid, so each sampled stack must be consumed by the
        # switch-out it belongs to, not by the last one in the batch.
        stream = [(s.time_ns, 0, s) for s in samples if s.pid != 0]
        stream += [(e.time_ns, 1, e) for e in switches if e.pid != 0]
        stream.sort(key=lambda t: (t[0], t[1]))
        emitted = 0
        for _ts, kind, ev in stream:
            if kind == 0:
                self._pending_stacks[ev.tid] = ev
                continue
            if ev.is_out:
                stack = self._pending_stacks.pop(ev.tid, None)
                if stack is not None and \
                        abs(stack.time_ns - ev.time_ns) > MATCH_WINDOW_NS:
                    stack = None  # stale sample from an earlier switch
                self._blocked[ev.tid] = (ev.time_ns, stack)
            else:
                entry = self._blocked.pop(ev.tid, None)
                if entry is None:
                    continue
                out_ns, stack = entry
                duration = ev.time_ns - out_ns
                if stack is None or duration < self.min_block_ns:
                    continue
                self._emit(stack, duration)
                emitted += 1
        # Bound the blocked map (tids that never come back).
        if len(self._blocked) > 65536:
            self._blocked.clear()
        if len(self._pending_stacks) > 65536:
            self._pending_stacks.clear()
        return emitted

    def _emit(self, s, duration_ns: int) -> None:
        frames = []
        proc = self.processes.ensure_maps(s.pid)
        for ip in s.user_ips:
            frames.append(self.resolver.resolve(s.pid, ip))
        kernel = []
        for ip in s.kernel_ips:
            name = self.kallsyms.lookup(ip) if self.kallsyms else None
            kernel.append(Frame(kind=FrameType.KERNEL, address=ip,
                                mapping=KERNEL_MAPPING,
                                function_name=name or ""))
        ordered = tuple(kernel + frames)
        if not ordered:
            return
        custom = self.native_labels.labels_for(s.pid, s.tid)
        meta = TraceEventMeta(
            timestamp_ns=self.clock.to_realtime(s.time_ns),
            comm=proc.comm if proc else "",
            pid=s.pid, tid=s.tid, cpu=s.cpu,
            origin=TraceOrigin.OFF_CPU, value=duration_ns)
        self.reporter.report_trace_event(
            Trace(frames=ordered, custom_labels=custom), meta)
        self.samples_emitted += 1
