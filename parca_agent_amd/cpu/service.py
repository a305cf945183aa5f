"""CPU sampler service: native perf events -> Traces -> Reporter.

The minimum end-to-end slice of SURVEY.md §7 stage 3: perf_event CPU
sampling at 19 Hz with frame-pointer unwinding, kernel stacks, process and
executable discovery, FileID hashing — feeding `Reporter.report_trace_event`
exactly like the fork's HandleTrace path feeds the reference's
arrowReporter (reference: main.go:596-607 -> parca_reporter.go:321).

Also performs the ktime->walltime conversion: perf timestamps are
CLOCK_MONOTONIC-domain; we keep a (realtime - monotonic) offset refreshed
periodically (reference: times.StartRealtimeSync, main.go:396-397).
"""

from __future__ import annotations

import logging
import threading
import time
from typing import Callable, List, Optional

from ..kallsyms import Kallsyms
from ..model import (
    Frame,
    FrameType,
    MappingFile,
    Trace,
    TraceEventMeta,
    TraceOrigin,
)
from ..procmaps import ExecutableCache, ProcessTable
from ..reporter.reporter import Reporter

log = logging.getLogger("parca_agent_amd.cpu")

KERNEL_MAPPING = MappingFile(path="[kernel.kallsyms]")
JIT_MAPPING = MappingFile(path="[jit]")


class ClockSync:
    """monotonic->realtime offset, refreshed on demand."""

    def __init__(self, refresh_interval: float = 180.0) -> None:
        self._interval = refresh_interval
        self._offset_ns = 0
        self._last = 0.0
        self._sync()

    def _sync(self) -> None:
        self._offset_ns = time.time_ns() - time.monotonic_ns()
        self._last = time.monotonic()

    def to_realtime(self, monotonic_ns: int) -> int:
        if time.monotonic() - self._last > self._interval:
            self._sync()
        return monotonic_ns + self._offset_ns


class CPUSamplerService:
    def __init__(
        self,
        reporter: Reporter,
        freq: int = 19,
        dwarf_stacks: bool = False,
        poll_interval: float = 0.1,
        ring_pages: int = 64,
        symbolize_kernel: bool = True,
        on_executable: Optional[Callable] = None,
        off_cpu_threshold: float = 0.0,
        target_pid: int = -1,
        unwinder=None,
        processes=None,
        executables=None,
        python_unwinder=None,
        ruby_unwinder=None,
        jvm_unwinder=None,
        php_unwinder=None,
    ) -> None:
        from ..native import sampler as native_sampler

        self.reporter = reporter
        self.freq = freq
        self.poll_interval = poll_interval
        self._native_mod = native_sampler()
        self._sampler = self._native_mod.PerfSampler(
            freq=freq, dwarf_stacks=dwarf_stacks, ring_pages=ring_pages,
            track_mmaps=True, target_pid=target_pid)
        self.processes = processes if processes is not None else ProcessTable()
        self.executables = executables if executables is not None else \
            ExecutableCache(load_symbols=False)
        self.clock = ClockSync()
        self.kallsyms = Kallsyms() if symbolize_kernel else None
        self.on_executable = on_executable
        self.unwinder = unwinder  # .eh_frame unwinder (cpu/unwind.py), optional
        if unwinder is not None and getattr(unwinder, "processes", 1) is None:
            unwinder.processes = self.processes
        # CPython interpreter unwinder (interp/python.py), optional.
        self.python_unwinder = python_unwinder
        self.ruby_unwinder = ruby_unwinder
        self.jvm_unwinder = jvm_unwinder
        self.php_unwinder = php_unwinder
        self.python_stacks = 0
        self.ruby_stacks = 0
        self.jvm_stacks = 0
        self.php_stacks = 0
        # Perl interpreter unwinder (interp/perl.py), optional.
        self.perl_unwinder = None
        self.perl_stacks = 0
        # Generic perf-map JIT symbolization (interp/perfmap.py).
        from ..interp.perfmap import PerfMapResolver

        self.perf_maps = PerfMapResolver()
        # Native custom labels: (pid, tid) -> labels published by
        # programs via include/parca_custom_labels.h (nativelabels.py).
        from ..nativelabels import NativeLabelReader

        self.native_labels = NativeLabelReader()
        self._seen_executables: set = set()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        # Converted-trace cache: identical raw stacks from the same
        # process skip frame resolution entirely (the trace-cache idea of
        # the reference, sized like main.go:682-703). Keyed by
        # (pid, kernel ips, user ips).
        from ..lru import LRU

        self._trace_cache: LRU = LRU(65536)
        # Frame interning: hot functions recur across most traces, so
        # distinct Frame objects number ~10^4-10^5 while the trace cache
        # would otherwise hold ~10^6 (65536 traces x stack depth) — the
        # difference is ~1 GB of agent RSS at full-machine scale.
        self._frame_intern: LRU = LRU(262144)
        # pid -> most recent Trace: the OOM watcher anchors kill reports
        # to what the victim was last seen doing (oom/watcher.py).
        self.last_trace_by_pid: LRU = LRU(4096, ttl_seconds=600)
        # Metrics
        self.samples_processed = 0
        self.samples_no_mapping = 0
        self.stacks_unwound_dwarf = 0
        self.trace_cache_hits = 0

    # -- lifecycle ---------------------------------------------------------

    def start(self) -> None:
        self._sampler.start()
        self._stop.clear()
        self._thread = threading.Thread(
            target=self._run, name="cpu-sampler", daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=10)
            self._thread = None
        self._sampler.stop()
        self._drain_once()

    @property
    def lost_samples(self) -> int:
        return self._sampler.lost

    @property
    def total_samples(self) -> int:
        return self._sampler.n_samples

    # -- drain loop --------------------------------------------------------

    def _run(self) -> None:
        next_sweep = time.monotonic() + 60.0
        while not self._stop.wait(self.poll_interval):
            try:
                self._drain_once()
                now = time.monotonic()
                if now >= next_sweep:
                    next_sweep = now + 60.0
                    # Janitor for label tables of crashed processes
                    # (exit events already handle the clean path).
                    self.native_labels.sweep()
            except Exception:
                log.error("sample drain failed", exc_info=True)

    def _drain_once(self) -> None:
        samples, proc_events = self._sampler.take()
        for ev in proc_events:
            self.processes.handle_proc_event(ev)
            if ev.kind == 2 and ev.pid == ev.tid:  # process exit
                self._drop_process_state(ev.pid)
        for s in samples:
            self._handle_sample(s)

    def _drop_process_state(self, pid: int) -> None:
        """Release per-pid state in every subsystem (the ProcessTable
        itself is handled by handle_proc_event): native unwinder mapping
        lists, interpreter-unwinder caches, trace anchors."""
        if self.unwinder is not None:
            self.unwinder.drop_process(pid)
        for interp in (self.python_unwinder, self.ruby_unwinder,
                       self.jvm_unwinder, self.php_unwinder,
                       self.perl_unwinder):
            if interp is not None and hasattr(interp, "drop_process"):
                interp.drop_process(pid)
        self.native_labels.forget(pid)

    # -- conversion --------------------------------------------------------

    def _handle_sample(self, s) -> None:
        if s.pid == 0:  # idle/swapper: skip like the reference's filter
            return

        user_ips = tuple(s.user_ips)
        # Frame-pointer walks of FP-less binaries stop after the sampled
        # IP (the kernel always emits at least that). When the FP chain
        # is implausibly short and we captured regs+stack, try the
        # .eh_frame unwinder and keep whichever recovered more frames.
        if self.unwinder is not None and s.regs and s.stack and \
                len(user_ips) < 4:
            unwound = self.unwinder.unwind(s.pid, s.regs, s.stack)
            if len(unwound) > len(user_ips):
                user_ips = tuple(unwound)
                self.stacks_unwound_dwarf += 1

        # CPython frames for interpreter processes (per-sample: the
        # python stack changes even when the native eval-loop stack does
        # not, so it cannot share the native trace cache).
        py_frames: tuple = ()
        if self.python_unwinder is not None and \
                self.python_unwinder.available:
            sp = s.regs[7] if s.regs else 0
            py_frames = tuple(self.python_unwinder.stack_for(
                s.pid, s.tid, sp))
            if py_frames:
                self.python_stacks += 1
        if not py_frames and self.ruby_unwinder is not None and \
                self.ruby_unwinder.available:
            py_frames = tuple(self.ruby_unwinder.stack_for(s.pid, s.tid))
            if py_frames:
                self.ruby_stacks += 1
        if not py_frames and self.jvm_unwinder is not None and \
                s.regs and self.jvm_unwinder.might_be_jvm(s.pid):
            stack = s.stack
            if stack:
                py_frames = tuple(self.jvm_unwinder.stack_for(
                    s.pid, s.regs[6], s.regs[7], stack))
                if py_frames:
                    self.jvm_stacks += 1
        if not py_frames and self.php_unwinder is not None and \
                self.php_unwinder.available:
            py_frames = tuple(self.php_unwinder.stack_for(s.pid, s.tid))
            if py_frames:
                self.php_stacks += 1
        if not py_frames and self.perl_unwinder is not None and \
                self.perl_unwinder.available:
            py_frames = tuple(self.perl_unwinder.stack_for(s.pid, s.tid))
            if py_frames:
                self.perl_stacks += 1

        kernel_ips = tuple(s.kernel_ips)
        # Key on 64-bit tuple hashes, not the tuples: retaining 65536
        # full ip tuples cost ~100 MB at capacity on busy nodes; the
        # collision probability across the cache is ~1e-10 per pair.
        cache_key = (s.pid, hash(kernel_ips), hash(user_ips))
        cached = self._trace_cache.get(cache_key)
        if cached is not None:
            trace, comm = cached
            self.trace_cache_hits += 1
        else:
            frames: List[Frame] = []
            proc = self.processes.ensure_maps(s.pid)
            for ip in kernel_ips:
                kf = self._frame_intern.get(ip)
                if kf is None:
                    name = self.kallsyms.lookup(ip) if self.kallsyms else None
                    kf = Frame(
                        kind=FrameType.KERNEL, address=ip,
                        mapping=KERNEL_MAPPING, function_name=name or "")
                    self._frame_intern.put(ip, kf)
                frames.append(kf)
            for ip in user_ips:
                frames.append(self._native_frame(proc, ip))
            if not frames:
                return
            # pprof order is leaf-first; kernel frames are innermost when
            # the interrupt hit kernel mode.
            trace = Trace(frames=tuple(frames))
            comm = proc.comm if proc else ""
            # Unknown frames may become resolvable shortly (perf-map JIT
            # entries appear as code is compiled): don't freeze them in
            # the cache.
            if not any(f.kind == FrameType.UNKNOWN for f in frames):
                self._trace_cache.put(cache_key, (trace, comm))

        if py_frames:
            # Interpreter frames sit between kernel frames (innermost)
            # and the native user stack: the logical python stack the
            # eval loop is executing.
            nk = len(kernel_ips)
            trace = Trace(frames=trace.frames[:nk] + py_frames +
                          trace.frames[nk:])

        custom = self.native_labels.labels_for(s.pid, s.tid)
        if custom:
            trace = Trace(frames=trace.frames, custom_labels=custom)

        meta = TraceEventMeta(
            timestamp_ns=self.clock.to_realtime(s.time_ns),
            comm=comm,
            pid=s.pid,
            tid=s.tid,
            cpu=s.cpu,
            origin=TraceOrigin.SAMPLING,
            value=1,
        )
        self.reporter.report_trace_event(trace, meta)
        self.last_trace_by_pid.put(s.pid, trace)
        self.samples_processed += 1

    def _native_frame(self, proc, ip: int) -> Frame:
        mapping = proc.find_mapping(ip) if proc else None
        if mapping is None or not mapping.path.startswith("/"):
            # JIT code lives in anonymous executable mappings; runtimes
            # following the perf-map convention (node --perf-basic-prof,
            # JVM perf-map-agent, ...) give us symbol names for them.
            if self.perf_maps is not None and proc is not None:
                name = self.perf_maps.lookup(proc.pid, ip)
                if name:
                    return Frame(kind=FrameType.JIT, address=ip,
                                 mapping=JIT_MAPPING, function_name=name)
            self.samples_no_mapping += 1
            path = mapping.path if mapping else ""
            return Frame(kind=FrameType.UNKNOWN, address=ip,
                         mapping=MappingFile(path=path) if path else None)
        info = self.executables.get(mapping.path)
        if info.error:
            return Frame(kind=FrameType.NATIVE, address=ip,
                         mapping=MappingFile(path=mapping.path))
        self._maybe_report_executable(info)
        addr = info.normalize(ip, mapping.start, mapping.file_offset)
        key = (info.file_id, addr)
        f = self._frame_intern.get(key)
        if f is None:
            f = Frame(
                kind=FrameType.NATIVE,
                address=addr,
                mapping=MappingFile(
                    file_id=info.file_id, path=mapping.path,
                    build_id=info.build_id or ""),
            )
            self._frame_intern.put(key, f)
        return f

    def _maybe_report_executable(self, info) -> None:
        if self.on_executable is None or info.file_id in self._seen_executables:
            return
        self._seen_executables.add(info.file_id)
        try:
            self.on_executable(info)
        except Exception:
            log.warning("on_executable callback failed", exc_info=True)
