"""OOM-kill reporting.

The reference integrates oomprof: eBPF-captured Go heap profiles shipped
on OOM kills via WriteRaw with job=oomprof labels (reference:
oom/oomprof.go:16-125). The native rebuild watches the kernel log
(/dev/kmsg) for oom-kill records and reports a `memory` origin sample
carrying the victim's last-known stack context: process identity,
anonymous/file RSS from the kill record, and — when the victim was
recently sampled by the CPU profiler — its most recent stack, giving the
flamegraph a "what was it doing when the kernel shot it" anchor. GPU
processes additionally surface HBM usage via their rings' last state.
"""

from __future__ import annotations

import logging
import os
import re
import threading
import time
from dataclasses import dataclass
from typing import Callable, Optional

from ..model import (
    Frame,
    FrameType,
    MappingFile,
    Trace,
    TraceEventMeta,
    TraceOrigin,
)

log = logging.getLogger("parca_agent_amd.oom")

# "Out of memory: Killed process 1234 (python3) total-vm:..kB,
#  anon-rss:..kB, file-rss:..kB, shmem-rss:..kB, UID:0 ..."
_OOM_RE = re.compile(
    r"Out of memory: Killed process (\d+) \(([^)]*)\)"
    r"(?:.*?total-vm:(\d+)kB)?(?:.*?anon-rss:(\d+)kB)?"
    r"(?:.*?file-rss:(\d+)kB)?")


@dataclass
class OOMKill:
    pid: int
    comm: str
    total_vm_kb: int = 0
    anon_rss_kb: int = 0
    file_rss_kb: int = 0
    timestamp_ns: int = 0


def parse_oom_kill(line: str) -> Optional[OOMKill]:
    m = _OOM_RE.search(line)
    if not m:
        return None
    return OOMKill(
        pid=int(m.group(1)),
        comm=m.group(2),
        total_vm_kb=int(m.group(3) or 0),
        anon_rss_kb=int(m.group(4) or 0),
        file_rss_kb=int(m.group(5) or 0),
    )


class OOMWatcher:
    def __init__(self, reporter,
                 last_stack_lookup: Optional[Callable] = None,
                 kmsg_path: str = "/dev/kmsg",
                 report_allocs: bool = False) -> None:
        self.reporter = reporter
        self.kmsg_path = kmsg_path
        # Reference semantics (flags.go:172, parca_reporter.go memory
        # origin): alloc_objects/alloc_space only with
        # --enable-oom-prof-allocs; inuse_* always.
        self.report_allocs = report_allocs
        # pid -> most recent Trace seen by the CPU sampler (optional).
        self.last_stack_lookup = last_stack_lookup
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.kills_reported = 0

    def start(self) -> None:
        self._stop.clear()
        self._thread = threading.Thread(target=self._run, name="oom-watch",
                                        daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=3)
            self._thread = None

    def _run(self) -> None:
        try:
            fd = os.open(self.kmsg_path, os.O_RDONLY | os.O_NONBLOCK)
        except OSError as e:
            log.warning("oom watcher: cannot open %s: %s",
                        self.kmsg_path, e)
            return
        # Seek to the end: only new records matter.
        try:
            os.lseek(fd, 0, os.SEEK_END)
        except OSError:
            pass
        try:
            while not self._stop.wait(0.5):
                while True:
                    try:
                        data = os.read(fd, 8192)
                    except BlockingIOError:
                        break
                    except OSError:
                        return
                    if not data:
                        break
                    self.handle_line(data.decode("utf-8", "replace"))
        finally:
            os.close(fd)

    def handle_line(self, line: str) -> None:
        kill = parse_oom_kill(line)
        if kill is None:
            return
        self.report(kill)

    def report(self, kill: OOMKill) -> None:
        # Real allocation profile when the victim ran under the
        # LD_PRELOAD heap sampler (oom/heap.py): the shm file outlives
        # the process.
        try:
            from .heap import heap_file_for, parse_heap_file, \
                report_heap_profile

            path = heap_file_for(kill.pid)
            prof = parse_heap_file(path)
            if prof is not None and prof.stacks:
                n = report_heap_profile(
                    self.reporter, prof, comm=kill.comm,
                    timestamp_ns=kill.timestamp_ns,
                    extra_labels=(("job", "oomprof"),),
                    report_allocs=self.report_allocs)
                log.info("reported %d heap samples for OOM-killed pid %d",
                         n, kill.pid)
            if prof is not None:
                try:
                    os.unlink(path)
                except OSError:
                    pass
        except Exception:
            log.debug("heap profile for pid %d unavailable", kill.pid,
                      exc_info=True)
        trace = None
        if self.last_stack_lookup is not None:
            try:
                trace = self.last_stack_lookup(kill.pid)
            except Exception:
                trace = None
        if trace is None:
            trace = Trace(frames=(Frame(
                kind=FrameType.ERROR, address=0,
                mapping=MappingFile(path="[oom]"),
                function_name=f"oom_killed:{kill.comm}"),))
        meta = TraceEventMeta(
            timestamp_ns=kill.timestamp_ns or time.time_ns(),
            comm=kill.comm, pid=kill.pid, tid=kill.pid,
            origin=TraceOrigin.MEMORY,
            value=kill.anon_rss_kb * 1024 or kill.total_vm_kb * 1024 or 1)
        self.reporter.report_trace_event(
            Trace(frames=trace.frames,
                  custom_labels=trace.custom_labels + (
                      ("job", "oom"),
                      ("oom_total_vm_kb", str(kill.total_vm_kb)))),
            meta)
        self.kills_reported += 1
        log.info("reported OOM kill of pid %d (%s), anon-rss %d kB",
                 kill.pid, kill.comm, kill.anon_rss_kb)
