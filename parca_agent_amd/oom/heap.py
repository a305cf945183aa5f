"""Post-mortem reader for the LD_PRELOAD heap sampler's shm files.

csrc/heap/heap_preload.c keeps a stack-aggregated allocation profile in
/dev/shm/parca_heap_<pid>; shm files outlive the process, so the OOM
watcher can ship REAL allocation profiles for the victim — the general-
process answer to the reference's Go-only oomprof integration
(oom/oomprof.go:16-125; 4 memory sample types, parca_reporter.go
memory origin)."""

from __future__ import annotations

import logging
import os
import struct
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

from ..model import (
    Frame,
    FrameType,
    MappingFile,
    SampleType,
    Trace,
    TraceEventMeta,
    TraceOrigin,
)

log = logging.getLogger("parca_agent_amd.oom.heap")

MAGIC = 0x48504150
MAX_FRAMES = 24
N_ENTRIES = 16384
N_LIVE = 65536
MAPS_CAP = 256 * 1024
HEADER = struct.Struct("<4I4Q")  # + pad to 64
HEADER_SIZE = 64
ENTRY = struct.Struct(f"<QII{MAX_FRAMES}Q4Q")
assert ENTRY.size == 240
LIVE_SIZE = 24

# The reference's four memory sample types (parca_reporter.go memory
# origin); period carries the sampling rate.
ALLOC_SPACE = SampleType("alloc_space", "bytes", "space", "bytes")
ALLOC_OBJECTS = SampleType("alloc_objects", "count", "space", "bytes")
INUSE_SPACE = SampleType("inuse_space", "bytes", "space", "bytes")
INUSE_OBJECTS = SampleType("inuse_objects", "count", "space", "bytes")


@dataclass
class HeapStack:
    ips: Tuple[int, ...]
    alloc_bytes: int
    alloc_count: int
    free_bytes: int
    free_count: int

    @property
    def inuse_bytes(self) -> int:
        return max(self.alloc_bytes - self.free_bytes, 0)

    @property
    def inuse_count(self) -> int:
        return max(self.alloc_count - self.free_count, 0)


@dataclass
class HeapProfile:
    pid: int
    sample_rate: int
    samples: int
    dropped: int
    stacks: List[HeapStack] = field(default_factory=list)
    # (start, end, file_offset, path) from the embedded maps snapshot
    mappings: List[Tuple[int, int, int, str]] = field(default_factory=list)

    def resolve(self, ip: int) -> Optional[Tuple[int, int, str]]:
        """(mapping_start, file_offset_of_ip, path) for an ip."""
        for start, end, off, path in self.mappings:
            if start <= ip < end:
                return start, ip - start + off, path
        return None


def heap_file_for(pid: int, directory: str = "") -> str:
    directory = directory or os.environ.get("PARCA_HEAP_DIR", "/dev/shm")
    return os.path.join(directory, f"parca_heap_{pid}")


def parse_heap_file(path: str) -> Optional[HeapProfile]:
    try:
        with open(path, "rb") as fh:
            data = fh.read()
    except OSError:
        return None
    if len(data) < HEADER_SIZE:
        return None
    magic, _ver, pid, n_entries, rate, samples, dropped, maps_len = \
        HEADER.unpack_from(data, 0)
    if magic != MAGIC or n_entries != N_ENTRIES:
        return None
    prof = HeapProfile(pid=pid, sample_rate=int(rate),
                       samples=int(samples), dropped=int(dropped))
    base = HEADER_SIZE
    for i in range(n_entries):
        off = base + i * ENTRY.size
        if off + ENTRY.size > len(data):
            break
        vals = ENTRY.unpack_from(data, off)
        if vals[0] == 0:
            continue
        n = min(vals[1], MAX_FRAMES)
        ips = tuple(vals[3:3 + n])
        prof.stacks.append(HeapStack(
            ips=ips,
            alloc_bytes=vals[3 + MAX_FRAMES],
            alloc_count=vals[3 + MAX_FRAMES + 1],
            free_bytes=vals[3 + MAX_FRAMES + 2],
            free_count=vals[3 + MAX_FRAMES + 3]))
    maps_off = HEADER_SIZE + n_entries * ENTRY.size + N_LIVE * LIVE_SIZE
    text = data[maps_off:maps_off + min(int(maps_len), MAPS_CAP)]
    for line in text.decode("utf-8", "replace").splitlines():
        parts = line.split()
        if len(parts) < 6 or "x" not in parts[1]:
            continue
        try:
            a, b = parts[0].split("-")
            prof.mappings.append((int(a, 16), int(b, 16),
                                  int(parts[2], 16), parts[5]))
        except (ValueError, IndexError):
            continue
    return prof


_FILE_ID_CACHE: Dict[str, str] = {}


def _mapping_file(path: str) -> MappingFile:
    fid = _FILE_ID_CACHE.get(path)
    if fid is None:
        fid = ""
        if path.startswith("/"):
            try:
                from ..elf import file_id

                fid = file_id(path)
            except OSError:
                fid = ""
        _FILE_ID_CACHE[path] = fid
    return MappingFile(path=path, file_id=fid)


def heap_trace(prof: HeapProfile, stack: HeapStack) -> Trace:
    frames = []
    for ip in stack.ips:
        hit = prof.resolve(ip)
        if hit is None:
            frames.append(Frame(kind=FrameType.NATIVE, address=ip,
                                mapping=MappingFile(path="[heap-unknown]")))
        else:
            _start, file_rel, path = hit
            frames.append(Frame(kind=FrameType.NATIVE, address=file_rel,
                                mapping=_mapping_file(path)))
    if not frames:
        frames = [Frame(kind=FrameType.ERROR, address=0,
                        mapping=MappingFile(path="[heap]"),
                        function_name="unknown_allocation_site")]
    return Trace(frames=tuple(frames))


def report_heap_profile(reporter, prof: HeapProfile, comm: str = "",
                        timestamp_ns: int = 0,
                        extra_labels: Tuple = (),
                        report_allocs: bool = True) -> int:
    """Emit the 4 memory sample types for every recorded stack; returns
    the number of samples reported."""
    import time

    ts = timestamp_ns or time.time_ns()
    n = 0
    for stack in prof.stacks:
        trace = heap_trace(prof, stack)
        trace = Trace(frames=trace.frames,
                      custom_labels=tuple(extra_labels))
        pairs = [
            (INUSE_SPACE, stack.inuse_bytes),
            (INUSE_OBJECTS, stack.inuse_count),
        ]
        if report_allocs:  # reference reportAllocs gate
            pairs += [(ALLOC_SPACE, stack.alloc_bytes),
                      (ALLOC_OBJECTS, stack.alloc_count)]
        for st, value in pairs:
            if value <= 0:
                continue
            meta = TraceEventMeta(
                timestamp_ns=ts, comm=comm, pid=prof.pid, tid=prof.pid,
                origin=TraceOrigin.MEMORY, value=int(value),
                sample_type=st, period=prof.sample_rate)
            reporter.report_trace_event(trace, meta)
            n += 1
    return n
