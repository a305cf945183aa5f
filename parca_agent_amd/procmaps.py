"""Process table: /proc-backed mapping + metadata resolution.

Combines the sampler's PERF_RECORD_{MMAP2,COMM,EXIT,FORK} stream with
lazy /proc/<pid>/ reads into a process table the symbolizer and reporter
query per sample (reference analog: the fork's process manager plus
reporter/metadata/process.go /proc parsing).
"""

from __future__ import annotations

import os
import time
from bisect import bisect_right
from dataclasses import dataclass, field
from typing import List, Optional, Tuple

from .elf import ELFFile, SymbolIndex, file_id
from .lru import LRU


@dataclass
class Mapping:
    start: int
    end: int
    file_offset: int
    path: str
    executable: bool = True

    def contains(self, addr: int) -> bool:
        return self.start <= addr < self.end


@dataclass
class ExecutableInfo:
    """Per-file info shared across processes (keyed by dev:inode-free path +
    mtime; good enough for an agent that resolves within seconds of load)."""

    path: str
    file_id: str = ""
    build_id: Optional[str] = None
    error: Optional[str] = None
    symbols: Optional[SymbolIndex] = None
    # PT_LOAD (offset, vaddr, filesz) triples for runtime->ELF-vaddr
    # normalization of sampled addresses.
    load_segments: Tuple[Tuple[int, int, int], ...] = ()

    def normalize(self, runtime_addr: int, map_start: int,
                  map_file_offset: int) -> int:
        """Turn a runtime address into the ELF virtual address it was
        linked at, which is what symbolizers (and the Parca server) expect:
        vaddr = (addr - map_start + map_file_offset) mapped through the
        PT_LOAD table."""
        file_offset = runtime_addr - map_start + map_file_offset
        for seg_off, seg_vaddr, seg_filesz in self.load_segments:
            if seg_off <= file_offset < seg_off + seg_filesz:
                return seg_vaddr + (file_offset - seg_off)
        return file_offset


class ExecutableCache:
    """FileID/build-id/symbols per executable path, LRU-bounded."""

    def __init__(self, max_size: int = 1024, load_symbols: bool = False) -> None:
        self._cache: LRU[Tuple[str, float], ExecutableInfo] = LRU(max_size)
        self.load_symbols = load_symbols

    def get(self, path: str) -> ExecutableInfo:
        try:
            mtime = os.stat(path).st_mtime
        except OSError as e:
            return ExecutableInfo(path=path, error=str(e))
        key = (path, mtime)
        info = self._cache.get(key)
        if info is not None:
            return info
        info = self._load(path)
        self._cache.put(key, info)
        return info

    def _load(self, path: str) -> ExecutableInfo:
        info = ExecutableInfo(path=path)
        try:
            info.file_id = file_id(path)
            with ELFFile.open(path) as elf:
                info.build_id = elf.build_id()
                info.load_segments = tuple(
                    (seg.offset, seg.vaddr, seg.filesz)
                    for seg in elf.segments if seg.p_type == 1)  # PT_LOAD
                if self.load_symbols:
                    info.symbols = SymbolIndex(elf.symbols())
        except (OSError, ValueError) as e:
            info.error = str(e)
        return info


_KEEP_SPECIAL = ("[vdso]", "[vsyscall]")


def _is_file_backed(path: str) -> bool:
    return bool(path) and path.startswith("/") and \
        not path.startswith("//") and " (deleted)" not in path


@dataclass
class Process:
    pid: int
    comm: str = ""
    exe: str = ""
    # Sorted, non-overlapping executable mappings.
    _starts: List[int] = field(default_factory=list)
    _maps: List[Mapping] = field(default_factory=list)
    maps_loaded: bool = False
    last_seen: float = field(default_factory=time.monotonic)

    def add_mapping(self, m: Mapping) -> None:
        i = bisect_right(self._starts, m.start)
        # Drop any previous overlapping mapping (remap over same range).
        while i > 0 and i - 1 < len(self._maps) and \
                self._maps[i - 1].end > m.start:
            del self._maps[i - 1]
            del self._starts[i - 1]
            i -= 1
        while i < len(self._maps) and self._maps[i].start < m.end:
            del self._maps[i]
            del self._starts[i]
        self._starts.insert(i, m.start)
        self._maps.insert(i, m)

    def find_mapping(self, addr: int) -> Optional[Mapping]:
        i = bisect_right(self._starts, addr) - 1
        if i < 0:
            return None
        m = self._maps[i]
        return m if m.contains(addr) else None

    @property
    def mappings(self) -> List[Mapping]:
        return list(self._maps)


class ProcessTable:
    def __init__(self, max_processes: int = 4096) -> None:
        self._procs: LRU[int, Process] = LRU(max_processes)
        # Recently-exited pids: samples of a dying process can trail its
        # PERF_RECORD_EXIT within a drain window (events are applied
        # before samples), and ensure_maps would resurrect the dead
        # entry — a leak under fork storms. Tombstones block re-creation
        # briefly; pid reuse after the TTL repopulates normally.
        self._tombstones: LRU = LRU(1024)

    def get(self, pid: int, create: bool = True) -> Optional[Process]:
        p = self._procs.get(pid)
        if p is None and create:
            import time as _time

            ts = self._tombstones.get(pid)
            if ts is not None and _time.monotonic() - ts < 2.0:
                return None
            p = Process(pid=pid)
            self._procs.put(pid, p)
        return p

    def remove(self, pid: int) -> None:
        import time as _time

        self._procs.remove(pid)
        self._tombstones.put(pid, _time.monotonic())

    def handle_proc_event(self, ev) -> None:
        """Apply one sampler ProcEvent (kind: 0=comm 1=mmap2 2=exit 3=fork)."""
        if ev.kind == 0:
            p = self.get(ev.pid)
            if p is not None:
                p.comm = ev.comm
        elif ev.kind == 1:
            if not _is_file_backed(ev.filename) and \
                    ev.filename not in _KEEP_SPECIAL:
                return
            p = self.get(ev.pid)
            if p is None:
                return
            p.add_mapping(Mapping(
                start=ev.addr, end=ev.addr + ev.len,
                file_offset=ev.pgoff, path=ev.filename))
        elif ev.kind == 2:
            if ev.pid == ev.tid:  # process (not thread) exit
                self.remove(ev.pid)

    # -- /proc fallback ----------------------------------------------------

    def ensure_maps(self, pid: int) -> Optional[Process]:
        """Load mappings from /proc for processes that predate the sampler
        (mmap2 events only cover post-attach mmaps)."""
        p = self.get(pid)
        if p is None or p.maps_loaded:
            return p
        try:
            with open(f"/proc/{pid}/maps") as fh:
                for line in fh:
                    parts = line.rstrip("\n").split(maxsplit=5)
                    if len(parts) < 5:
                        continue
                    addrs, perms, offset = parts[0], parts[1], parts[2]
                    path = parts[5] if len(parts) > 5 else ""
                    if "x" not in perms:
                        continue
                    start_s, _, end_s = addrs.partition("-")
                    p.add_mapping(Mapping(
                        start=int(start_s, 16), end=int(end_s, 16),
                        file_offset=int(offset, 16), path=path))
            if not p.comm:
                p.comm = read_comm(pid) or ""
            if not p.exe:
                p.exe = read_exe(pid) or ""
            p.maps_loaded = True
        except OSError:
            return p
        return p


def read_comm(pid: int) -> Optional[str]:
    try:
        with open(f"/proc/{pid}/comm") as fh:
            return fh.read().strip()
    except OSError:
        return None


def read_exe(pid: int) -> Optional[str]:
    try:
        return os.readlink(f"/proc/{pid}/exe")
    except OSError:
        return None


def read_cmdline(pid: int) -> Optional[str]:
    try:
        with open(f"/proc/{pid}/cmdline", "rb") as fh:
            return fh.read().replace(b"\x00", b" ").decode(
                "utf-8", "replace").strip()
    except OSError:
        return None


def read_cgroup(pid: int) -> Optional[str]:
    """The v2 (or first) cgroup path — container-id extraction input
    (reference: reporter/metadata/process.go:199-246)."""
    try:
        with open(f"/proc/{pid}/cgroup") as fh:
            first = None
            for line in fh:
                parts = line.strip().split(":", 2)
                if len(parts) != 3:
                    continue
                if first is None:
                    first = parts[2]
                if parts[0] == "0":
                    return parts[2]
            return first
    except OSError:
        return None


def read_stat_starttime(pid: int) -> Optional[int]:
    try:
        with open(f"/proc/{pid}/stat") as fh:
            data = fh.read()
        # comm can contain spaces/parens; field 22 counted after the last ')'
        rest = data[data.rindex(")") + 2:].split()
        return int(rest[19])  # starttime is field 22 overall
    except (OSError, ValueError, IndexError):
        return None
