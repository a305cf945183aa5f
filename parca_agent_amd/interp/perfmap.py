"""JIT symbolization via the Linux perf map convention.

JIT runtimes that opt in (node --perf-basic-prof, JVM with
perf-map-agent, .NET, wasmtime, LuaJIT builds, ...) write
`/tmp/perf-<pid>.map` lines of `START SIZE symbol-name`. The reference
covers these engines with dedicated fork unwinders; this generic path
gives frame NAMES for any conforming runtime: sampled IPs landing in
anonymous executable mappings are resolved against the map.

Maps grow as code is jitted: the index refreshes incrementally when the
file grows (re-parsing only the appended tail) and fully when it is
replaced (JIT code GC/rewrites).
"""

from __future__ import annotations

import bisect
import logging
import os
import struct
from dataclasses import dataclass, field
from typing import List, Optional, Tuple

from ..lru import LRU

log = logging.getLogger("parca_agent_amd.interp.perfmap")


@dataclass
class _PerfMap:
    path: str
    entries: List[Tuple[int, int, str]] = field(default_factory=list)
    sorted_addrs: List[int] = field(default_factory=list)
    size_parsed: int = 0
    mtime: float = 0.0
    dirty: bool = True

    def refresh(self) -> None:
        try:
            st = os.stat(self.path)
        except OSError:
            self.entries = []
            self.sorted_addrs = []
            self.size_parsed = 0
            return
        if st.st_size == self.size_parsed and st.st_mtime == self.mtime:
            return
        start_from = self.size_parsed
        if st.st_size < self.size_parsed:
            # truncated/replaced: full reparse
            self.entries = []
            start_from = 0
        try:
            with open(self.path, "rb") as fh:
                fh.seek(start_from)
                data = fh.read()
        except OSError:
            return
        for line in data.splitlines():
            parts = line.split(b" ", 2)
            if len(parts) != 3:
                continue
            try:
                start = int(parts[0], 16)
                size = int(parts[1], 16)
            except ValueError:
                continue
            name = parts[2].decode("utf-8", "replace").strip()
            if name:
                self.entries.append((start, size, name))
        self.size_parsed = start_from + len(data)
        self.mtime = st.st_mtime
        # Later entries win on overlap (JIT re-tiering re-emits ranges).
        self.entries.sort(key=lambda e: e[0])
        self.sorted_addrs = [e[0] for e in self.entries]

    def lookup(self, addr: int) -> Optional[str]:
        i = bisect.bisect_right(self.sorted_addrs, addr) - 1
        # Walk back over overlapping ranges: prefer the innermost match.
        while i >= 0:
            start, size, name = self.entries[i]
            if start <= addr < start + size:
                return name
            if addr - start > (1 << 24):
                break  # far past any plausible entry
            i -= 1
        return None


class _JitDump:
    """Reader for the perf jitdump format (jit-<pid>.dump), the SECOND
    JIT-symbol convention: Julia, LLVM ORC/MCJIT-based runtimes and
    perf-inject-style emitters write JIT_CODE_LOAD records instead of
    text perf maps. Parsed incrementally like _PerfMap."""

    MAGIC = 0x4A695444  # 'JiTD'
    HEADER = struct.Struct("<IIIIIIQQ")
    REC = struct.Struct("<IIQ")
    LOAD = struct.Struct("<IIQQQQ")
    JIT_CODE_LOAD = 0

    def __init__(self, path: str) -> None:
        self.path = path
        self.entries: List[Tuple[int, int, str]] = []
        self.sorted_addrs: List[int] = []
        self.size_parsed = 0
        self._header_ok = False

    def refresh(self) -> None:
        try:
            st = os.stat(self.path)
        except OSError:
            return
        if st.st_size <= self.size_parsed:
            return
        try:
            with open(self.path, "rb") as fh:
                if not self._header_ok:
                    hdr = fh.read(self.HEADER.size)
                    if len(hdr) < self.HEADER.size:
                        return
                    magic, _ver, total_size = struct.unpack_from(
                        "<III", hdr, 0)
                    if magic != self.MAGIC:
                        self.size_parsed = st.st_size  # not a jitdump
                        return
                    self._header_ok = True
                    self.size_parsed = max(total_size, self.HEADER.size)
                fh.seek(self.size_parsed)
                data = fh.read()
        except OSError:
            return
        pos = 0
        while pos + self.REC.size <= len(data):
            rec_id, total, _ts = self.REC.unpack_from(data, pos)
            if total < self.REC.size or pos + total > len(data):
                break  # partially-written record: retry next refresh
            if rec_id == self.JIT_CODE_LOAD and \
                    total >= self.REC.size + self.LOAD.size:
                body = pos + self.REC.size
                (_pid, _tid, _vma, code_addr, code_size,
                 _idx) = self.LOAD.unpack_from(data, body)
                name_start = body + self.LOAD.size
                nul = data.find(b"\x00", name_start, pos + total)
                if nul > 0:
                    name = data[name_start:nul].decode("utf-8", "replace")
                    if name:
                        self.entries.append(
                            (code_addr, code_size, name))
            pos += total
        self.size_parsed += pos
        self.entries.sort(key=lambda e: e[0])
        self.sorted_addrs = [e[0] for e in self.entries]

    # identical lookup semantics to _PerfMap
    lookup = _PerfMap.lookup


class PerfMapResolver:
    def __init__(self, refresh_interval: float = 2.0,
                 max_processes: int = 1024) -> None:
        self._maps: LRU[int, Optional[_PerfMap]] = LRU(
            max_processes, ttl_seconds=600)
        self._refresh_interval = refresh_interval
        self._last_refresh: dict = {}
        self.symbols_resolved = 0

    def _map_for(self, pid: int) -> Optional[_PerfMap]:
        import time

        pm = self._maps.get(pid, default="MISS")
        if isinstance(pm, float):
            # Negative cache: the map did not exist yet. JIT runtimes
            # create it lazily (first compilation), so recheck
            # periodically rather than caching the miss forever.
            if time.monotonic() - pm < self._refresh_interval:
                return None
            pm = "MISS"
        if pm == "MISS":
            path = f"/tmp/perf-{pid}.map"
            # Also honour containerized processes whose /tmp differs.
            if not os.path.exists(path):
                alt = f"/proc/{pid}/root/tmp/perf-{pid}.map"
                path = alt if os.path.exists(alt) else None
            if path is not None:
                pm = _PerfMap(path=path)
                self._maps.put(pid, pm)
                return pm
            # jitdump convention (Julia, LLVM-JIT runtimes): written to
            # the process cwd or TMPDIR as jit-<pid>.dump.
            for cand in (f"/proc/{pid}/cwd/jit-{pid}.dump",
                         f"/tmp/jit-{pid}.dump",
                         f"/proc/{pid}/root/tmp/jit-{pid}.dump"):
                if os.path.exists(cand):
                    pm = _JitDump(path=cand)
                    self._maps.put(pid, pm)
                    return pm
            self._maps.put(pid, time.monotonic())
            return None
        return pm

    def lookup(self, pid: int, addr: int) -> Optional[str]:
        pm = self._map_for(pid)
        if pm is None:
            return None
        import time

        now = time.monotonic()
        if now - self._last_refresh.get(pid, 0.0) > self._refresh_interval:
            self._last_refresh[pid] = now
            pm.refresh()
        name = pm.lookup(addr)
        if name:
            self.symbols_resolved += 1
        return name
