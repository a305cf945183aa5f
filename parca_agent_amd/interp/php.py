"""PHP (Zend engine) interpreter unwinder: remote PHP stacks.

Reference coverage via the fork's PHP unwinder
(/root/reference/README.md:23-30). Target-side invariant calibration in
the style of interp/python.py / interp/ruby.py:

Anchors: the exported ``executor_globals`` symbol (non-ZTS builds —
php-cli and php-fpm as distros ship them) in the php binary / libphp.

Invariants (stable across PHP 7.0-8.3):
  * ``zend_string`` layout is fixed: gc header at +0, hash at +8,
    length at +16, bytes at +24 with a NUL terminator — a candidate
    pointer is validated by decoding it.
  * ``zend_function.common.function_name`` is the zend_string pointer
    at offset 8 of every function (the union's common prefix).
  * ``zend_execute_data`` is discovered inside executor_globals as the
    pointer whose struct contains (a) a func pointer — something whose
    +8 decodes as a plausible function name — and (b) a
    prev_execute_data pointer passing the same test recursively. The
    two field offsets are calibrated, not assumed.
  * ``op_array.filename`` is the zend_string field of the func struct
    whose content is path-shaped.

Offsets cache per build FileID; synthetic-layout tests in
tests/test_php_unwinder.py (no PHP in this image — same strategy as
the CPython 3.11+/Ruby/JVM eras).
"""

from __future__ import annotations

import logging
import os
import struct
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

from ..elf import ELFFile, file_id
from ..gpu.codeobj import read_process_memory
from ..lru import LRU
from ..model import Frame, FrameType, MappingFile
from .python import RemoteMem, _plausible

log = logging.getLogger("parca_agent_amd.interp.php")

_PHP_MAPPING = MappingFile(path="<php>")
_MAX_FRAMES = 96
FUNC_NAME_OFF = 8  # zend_function.common.function_name (all PHP 7/8)


def read_zend_string(mem: RemoteMem, addr: int,
                     limit: int = 512) -> Optional[str]:
    """Decode a zend_string: len at +16, bytes at +24, NUL-terminated."""
    if not _plausible(addr):
        return None
    hdr = mem.read(addr, 24)
    if hdr is None:
        return None
    (length,) = struct.unpack_from("<Q", hdr, 16)
    if not (0 < length <= limit):
        return None
    data = mem.read(addr + 24, int(length) + 1)
    if data is None or data[int(length)] != 0:
        return None
    body = data[:int(length)]
    if any(b < 0x09 for b in body):
        return None
    return body.decode("utf-8", "replace")


@dataclass
class PhpOffsets:
    eg_current_ex: int = -1   # executor_globals -> current_execute_data
    ex_func: int = -1         # execute_data -> func
    ex_prev: int = -1         # execute_data -> prev_execute_data
    func_filename: int = -1   # zend_function -> op_array.filename (opt)

    def complete(self) -> bool:
        return (self.eg_current_ex >= 0 and self.ex_func >= 0
                and self.ex_prev >= 0)


class PhpCalibrator:
    def __init__(self, mem: RemoteMem, eg: int) -> None:
        self.mem = mem
        self.eg = eg

    def _func_name(self, func: int) -> Optional[str]:
        if not _plausible(func):
            return None
        p = self.mem.word(func + FUNC_NAME_OFF)
        if p is None:
            return None
        if p == 0:
            return ""  # main scope: anonymous (still a valid func)
        if not _plausible(p):
            return None
        s = read_zend_string(self.mem, p, limit=256)
        if s is None or not s:
            return None
        return s

    def _probe_ex(self, ex: int) -> Optional[Tuple[int, int]]:
        """(func_off, prev_off) if ex looks like a zend_execute_data
        whose prev chain is consistent."""
        data = self.mem.read(ex, 96)
        if data is None:
            return None
        func_offs = []
        for o in range(0, 96 - 8 + 1, 8):
            (f,) = struct.unpack_from("<Q", data, o)
            name = self._func_name(f) if _plausible(f) else None
            if name:  # non-empty function name decodes
                func_offs.append(o)
        for fo in func_offs:
            for po in range(0, 96 - 8 + 1, 8):
                if po == fo:
                    continue
                (prev,) = struct.unpack_from("<Q", data, po)
                if prev == 0 or prev == ex or not _plausible(prev):
                    continue
                pf = self.mem.word(prev + fo)
                if pf is None or not _plausible(pf):
                    continue
                if self._func_name(pf) is not None:
                    # prev-of-prev must be 0 or chain again
                    pp = self.mem.word(prev + po)
                    if pp == 0 or (pp is not None and _plausible(pp)):
                        return fo, po
        return None

    def run(self) -> Optional[PhpOffsets]:
        egdata = self.mem.read_some(self.eg, 2048)
        for o in range(0, len(egdata) - 8 + 1, 8):
            (ex,) = struct.unpack_from("<Q", egdata, o)
            if not _plausible(ex):
                continue
            hit = self._probe_ex(ex)
            if hit is None:
                continue
            off = PhpOffsets(eg_current_ex=o, ex_func=hit[0],
                             ex_prev=hit[1])
            self._find_filename(ex, off)
            return off
        return None

    def _find_filename(self, ex: int, off: PhpOffsets) -> None:
        func = self.mem.word(ex + off.ex_func)
        if func is None or not _plausible(func):
            return
        fdata = self.mem.read(func, 256)
        if fdata is None:
            return
        for o in range(16, 256 - 8 + 1, 8):
            (p,) = struct.unpack_from("<Q", fdata, o)
            if not _plausible(p):
                continue
            s = read_zend_string(self.mem, p, limit=512)
            if s and ("/" in s or s.endswith(".php")):
                off.func_filename = o
                return


@dataclass
class PhpProcess:
    pid: int
    eg_addr: int
    offsets: Optional[PhpOffsets] = None
    mem: Optional[RemoteMem] = None

    @property
    def usable(self) -> bool:
        return self.offsets is not None


def _php_module_of(pid) -> Tuple[Optional[str], int, int]:
    best = None
    try:
        with open(f"/proc/{pid}/maps") as fh:
            for line in fh:
                parts = line.split()
                if len(parts) < 6 or not parts[1].startswith("r"):
                    continue
                path = parts[5]
                base = path.rsplit("/", 1)[-1]
                if base == "php" or base.startswith("php8") or \
                        base.startswith("php7") or \
                        base.startswith("libphp") or \
                        base.startswith("php-fpm"):
                    start = int(parts[0].split("-")[0], 16)
                    foff = int(parts[2], 16)
                    if best is None or start < best[1]:
                        best = (path, start, foff)
    except OSError:
        return (None, 0, 0)
    return best if best else (None, 0, 0)


class PhpUnwinder:
    def __init__(self, processes=None) -> None:
        self.processes = processes
        self._procs: LRU[int, Optional[PhpProcess]] = LRU(
            2048, ttl_seconds=300)
        self._offsets_by_build: Dict[str, PhpOffsets] = {}
        self._func_cache: LRU[Tuple[int, int], tuple] = LRU(65536)
        self.stacks_resolved = 0
        self.resolve_failures = 0
        self.calibrations = 0

    @property
    def available(self) -> bool:
        return True

    def drop_process(self, pid: int) -> None:
        self._procs.remove(pid)

    def _probe(self, pid: int) -> Optional[PhpProcess]:
        path, base, foff = _php_module_of(pid)
        if path is None:
            return None
        rooted = f"/proc/{pid}/root{path}"
        elf_path = rooted if os.path.exists(rooted) else path
        eg_vaddr = None
        try:
            with ELFFile.open(elf_path) as elf:
                for sym in elf.symbols():
                    if sym.name == "executor_globals" and sym.value:
                        eg_vaddr = sym.value
                        break
                map_vaddr = elf.vaddr_for_file_offset(foff)
        except (OSError, ValueError):
            return None
        if eg_vaddr is None or map_vaddr is None:
            return None
        bias = base - map_vaddr
        mem = RemoteMem(lambda a, n: read_process_memory(pid, a, n))
        info = PhpProcess(pid=pid, eg_addr=bias + eg_vaddr, mem=mem)
        try:
            build = file_id(elf_path)
        except OSError:
            build = ""
        offsets = self._offsets_by_build.get(build) if build else None
        if offsets is None:
            self.calibrations += 1
            offsets = PhpCalibrator(mem, info.eg_addr).run()
            if offsets is None:
                return None
            if build:
                self._offsets_by_build[build] = offsets
        info.offsets = offsets
        return info

    def _process(self, pid: int) -> Optional[PhpProcess]:
        cached = self._procs.get(pid, default="MISS")
        if cached != "MISS":
            return cached
        try:
            info = self._probe(pid)
        except Exception:
            log.debug("php probe failed for pid %d", pid, exc_info=True)
            info = None
        self._procs.put(pid, info)
        return info

    def _func_names(self, info: PhpProcess, func: int) -> Tuple[str, str]:
        key = (info.pid, func)
        cached = self._func_cache.get(key)
        if cached is not None:
            return cached
        mem, off = info.mem, info.offsets
        name = filename = ""
        p = mem.word(func + FUNC_NAME_OFF)
        if p and _plausible(p):
            name = read_zend_string(mem, p) or ""
        if off.func_filename >= 0:
            fp = mem.word(func + off.func_filename)
            if fp and _plausible(fp):
                filename = read_zend_string(mem, fp) or ""
        result = (name, filename)
        if name:
            self._func_cache.put(key, result)
        return result

    def stack_for(self, pid: int, tid: int,
                  max_frames: int = _MAX_FRAMES) -> List[Frame]:
        """PHP frames (leaf-first), or []. Non-main threads skipped
        (non-ZTS builds run PHP on the main thread)."""
        if tid != pid:
            return []
        info = self._process(pid)
        if info is None or not info.usable:
            return []
        mem, off = info.mem, info.offsets
        ex = mem.word(info.eg_addr + off.eg_current_ex)
        out: List[Frame] = []
        seen = set()
        hops = 0
        while ex and _plausible(ex) and ex not in seen and \
                hops < max_frames:
            seen.add(ex)
            hops += 1
            data = mem.read(ex, max(off.ex_func, off.ex_prev) + 8)
            if data is None:
                break
            (func,) = struct.unpack_from("<Q", data, off.ex_func)
            (prev,) = struct.unpack_from("<Q", data, off.ex_prev)
            if func and _plausible(func):
                name, filename = self._func_names(info, func)
                if name:
                    out.append(Frame(
                        kind=FrameType.PHP, address=0,
                        mapping=_PHP_MAPPING, function_name=name,
                        source_file=filename))
                elif not out and prev == 0:
                    out.append(Frame(
                        kind=FrameType.PHP, address=0,
                        mapping=_PHP_MAPPING, function_name="{main}",
                        source_file=filename))
            ex = prev
        if out:
            self.stacks_resolved += 1
        else:
            self.resolve_failures += 1
        return out
