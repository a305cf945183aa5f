"""Perl interpreter unwinding via build-exact struct offsets.

The reference profiles perl through its fork's perl unwinder, which
ships hand-maintained per-version struct-offset tables (SURVEY.md §2.9
interpreter support). This build extracts the offsets from the LOCAL
perl build instead: build_native.py compiles `tools/perl_offsets.c`
against the installed CORE headers — ground truth for this exact binary
— and stores them keyed by the perl executable's FileID. At runtime the
offsets are applied only to target processes whose perl build FileID
matches, the same safety gate the CPython unwinder uses
(interp/python.py): foreign builds are skipped, never misread.

Debian/Ubuntu perl is built with ithreads+multiplicity, so every
interpreter variable hangs off `my_perl`. For the overwhelmingly common
single-ithread process, the exported global `PL_curinterp` IS that
pointer; the walker reads it and descends PL_curstackinfo's PERL_SI /
PERL_CONTEXT chain remotely (CXt_SUB/CXt_FORMAT frames carry the sub CV
and the caller COP; PL_curcop supplies the innermost file:line).
Processes that spawned additional ithreads are skipped for non-main
threads (PL_curinterp then tracks whichever thread last set context).
"""

from __future__ import annotations

import json
import logging
import os
import struct
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

from ..elf import ELFFile, file_id
from ..gpu.codeobj import read_process_memory
from ..lru import LRU
from ..model import Frame, FrameType, MappingFile

log = logging.getLogger("parca_agent_amd.interp.perl")

OFFSETS_JSON = os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    "native", "perl_offsets.json")

_PERL_MAPPING = MappingFile(path="<perl>")

# Walk bounds: live mutating structures are read racily, so every loop
# is capped and every pointer bounds-checked.
_MAX_FRAMES = 96
_MAX_CONTEXTS_PER_SI = 192
_MAX_SI = 8
_PTR_LIMIT = 1 << 48


@dataclass
class _PerlProcess:
    pid: int
    curinterp_addr: int  # remote address of PL_curinterp, 0 = not perl
    usable: bool


class PerlUnwinder:
    def __init__(self, processes=None) -> None:
        self.processes = processes
        self.offsets: Optional[dict] = None
        self.build_file_id = ""
        try:
            with open(OFFSETS_JSON) as fh:
                data = json.load(fh)
            self.offsets = data["offsets"]
            self.build_file_id = data["file_id"]
        except (OSError, ValueError, KeyError):
            pass
        self._procs: LRU = LRU(2048, ttl_seconds=600)
        # (path) -> PL_curinterp ELF vaddr (symbol), or -1 if absent.
        self._sym_vaddr: Dict[str, int] = {}
        self._str_cache: LRU = LRU(65536)
        self._cv_cache: LRU = LRU(65536)
        self.stacks_resolved = 0
        self.resolve_failures = 0

    @property
    def available(self) -> bool:
        return self.offsets is not None

    # -- process discovery -------------------------------------------------

    def _curinterp_vaddr(self, path: str) -> int:
        v = self._sym_vaddr.get(path)
        if v is not None:
            return v
        v = -1
        try:
            with ELFFile.open(path) as elf:
                for sym in elf.symbols():
                    if sym.name == "PL_curinterp":
                        v = sym.value
                        break
        except (OSError, ValueError):
            pass
        self._sym_vaddr[path] = v
        return v

    def _process(self, pid: int) -> Optional[_PerlProcess]:
        info = self._procs.get(pid)
        if info is not None:
            return info
        info = _PerlProcess(pid=pid, curinterp_addr=0, usable=False)
        best: Optional[Tuple[str, int, int]] = None
        try:
            with open(f"/proc/{pid}/maps") as fh:
                for line in fh:
                    parts = line.split()
                    if len(parts) < 6 or not parts[1].startswith("r"):
                        continue
                    path = parts[5]
                    base_name = path.rsplit("/", 1)[-1]
                    if base_name == "perl" or \
                            base_name.startswith("perl5") or \
                            base_name.startswith("libperl"):
                        start = int(parts[0].split("-")[0], 16)
                        off = int(parts[2], 16)
                        if best is None or start < best[1]:
                            best = (path, start, off)
        except OSError:
            self._procs.put(pid, info)
            return info
        if best is not None:
            path, start, map_off = best
            try:
                # FileID gate: offsets only apply to the exact build they
                # were extracted from.
                if file_id(f"/proc/{pid}/root{path}"
                           if os.path.exists(f"/proc/{pid}/root{path}")
                           else path) == self.build_file_id:
                    sym_vaddr = self._curinterp_vaddr(path)
                    if sym_vaddr >= 0:
                        with ELFFile.open(path) as elf:
                            map_vaddr = elf.vaddr_for_file_offset(map_off)
                        if map_vaddr is not None:
                            bias = start - map_vaddr
                            info.curinterp_addr = bias + sym_vaddr
                            info.usable = True
            except (OSError, ValueError):
                pass
        self._procs.put(pid, info)
        return info

    def drop_process(self, pid: int) -> None:
        self._procs.remove(pid)

    # -- remote reads ------------------------------------------------------

    def _word(self, pid: int, addr: int) -> int:
        if addr == 0 or addr > _PTR_LIMIT:
            return 0
        try:
            data = read_process_memory(pid, addr, 8)
        except OSError:
            return 0
        return struct.unpack("<Q", data)[0] if len(data) >= 8 else 0

    def _u32(self, pid: int, addr: int) -> int:
        try:
            data = read_process_memory(pid, addr, 4)
        except OSError:
            return 0
        return struct.unpack("<I", data)[0] if len(data) >= 4 else 0

    def _cstring(self, pid: int, addr: int, limit: int = 256) -> str:
        if addr == 0 or addr > _PTR_LIMIT:
            return ""
        key = (pid, addr)
        cached = self._str_cache.get(key)
        if cached is not None:
            return cached
        try:
            data = read_process_memory(pid, addr, limit)
        except OSError:
            return ""
        end = data.find(b"\x00")
        s = data[:end if end >= 0 else limit].decode("utf-8", "replace")
        self._str_cache.put(key, s)
        return s

    def _hek_name(self, pid: int, hek: int) -> str:
        off = self.offsets
        if hek == 0 or hek > _PTR_LIMIT:
            return ""
        try:
            head = read_process_memory(pid, hek, off["hek_key"] + 0)
        except OSError:
            return ""
        if len(head) < off["hek_len"] + 4:
            return ""
        (length,) = struct.unpack_from("<i", head, off["hek_len"])
        if not 0 < length <= 512:
            return ""
        try:
            raw = read_process_memory(pid, hek + off["hek_key"], length)
        except OSError:
            return ""
        return raw.decode("utf-8", "replace")

    def _cv_name(self, pid: int, cv: int) -> str:
        """Sub name from a CV: the CvNAMED hek, or CvGV's name hek."""
        key = (pid, cv)
        cached = self._cv_cache.get(key)
        if cached is not None:
            return cached
        off = self.offsets
        name = ""
        xpvcv = self._word(pid, cv + off["sv_any"])
        if xpvcv:
            flags = self._u32(pid, xpvcv + off["xpvcv_flags"])
            gv_u = self._word(pid, xpvcv + off["xpvcv_gv_u"])
            if flags & off["cvf_named"]:
                name = self._hek_name(pid, gv_u)
            elif gv_u:
                gv_any = self._word(pid, gv_u + off["sv_any"])
                if gv_any:
                    hek = self._word(pid, gv_any + off["xpvgv_namehek"])
                    name = self._hek_name(pid, hek)
        if name:  # a racy empty read must not pin "(anon)" forever
            self._cv_cache.put(key, name)
        return name

    # -- walking -----------------------------------------------------------

    def _cop_location(self, pid: int, cop: int) -> Tuple[str, int]:
        if cop == 0 or cop > _PTR_LIMIT:
            return ("", 0)
        off = self.offsets
        line = self._u32(pid, cop + off["cop_line"])
        fileptr = self._word(pid, cop + off["cop_file"])
        fname = self._cstring(pid, fileptr) if off["cop_file_is_char"] \
            else ""
        return (fname, line if line < 10_000_000 else 0)

    def stack_for(self, pid: int, tid: int,
                  max_frames: int = _MAX_FRAMES) -> List[Frame]:
        """Perl frames (leaf-first) for the sampled thread, or []."""
        if not self.available:
            return []
        if tid != pid:
            return []  # ithread targets: PL_curinterp is not this thread's
        info = self._process(pid)
        if info is None or not info.usable:
            return []
        off = self.offsets
        my_perl = self._word(pid, info.curinterp_addr)
        if my_perl == 0:
            return []
        cop = self._word(pid, my_perl + off["interp_curcop"])
        si = self._word(pid, my_perl + off["interp_curstackinfo"])
        out: List[Frame] = []
        n_si = 0
        while si and n_si < _MAX_SI and len(out) < max_frames:
            n_si += 1
            cxstack = self._word(pid, si + off["si_cxstack"])
            cxix = struct.unpack(
                "<i", struct.pack("<I", self._u32(pid,
                                                  si + off["si_cxix"])))[0]
            if cxstack and 0 <= cxix < _MAX_CONTEXTS_PER_SI:
                for ix in range(cxix, -1, -1):
                    if len(out) >= max_frames:
                        break
                    cx = cxstack + ix * off["cx_size"]
                    ctype = self._u32(pid, cx + off["cx_type"]) & 0xFF
                    base = ctype & off["cxtypemask"]
                    if base in (off["cxt_sub"], off["cxt_format"]):
                        cv = self._word(pid, cx + off["cx_sub_cv"])
                        name = self._cv_name(pid, cv) or "(anon)"
                        fname, line = self._cop_location(pid, cop)
                        out.append(Frame(
                            kind=FrameType.PERL, address=0,
                            mapping=_PERL_MAPPING, function_name=name,
                            source_file=fname, source_line=line))
                        cop = self._word(pid, cx + off["cx_oldcop"])
                    elif base == off["cxt_eval"]:
                        fname, line = self._cop_location(pid, cop)
                        out.append(Frame(
                            kind=FrameType.PERL, address=0,
                            mapping=_PERL_MAPPING, function_name="(eval)",
                            source_file=fname, source_line=line))
                        cop = self._word(pid, cx + off["cx_oldcop"])
            si = self._word(pid, si + off["si_prev"])
        # Outermost: whatever file:line the remaining COP points at
        # (top-level code of the script).
        fname, line = self._cop_location(pid, cop)
        if fname:
            out.append(Frame(
                kind=FrameType.PERL, address=0, mapping=_PERL_MAPPING,
                function_name="main::", source_file=fname,
                source_line=line))
        if out:
            self.stacks_resolved += 1
        else:
            self.resolve_failures += 1
        return out
