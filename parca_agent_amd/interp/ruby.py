"""CRuby interpreter unwinder: remote ruby stacks for sampled processes.

The reference profiles ruby through its fork's ruby unwinder with
hand-maintained per-version struct tables (SURVEY.md §2.9;
/root/reference/README.md:23-30). Like the round-2 CPython unwinder
(interp/python.py), this walker avoids version tables: it calibrates
against EACH TARGET using invariants that hold across CRuby 3.0-3.3
and verifies them remotely before use.

Anchors (exported symbols in ruby / libruby, resolved via load bias):
  * ``ruby_single_main_ractor`` — the main ractor (3.0+; single-ractor
    processes, i.e. essentially all of them).
  * ``ruby_current_vm_ptr`` — fallback: rb_vm_t, which contains the
    main ractor pointer (found by the same ec-scan, one indirection
    deeper).
  * ``rb_cString`` — klass pointer for validating VALUE strings.

Invariants used:
  * ``rb_execution_context_t`` begins ``{VALUE *vm_stack; size_t
    vm_stack_size; rb_control_frame_t *cfp;}`` in every 3.x release —
    a candidate pointer E inside the ractor is an ec iff word0 is a
    plausible pointer, word1 a sane element count, and word2 lies
    inside [word0, word0 + word1*8).
  * ``rb_control_frame_t`` starts ``{pc; sp; iseq; self; ep;}``; its
    SIZE varies by version (jit_return etc.), so the frame stride is
    calibrated by trying plausible strides and checking that successive
    frames keep sp/ep inside the vm_stack.
  * ``iseq->body`` is the pointer field whose target contains VALUE
    strings (klass == rb_cString) for the label and a path object —
    found by scanning candidate offsets and validating string decode.
  * RString decode: RBasic{flags, klass} then either embedded bytes or
    {len, ptr} heap form, disambiguated by RSTRING_NOEMBED (flag bit
    13) — stable across 3.x.

Caching: offsets per ruby-build FileID; per-pid anchors with TTL.
All reads are process_vm_readv, bounds-checked and capped.
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

log = logging.getLogger("parca_agent_amd.interp.ruby")

_RUBY_MAPPING = MappingFile(path="<ruby>")

_MAX_FRAMES = 128
_RSTRING_NOEMBED = 1 << 13
# Plausible rb_control_frame_t strides (bytes) across 3.0-3.3: 5-8
# pointer-sized fields.
_CFP_STRIDES = (40, 48, 56, 64)
_ANCHOR_SYMS = ("ruby_single_main_ractor", "ruby_current_vm_ptr",
                "rb_cString")


@dataclass
class RubyOffsets:
    ec_in_ractor: int = -1       # offset of running_ec within rb_ractor_t
    cfp_stride: int = -1
    iseq_body: int = -1          # offset of body within rb_iseq_t
    body_label: int = -1         # offset of the label VALUE within body
    body_path: int = -1          # offset of the pathobj VALUE within body

    def complete(self) -> bool:
        return (self.ec_in_ractor >= 0 and self.cfp_stride > 0
                and self.iseq_body >= 0 and self.body_label >= 0)


def _ruby_module_of(pid) -> Tuple[Optional[str], int, int]:
    best = None
    try:
        with open(f"/proc/{pid}/maps") as fh:
            for line in fh:
                parts = line.split()
                if len(parts) < 6 or not parts[1].startswith("r"):
                    continue
                path = parts[5]
                base = path.rsplit("/", 1)[-1]
                if base == "ruby" or base.startswith("ruby3") or \
                        base.startswith("libruby"):
                    start = int(parts[0].split("-")[0], 16)
                    off = int(parts[2], 16)
                    if best is None or start < best[1]:
                        best = (path, start, off)
    except OSError:
        return (None, 0, 0)
    return best if best else (None, 0, 0)


class RubyStringReader:
    """Decode a VALUE known/suspected to be an RString."""

    def __init__(self, mem: RemoteMem, cstring_klass: int) -> None:
        self.mem = mem
        self.klass = cstring_klass

    def is_string(self, value: int) -> bool:
        if not _plausible(value):
            return False
        k = self.mem.word(value + 8)
        return k == self.klass

    def read(self, value: int, limit: int = 256) -> str:
        data = self.mem.read(value, 40)
        if data is None:
            return ""
        (flags,) = struct.unpack_from("<Q", data, 0)
        if flags & _RSTRING_NOEMBED:
            (length,) = struct.unpack_from("<q", data, 16)
            (ptr,) = struct.unpack_from("<Q", data, 24)
            if not (0 < length <= limit) or not _plausible(ptr):
                return ""
            raw = self.mem.read(ptr, int(length))
        else:
            # embedded: 3.x stores len then bytes (>=3.2: long len at
            # +16, bytes at +24; <=3.1: bytes at +16, len packed in
            # flags). Try the long-len form first, fall back to
            # flag-packed.
            (length,) = struct.unpack_from("<q", data, 16)
            if 0 < length <= 16:  # embedded capacity is small
                raw = self.mem.read(value + 24, int(length))
            else:
                length = (flags >> 15) & 0x1F  # RSTRING_EMBED_LEN (<=3.1)
                if not (0 < length <= 24):
                    return ""
                raw = self.mem.read(value + 16, int(length))
        if raw is None:
            return ""
        return raw.decode("utf-8", "replace")


class RubyCalibrator:
    def __init__(self, mem: RemoteMem, ractor: int,
                 strings: RubyStringReader) -> None:
        self.mem = mem
        self.ractor = ractor
        self.strings = strings

    def _is_ec(self, addr: int) -> Optional[Tuple[int, int, int]]:
        """(vm_stack, n_slots, cfp) if addr looks like an ec."""
        data = self.mem.read(addr, 24)
        if data is None:
            return None
        vm_stack, size, cfp = struct.unpack("<3Q", data)
        if not _plausible(vm_stack) or not (64 <= size <= (1 << 24)):
            return None
        if not (vm_stack <= cfp < vm_stack + size * 8):
            return None
        if cfp % 8:
            return None
        return vm_stack, size, cfp

    def run(self) -> Optional[RubyOffsets]:
        rdata = self.mem.read_some(self.ractor, 4096)
        for off in range(0, len(rdata) - 8 + 1, 8):
            (cand,) = struct.unpack_from("<Q", rdata, off)
            if not _plausible(cand):
                continue
            ec = self._is_ec(cand)
            if ec is None:
                continue
            offsets = RubyOffsets(ec_in_ractor=off)
            if self._calibrate_frames(ec, offsets):
                return offsets
        return None

    def _calibrate_frames(self, ec: Tuple[int, int, int],
                          off: RubyOffsets) -> bool:
        vm_stack, size, cfp = ec
        stack_end = vm_stack + size * 8
        # Prefer a stride that yields >= 2 consistent iseq frames: a
        # wrong stride can read ONE valid frame (the first) before its
        # bounds checks fire, so single-frame evidence is only accepted
        # when nothing better exists.
        for min_frames in (2, 1):
            for stride in _CFP_STRIDES:
                iseqs = self._walk(cfp, stack_end, stride, vm_stack)
                if len(iseqs) < min_frames:
                    continue
                if self._calibrate_iseq(iseqs, off):
                    off.cfp_stride = stride
                    return True
        return False

    def _walk(self, cfp: int, stack_end: int, stride: int,
              vm_stack: int) -> List[int]:
        """iseq pointers from walking cfp with the given stride;
        empty when the stride is inconsistent."""
        iseqs: List[int] = []
        cur = cfp
        hops = 0
        while cur + stride <= stack_end and hops < _MAX_FRAMES:
            hops += 1
            data = self.mem.read(cur, 40)
            if data is None:
                return []
            pc, sp, iseq, self_v, ep = struct.unpack("<5Q", data)
            # sp/ep of live ruby frames point into the VM stack (dummy
            # frames may have pc==0); violations mean a wrong stride.
            if sp and not (vm_stack <= sp <= stack_end):
                return []
            if pc and iseq and _plausible(iseq):
                iseqs.append(iseq)
            cur += stride
        return iseqs

    def _calibrate_iseq(self, iseqs: List[int],
                        off: RubyOffsets) -> bool:
        """Find iseq->body and the label/path VALUEs within body."""
        for iseq in iseqs[:8]:
            idata = self.mem.read(iseq, 40)
            if idata is None:
                continue
            for boff in range(8, 40 - 8 + 1, 8):
                (body,) = struct.unpack_from("<Q", idata, boff)
                if not _plausible(body):
                    continue
                found = self._scan_body(body)
                if found is not None:
                    off.iseq_body = boff
                    off.body_label, off.body_path = found
                    return True
        return False

    def _scan_body(self, body: int) -> Optional[Tuple[int, int]]:
        data = self.mem.read(body, 256)
        if data is None:
            return None
        label_off = path_off = -1
        for o in range(0, 256 - 8 + 1, 8):
            (v,) = struct.unpack_from("<Q", data, o)
            if not _plausible(v) or not self.strings.is_string(v):
                continue
            s = self.strings.read(v)
            if not s:
                continue
            if ("/" in s or s.endswith(".rb") or s.startswith("<")) and \
                    path_off < 0:
                path_off = o
            elif label_off < 0:
                label_off = o
        if label_off < 0 and path_off < 0:
            return None
        if label_off < 0:
            label_off = path_off
        return label_off, path_off


@dataclass
class RubyProcess:
    pid: int
    ractor_ptr_addr: int  # address holding the main-ractor pointer
    cstring_addr: int
    offsets: Optional[RubyOffsets] = None
    mem: Optional[RemoteMem] = None

    @property
    def usable(self) -> bool:
        return self.offsets is not None


class RubyUnwinder:
    def __init__(self, processes=None) -> None:
        self.processes = processes
        self._procs: LRU[int, Optional[RubyProcess]] = LRU(
            2048, ttl_seconds=300)
        self._offsets_by_build: Dict[str, RubyOffsets] = {}
        self._anchor_cache: Dict[str, Optional[Dict[str, int]]] = {}
        self._iseq_cache: LRU[Tuple[int, int], tuple] = LRU(65536)
        self.stacks_resolved = 0
        self.resolve_failures = 0
        self.calibrations = 0

    @property
    def available(self) -> bool:
        return True

    def drop_process(self, pid: int) -> None:
        self._procs.remove(pid)

    # -- probing -----------------------------------------------------------

    def _anchors(self, elf_path: str) -> Optional[Dict[str, int]]:
        if elf_path in self._anchor_cache:
            return self._anchor_cache[elf_path]
        out: Optional[Dict[str, int]] = None
        try:
            with ELFFile.open(elf_path) as elf:
                vals: Dict[str, int] = {}
                for sym in elf.symbols():
                    if sym.name in _ANCHOR_SYMS and sym.value:
                        vals.setdefault(sym.name, sym.value)
            if "rb_cString" in vals and (
                    "ruby_single_main_ractor" in vals or
                    "ruby_current_vm_ptr" in vals):
                out = vals
        except (OSError, ValueError):
            out = None
        self._anchor_cache[elf_path] = out
        return out

    def _probe(self, pid: int) -> Optional[RubyProcess]:
        path, base, file_off = _ruby_module_of(pid)
        if path is None:
            return None
        rooted = f"/proc/{pid}/root{path}"
        elf_path = rooted if os.path.exists(rooted) else path
        vaddrs = self._anchors(elf_path)
        if vaddrs is None:
            return None
        try:
            with ELFFile.open(elf_path) as elf:
                map_vaddr = elf.vaddr_for_file_offset(file_off)
        except (OSError, ValueError):
            return None
        if map_vaddr is None:
            return None
        bias = base - map_vaddr
        mem = RemoteMem(lambda a, n: read_process_memory(pid, a, n))

        # rb_cString is a VALUE variable: the class object pointer is
        # stored AT the symbol address.
        cstring = mem.word(bias + vaddrs["rb_cString"]) or 0
        if not _plausible(cstring):
            return None
        ractor_ptr = 0
        if "ruby_single_main_ractor" in vaddrs:
            ractor_ptr = bias + vaddrs["ruby_single_main_ractor"]
        info = RubyProcess(pid=pid, ractor_ptr_addr=ractor_ptr,
                           cstring_addr=cstring, mem=mem)

        ractor = mem.word(ractor_ptr) if ractor_ptr else None
        if ractor is None or not _plausible(ractor):
            # multi-ractor or pre-3.0: fall back through the vm struct
            if "ruby_current_vm_ptr" not in vaddrs:
                return None
            vm = mem.word(bias + vaddrs["ruby_current_vm_ptr"])
            if vm is None or not _plausible(vm):
                return None
            ractor = self._find_ractor_via_vm(mem, vm, cstring)
            if ractor is None:
                return None

        try:
            build = file_id(elf_path)
        except OSError:
            build = ""
        offsets = self._offsets_by_build.get(build) if build else None
        if offsets is None:
            self.calibrations += 1
            strings = RubyStringReader(mem, cstring)
            offsets = RubyCalibrator(mem, ractor, strings).run()
            if offsets is None:
                return None
            if build:
                self._offsets_by_build[build] = offsets
        info.offsets = offsets
        info.ractor = ractor  # type: ignore[attr-defined]
        return info

    def _find_ractor_via_vm(self, mem: RemoteMem, vm: int,
                            cstring: int) -> Optional[int]:
        """Scan rb_vm_t for a pointer to a struct that contains an
        ec-shaped field (i.e. the main ractor)."""
        strings = RubyStringReader(mem, cstring)
        vdata = mem.read_some(vm, 2048)
        for o in range(0, len(vdata) - 8 + 1, 8):
            (cand,) = struct.unpack_from("<Q", vdata, o)
            if not _plausible(cand):
                continue
            cal = RubyCalibrator(mem, cand, strings)
            rdata = mem.read_some(cand, 1024)
            for off in range(0, len(rdata) - 8 + 1, 8):
                (e,) = struct.unpack_from("<Q", rdata, off)
                if _plausible(e) and cal._is_ec(e) is not None:
                    return cand
        return None

    def _process(self, pid: int) -> Optional[RubyProcess]:
        cached = self._procs.get(pid, default="MISS")
        if cached != "MISS":
            return cached
        try:
            info = self._probe(pid)
        except Exception:
            log.debug("ruby probe failed for pid %d", pid, exc_info=True)
            info = None
        self._procs.put(pid, info)
        return info

    # -- walking -----------------------------------------------------------

    def _iseq_names(self, info: RubyProcess, iseq: int) -> Tuple[str, str]:
        key = (info.pid, iseq)
        cached = self._iseq_cache.get(key)
        if cached is not None:
            return cached
        mem, off = info.mem, info.offsets
        strings = RubyStringReader(mem, info.cstring_addr)
        label = path = ""
        body = mem.word(iseq + off.iseq_body)
        if body is not None and _plausible(body):
            span = max(off.body_label, off.body_path
                       if off.body_path >= 0 else 0) + 8
            data = mem.read(body, span)
            if data is not None:
                (lv,) = struct.unpack_from("<Q", data, off.body_label)
                if _plausible(lv) and strings.is_string(lv):
                    label = strings.read(lv)
                if off.body_path >= 0:
                    (pv,) = struct.unpack_from("<Q", data, off.body_path)
                    if _plausible(pv) and strings.is_string(pv):
                        path = strings.read(pv)
        result = (label, path)
        if label:
            self._iseq_cache.put(key, result)
        return result

    def stack_for(self, pid: int, tid: int,
                  max_frames: int = _MAX_FRAMES) -> List[Frame]:
        """Ruby frames (leaf-first) for the sampled process, or []."""
        if tid != pid:
            return []  # single-ractor main thread only
        info = self._process(pid)
        if info is None or not info.usable:
            return []
        mem, off = info.mem, info.offsets
        ec = mem.word(getattr(info, "ractor", 0) + off.ec_in_ractor)
        if ec is None or not _plausible(ec):
            self.resolve_failures += 1
            return []
        head = mem.read(ec, 24)
        if head is None:
            self.resolve_failures += 1
            return []
        vm_stack, size, cfp = struct.unpack("<3Q", head)
        stack_end = vm_stack + size * 8
        out: List[Frame] = []
        cur = cfp
        hops = 0
        while cur + off.cfp_stride <= stack_end and hops < max_frames:
            hops += 1
            data = mem.read(cur, 24)
            if data is None:
                break
            pc, _sp, iseq = struct.unpack("<3Q", data)
            cur += off.cfp_stride
            if not pc or not iseq or not _plausible(iseq):
                continue  # C frames / dummy frames carry no iseq pc
            label, path = self._iseq_names(info, iseq)
            if label:
                out.append(Frame(
                    kind=FrameType.RUBY, address=0, mapping=_RUBY_MAPPING,
                    function_name=label, source_file=path))
        if out:
            self.stacks_resolved += 1
        else:
            self.resolve_failures += 1
        return out
