"""OpenJDK (HotSpot) interpreter unwinder: remote Java stacks.

The reference covers JVM bytecode stacks through its fork's unwinder
(/root/reference/README.md:23-30). HotSpot's template
interpreter keeps a classic rbp-linked frame and stores the executing
``Method*`` in a fixed slot of every interpreter frame
(``interpreter_frame_method_offset`` = fp - 3 words on x86-64, stable
across JDK 8-21), so Java frames are recoverable from the stack dump
the CPU sampler already captures — no in-process agent needed.

Like interp/python.py and interp/ruby.py this avoids per-version struct
tables. Calibration against the target validates every step:

  * A candidate ``Method*`` must carry a vtable pointer into libjvm's
    mapped range (Method is polymorphic Metadata; its vtable lives in
    libjvm.so), preferably the exact exported ``_ZTV6Method`` address.
  * ``Method->_constMethod`` and ``ConstMethod->_constants`` offsets
    are found by scanning for the pointer whose target itself carries
    a libjvm vtable (ConstantPool is also polymorphic Metadata).
  * ``_name_index``/``_signature_index`` (adjacent u2 fields in
    ConstMethod) and the constant-pool base offset are solved jointly:
    the signature Symbol must decode to a string starting with ``(``
    (method descriptors always do) while the name Symbol decodes to a
    Java identifier — a constraint pair with essentially no false
    positives.
  * Symbol layout is probed between the two eras: length u2 at +4 and
    body at +6 (JDK15+, hash+refcount word first) vs length at +0 and
    body at +8 (JDK 8/11).
  * The declaring class is found by scanning ConstantPool for the
    ``_pool_holder`` (a Klass, vtable in libjvm) whose ``_name``
    Symbol looks like an internal class path (``java/lang/Thread``).

Offsets cache per libjvm FileID. Compiled (JIT) Java frames are
covered by the perf-map path (interp/perfmap.py with perf-map-agent or
-XX:+DumpPerfMapAtExit); this walker adds the interpreter frames those
maps cannot see. No JVM ships in this image, so tests mirror both
layout eras synthetically (tests/test_jvm_unwinder.py), the same
strategy as the CPython 3.11+/Ruby eras.
"""

from __future__ import annotations

import logging
import os
import re
import struct
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

from ..elf import ELFFile, file_id
from ..gpu.codeobj import read_process_memory
from ..lru import LRU
from ..model import Frame, FrameType, MappingFile
from .python import RemoteMem, _plausible

log = logging.getLogger("parca_agent_amd.interp.jvm")

_JVM_MAPPING = MappingFile(path="<jvm>")

# fp-relative slot of Method* in an interpreter frame (x86-64 template
# interpreter: interpreter_frame_method_offset = -3 words).
METHOD_SLOT = -3 * 8
_MAX_FRAMES = 64

_IDENT = re.compile(r"^[A-Za-z_$<][A-Za-z0-9_$<>]*$")
_CLASSPATH = re.compile(r"^[A-Za-z_$][A-Za-z0-9_$]*(/[A-Za-z0-9_$]+)+$")


@dataclass
class JvmOffsets:
    const_method: int = -1    # Method -> _constMethod
    constants: int = -1       # ConstMethod -> _constants
    name_index: int = -1      # ConstMethod u2 (signature at +2)
    cp_base: int = -1         # ConstantPool header size
    sym_len: int = -1         # Symbol length offset (4 or 0)
    sym_body: int = -1        # Symbol body offset (6 or 8)
    pool_holder: int = -1     # ConstantPool -> _pool_holder (optional)
    klass_name: int = -1      # Klass -> _name (optional)

    def complete(self) -> bool:
        return (self.const_method >= 0 and self.constants >= 0
                and self.name_index >= 0 and self.cp_base >= 0
                and self.sym_len >= 0)


class SymbolReader:
    def __init__(self, mem: RemoteMem) -> None:
        self.mem = mem

    def read_with(self, addr: int, len_off: int, body_off: int,
                  limit: int = 256) -> str:
        hdr = self.mem.read(addr, body_off + 0)
        if hdr is None and body_off > 0:
            return ""
        data = self.mem.read(addr, body_off + limit)
        if data is None:
            data = self.mem.read(addr, body_off + 64)
            if data is None:
                return ""
        (length,) = struct.unpack_from("<H", data, len_off)
        if not (0 < length <= limit):
            return ""
        body = data[body_off:body_off + length]
        if len(body) < length:
            more = self.mem.read(addr + body_off, length)
            if more is None:
                return ""
            body = more
        try:
            return body.decode("utf-8")
        except UnicodeDecodeError:
            return ""


class JvmCalibrator:
    """Solve the offsets from one validated Method* observed live."""

    def __init__(self, mem: RemoteMem, jvm_range: Tuple[int, int]) -> None:
        self.mem = mem
        self.jvm_lo, self.jvm_hi = jvm_range
        self.symbols = SymbolReader(mem)

    def vptr_in_jvm(self, obj: int) -> bool:
        if not _plausible(obj):
            return False
        v = self.mem.word(obj)
        return v is not None and self.jvm_lo <= v < self.jvm_hi

    def run(self, method: int) -> Optional[JvmOffsets]:
        if not self.vptr_in_jvm(method):
            return None
        mdata = self.mem.read(method, 64)
        if mdata is None:
            return None
        for cm_off in range(8, 64 - 8 + 1, 8):
            (cm,) = struct.unpack_from("<Q", mdata, cm_off)
            if not _plausible(cm):
                continue
            cmdata = self.mem.read(cm, 128)
            if cmdata is None:
                continue
            for cp_off in range(0, 64 - 8 + 1, 8):
                (cp,) = struct.unpack_from("<Q", cmdata, cp_off)
                if not _plausible(cp) or not self.vptr_in_jvm(cp):
                    continue
                off = self._solve_indices(cm, cmdata, cp)
                if off is not None:
                    off.const_method = cm_off
                    off.constants = cp_off
                    self._solve_holder(cp, off)
                    return off
        return None

    def _solve_indices(self, cm: int, cmdata: bytes,
                       cp: int) -> Optional[JvmOffsets]:
        for sym_len, sym_body in ((4, 6), (0, 8)):
            for no in range(16, len(cmdata) - 4 + 1, 2):
                (name_idx,) = struct.unpack_from("<H", cmdata, no)
                (sig_idx,) = struct.unpack_from("<H", cmdata, no + 2)
                if not (0 < name_idx < 60000 and 0 < sig_idx < 60000):
                    continue
                if name_idx == sig_idx:
                    continue
                for base in range(40, 128 + 1, 8):
                    sig_p = self.mem.word(cp + base + sig_idx * 8)
                    if sig_p is None or not _plausible(sig_p):
                        continue
                    sig = self.symbols.read_with(sig_p, sym_len, sym_body)
                    if not sig.startswith("(") or ")" not in sig:
                        continue
                    name_p = self.mem.word(cp + base + name_idx * 8)
                    if name_p is None or not _plausible(name_p):
                        continue
                    name = self.symbols.read_with(name_p, sym_len, sym_body)
                    if not name or not _IDENT.match(name):
                        continue
                    return JvmOffsets(name_index=no, cp_base=base,
                                      sym_len=sym_len, sym_body=sym_body)
        return None

    def _solve_holder(self, cp: int, off: JvmOffsets) -> None:
        """Optional: ConstantPool->_pool_holder and Klass->_name, for
        Class.method naming."""
        cpdata = self.mem.read(cp, off.cp_base)
        if cpdata is None:
            return
        for ho in range(8, len(cpdata) - 8 + 1, 8):
            (k,) = struct.unpack_from("<Q", cpdata, ho)
            if not _plausible(k) or not self.vptr_in_jvm(k):
                continue
            kdata = self.mem.read(k, 64)
            if kdata is None:
                continue
            for kn in range(8, 64 - 8 + 1, 8):
                (sym,) = struct.unpack_from("<Q", kdata, kn)
                if not _plausible(sym):
                    continue
                s = self.symbols.read_with(sym, off.sym_len, off.sym_body)
                if s and _CLASSPATH.match(s):
                    off.pool_holder = ho
                    off.klass_name = kn
                    return


@dataclass
class JvmProcess:
    pid: int
    jvm_lo: int = 0
    jvm_hi: int = 0
    offsets: Optional[JvmOffsets] = None
    mem: Optional[RemoteMem] = None

    @property
    def usable(self) -> bool:
        return self.offsets is not None


def _libjvm_range(pid) -> Tuple[int, int]:
    lo = hi = 0
    try:
        with open(f"/proc/{pid}/maps") as fh:
            for line in fh:
                parts = line.split()
                if len(parts) < 6:
                    continue
                if parts[5].rsplit("/", 1)[-1].startswith("libjvm.so"):
                    a, b = parts[0].split("-")
                    s, e = int(a, 16), int(b, 16)
                    lo = s if lo == 0 else min(lo, s)
                    hi = max(hi, e)
    except OSError:
        pass
    return lo, hi


def _libjvm_path(pid) -> Optional[str]:
    try:
        with open(f"/proc/{pid}/maps") as fh:
            for line in fh:
                parts = line.split()
                if len(parts) >= 6 and \
                        parts[5].rsplit("/", 1)[-1].startswith("libjvm.so"):
                    return parts[5]
    except OSError:
        pass
    return None


class JvmUnwinder:
    """Per-agent singleton; extracts Java interpreter frames from the
    sampled stack dump's rbp chain."""

    def __init__(self, processes=None) -> None:
        self.processes = processes
        self._procs: LRU[int, Optional[JvmProcess]] = LRU(
            2048, ttl_seconds=300)
        self._offsets_by_build: Dict[str, JvmOffsets] = {}
        self._method_cache: LRU[Tuple[int, int], str] = LRU(65536)
        self.stacks_resolved = 0
        self.resolve_failures = 0
        self.calibrations = 0

    @property
    def available(self) -> bool:
        return True

    def drop_process(self, pid: int) -> None:
        self._procs.remove(pid)

    def might_be_jvm(self, pid: int) -> bool:
        """Cheap cached gate so the CPU service does not materialize
        stack bytes for non-JVM processes."""
        return self._process(pid) is not None

    def _process(self, pid: int) -> Optional[JvmProcess]:
        cached = self._procs.get(pid, default="MISS")
        if cached != "MISS":
            return cached
        lo, hi = _libjvm_range(pid)
        info = None
        if lo:
            mem = RemoteMem(lambda a, n: read_process_memory(pid, a, n))
            info = JvmProcess(pid=pid, jvm_lo=lo, jvm_hi=hi, mem=mem)
            path = _libjvm_path(pid)
            if path:
                rooted = f"/proc/{pid}/root{path}"
                try:
                    info.build = file_id(  # type: ignore[attr-defined]
                        rooted if os.path.exists(rooted) else path)
                except OSError:
                    pass
        self._procs.put(pid, info)
        return info

    def _method_name(self, info: JvmProcess, method: int) -> str:
        key = (info.pid, method)
        cached = self._method_cache.get(key)
        if cached is not None:
            return cached
        mem, off = info.mem, info.offsets
        sym = SymbolReader(mem)
        cm = mem.word(method + off.const_method)
        if cm is None or not _plausible(cm):
            return ""
        cp = mem.word(cm + off.constants)
        if cp is None or not _plausible(cp):
            return ""
        idx_data = mem.read(cm + off.name_index, 4)
        if idx_data is None:
            return ""
        name_idx, _sig_idx = struct.unpack("<2H", idx_data)
        name_p = mem.word(cp + off.cp_base + name_idx * 8)
        if name_p is None or not _plausible(name_p):
            return ""
        name = sym.read_with(name_p, off.sym_len, off.sym_body)
        if not name:
            return ""
        if off.pool_holder >= 0:
            k = mem.word(cp + off.pool_holder)
            if k is not None and _plausible(k):
                ks = mem.word(k + off.klass_name)
                if ks is not None and _plausible(ks):
                    cls = sym.read_with(ks, off.sym_len, off.sym_body)
                    if cls:
                        name = f"{cls.replace('/', '.')}.{name}"
        self._method_cache.put(key, name)
        return name

    def stack_for(self, pid: int, bp: int, sp: int,
                  stack: bytes) -> List[Frame]:
        """Java interpreter frames (leaf-first) recovered from the
        copied stack's rbp chain, or []."""
        info = self._process(pid)
        if info is None:
            return []
        mem = info.mem

        def stack_word(addr: int) -> Optional[int]:
            rel = addr - sp
            if rel < 0 or rel + 8 > len(stack):
                return None
            return struct.unpack_from("<Q", stack, rel)[0]

        out: List[Frame] = []
        fp = bp
        hops = 0
        while fp and hops < _MAX_FRAMES:
            hops += 1
            method = stack_word(fp + METHOD_SLOT)
            nxt = stack_word(fp)
            if method is not None and _plausible(method):
                if info.offsets is None:
                    # first plausible candidate calibrates the build
                    cal = JvmCalibrator(mem, (info.jvm_lo, info.jvm_hi))
                    if cal.vptr_in_jvm(method):
                        build = getattr(info, "build", "")
                        offs = self._offsets_by_build.get(build) \
                            if build else None
                        if offs is None:
                            self.calibrations += 1
                            offs = cal.run(method)
                            if offs is not None and build:
                                self._offsets_by_build[build] = offs
                        info.offsets = offs
                if info.offsets is not None:
                    # validate vtable before trusting the slot: compiled
                    # and native frames also pass through this chain.
                    v = mem.word(method)
                    if v is not None and \
                            info.jvm_lo <= v < info.jvm_hi:
                        name = self._method_name(info, method)
                        if name:
                            out.append(Frame(
                                kind=FrameType.JVM, address=0,
                                mapping=_JVM_MAPPING,
                                function_name=name))
            if nxt is None or nxt <= fp:
                break
            fp = nxt
        if out:
            self.stacks_resolved += 1
        return out
