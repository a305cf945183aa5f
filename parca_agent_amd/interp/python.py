"""CPython interpreter unwinder: remote Python stacks for sampled
processes.

The reference gets interpreter unwinding (Python, Ruby, JVM, ...) from
its eBPF fork (SURVEY.md §2.9 "interpreter unwinders"). This native
rebuild reads the target's CPython state with process_vm_readv, walking
_PyRuntime -> PyInterpreterState -> PyThreadState -> frame chain and
decoding code-object names, and attaches python frames to CPU samples of
python threads.

Struct offsets are NOT hardcoded per version: the unwinder calibrates
them at startup against the agent's OWN interpreter — ctypes provides
ground-truth addresses (PyThreadState_Get, id(frame), id(code), ...),
and the calibrator scans the surrounding structs for those known pointer
values. The calibrated offsets are then applied only to target processes
whose python binary/libpython has the SAME FileID as the agent's
(identical build => identical layout); other interpreters are skipped
rather than misread. Works on 3.9-3.10 frame layouts (PyFrameObject
chain); 3.11+ would calibrate the _PyInterpreterFrame chain the same way
and is left as a follow-up.
"""

from __future__ import annotations

import ctypes
import logging
import struct
import sys
import threading
from dataclasses import dataclass
from typing import List, Optional, Tuple

from ..elf import ELFFile, file_id
from ..gpu.codeobj import read_process_memory
from ..lru import LRU
from ..model import Frame, FrameType, MappingFile

log = logging.getLogger("parca_agent_amd.interp.python")

_WORD = 8


def _read_words(data: bytes) -> List[int]:
    n = len(data) // _WORD
    return list(struct.unpack(f"<{n}Q", data[: n * _WORD]))


def _scan_for(haystack_addr: int, data: bytes, needle: int) -> Optional[int]:
    """Offset of the 8-byte little-endian `needle` within data."""
    packed = struct.pack("<Q", needle)
    idx = data.find(packed)
    while idx >= 0:
        if idx % _WORD == 0:
            return idx
        idx = data.find(packed, idx + 1)
    return None


def _self_read(addr: int, size: int) -> bytes:
    """Safe self-memory read: process_vm_readv fails with EFAULT on bad
    addresses where ctypes.string_at would segfault (the calibrator walks
    candidate pointers)."""
    import os

    return read_process_memory(os.getpid(), addr, size)


@dataclass
class PyOffsets:
    """Calibrated offsets; all relative to their containing struct."""

    runtime_interp_head: int = -1
    interp_tstate_head: int = -1
    interp_next: int = -1
    tstate_next: int = -1
    tstate_frame: int = -1
    tstate_thread_id: int = -1
    tstate_native_id: int = -1  # -1 on 3.10 (absent)
    frame_back: int = -1
    frame_code: int = -1
    code_name: int = -1
    code_filename: int = -1
    unicode_length: int = 16   # PyASCIIObject.length (stable)
    unicode_data: int = -1     # compact-ASCII payload offset
    # glibc struct pthread: kernel tid at this offset from pthread_t
    # (which is the struct's own address). Lets 3.10 — where
    # PyThreadState has no native_thread_id — match tstates to sampled
    # tids EXACTLY instead of by stack-distance heuristics.
    pthread_tid: int = -1

    def complete(self) -> bool:
        return all(v >= 0 for v in (
            self.runtime_interp_head, self.interp_tstate_head,
            self.tstate_next, self.tstate_frame, self.tstate_thread_id,
            self.frame_back, self.frame_code, self.code_name,
            self.code_filename, self.unicode_data))


def calibrate() -> Optional[PyOffsets]:
    """Derive offsets from this process's interpreter."""
    try:
        api = ctypes.pythonapi
        api.PyInterpreterState_Head.restype = ctypes.c_void_p
        api.PyThreadState_Get.restype = ctypes.c_void_p

        off = PyOffsets()

        interp = api.PyInterpreterState_Head()
        runtime_addr = _find_pyruntime_self()
        if runtime_addr:
            data = _self_read(runtime_addr, 2048)
            o = _scan_for(runtime_addr, data, interp)
            if o is not None:
                off.runtime_interp_head = o

        # Ground truth from a helper thread (gives a second tstate and a
        # deep frame chain to scan against).
        info = {}
        ready = threading.Event()
        release = threading.Event()

        def helper():
            def inner():
                info["tstate"] = api.PyThreadState_Get()
                info["ident"] = threading.get_ident()
                info["native"] = getattr(threading, "get_native_id",
                                         lambda: -1)()
                info["frame"] = id(sys._getframe(0))
                info["frame_back"] = id(sys._getframe(1))
                info["code"] = id(sys._getframe(0).f_code)
                ready.set()
                release.wait(10)

            inner()

        t = threading.Thread(target=helper)
        t.start()
        ready.wait(10)

        main_tstate = api.PyThreadState_Get()
        helper_tstate = info["tstate"]

        interp_data = _self_read(interp, 4096)
        # tstate_head points at the most recently created tstate — but
        # threads come and go; scan for either known tstate.
        for needle in (helper_tstate, main_tstate):
            o = _scan_for(interp, interp_data, needle)
            if o is not None:
                off.interp_tstate_head = o
                break

        # Within the helper tstate: next (-> another tstate: scan for
        # main's), frame, thread_id, native id.
        ts_data = _self_read(helper_tstate, 1024)
        o = _scan_for(helper_tstate, ts_data, main_tstate)
        if o is not None:
            off.tstate_next = o
        o = _scan_for(helper_tstate, ts_data, info["ident"])
        if o is not None:
            off.tstate_thread_id = o
        if info["native"] != -1:
            o = _scan_for(helper_tstate, ts_data, info["native"])
            if o is not None and o != off.tstate_thread_id:
                off.tstate_native_id = o

        frame_data = _self_read(info["frame"], 512)
        o = _scan_for(info["frame"], frame_data, info["frame_back"])
        if o is not None:
            off.frame_back = o
        o = _scan_for(info["frame"], frame_data, info["code"])
        if o is not None:
            off.frame_code = o

        # tstate->frame points at the helper thread's CURRENT top frame
        # (deep inside Event.wait), not at inner()'s frame — so find the
        # offset whose value, walked via frame_back, reaches the known
        # ancestor frame id within a few hops.
        if off.frame_back >= 0:
            words = _read_words(ts_data)
            for i, v in enumerate(words):
                if not (0x1000 < v < (1 << 48)):
                    continue
                cur = v
                for _ in range(64):
                    if cur == info["frame"]:
                        off.tstate_frame = i * _WORD
                        break
                    try:
                        cur = struct.unpack(
                            "<Q", _self_read(cur + off.frame_back, 8))[0]
                    except (OSError, ValueError, ctypes.ArgumentError):
                        break
                    if not (0x1000 < cur < (1 << 48)):
                        break
                if off.tstate_frame >= 0:
                    break

        code_obj = None
        for frame_attr in (sys._getframe(0),):
            code_obj = frame_attr.f_code
        code_data = _self_read(id(code_obj), 512)
        o = _scan_for(id(code_obj), code_data, id(code_obj.co_name))
        if o is not None:
            off.code_name = o
        o = _scan_for(id(code_obj), code_data, id(code_obj.co_filename))
        if o is not None:
            off.code_filename = o

        # Unicode payload: known-content ASCII string.
        probe = sys.intern("parca_unicode_probe_0123")
        u_data = _self_read(id(probe), 128)
        idx = u_data.find(b"parca_unicode_probe_0123")
        if idx > 0:
            off.unicode_data = idx
        # length field: scan for the value 24 at an 8-aligned offset
        # below the payload.
        for cand in range(8, off.unicode_data if off.unicode_data > 0
                          else 64, 8):
            (v,) = struct.unpack_from("<Q", u_data, cand)
            if v == len("parca_unicode_probe_0123"):
                off.unicode_length = cand
                break

        # glibc pthread tid offset: pthread_t == address of struct
        # pthread; scan it for the helper's kernel tid, verified against
        # the main thread at the same offset.
        if info["native"] != -1:
            helper_pt = info["ident"]
            pt_data = _self_read(helper_pt, 2048)
            main_pt = threading.get_ident()
            main_nid = threading.get_native_id()
            needle = struct.pack("<I", info["native"])
            idx = pt_data.find(needle)
            while idx >= 0:
                if idx % 4 == 0:
                    try:
                        main_data = _self_read(main_pt + idx, 4)
                        if struct.unpack("<I", main_data)[0] == main_nid:
                            off.pthread_tid = idx
                            break
                    except OSError:
                        pass
                idx = pt_data.find(needle, idx + 1)

        release.set()
        t.join(timeout=5)

        if not off.complete():
            log.info("python offset calibration incomplete: %s", off)
            return None
        return off
    except Exception:
        log.warning("python offset calibration failed", exc_info=True)
        return None


def _runtime_addr_for(path: str, map_start: int,
                      map_file_off: int) -> Optional[int]:
    """Remote address of _PyRuntime given one load mapping of the python
    module: load_bias = map_start - vaddr(map_file_off); addr = bias +
    sym_vaddr. (_PyRuntime lives in the rw data segment, so naive
    file-offset arithmetic against the text mapping is wrong.)"""
    try:
        with ELFFile.open(path) as elf:
            map_vaddr = elf.vaddr_for_file_offset(map_file_off)
            if map_vaddr is None:
                return None
            bias = map_start - map_vaddr
            for sym in elf.symbols():
                if sym.name == "_PyRuntime":
                    return bias + sym.value
    except (OSError, ValueError):
        return None
    return None


def _find_pyruntime_self() -> Optional[int]:
    """Address of _PyRuntime in this process (symbol + load bias)."""
    exe_path, base, file_off = _python_module_of_self()
    if exe_path is None:
        return None
    return _runtime_addr_for(exe_path, base, file_off)


def _python_module_of(pid) -> Tuple[Optional[str], int, int]:
    """(path, map_start, file_offset) of the lowest readable mapping of
    the CPython binary/libpython in `pid`. The SAME selection must be
    used for self-calibration and target probing so any systematic bias
    cancels."""
    best = None
    try:
        with open(f"/proc/{pid}/maps") as fh:
            for line in fh:
                parts = line.split()
                if len(parts) < 6:
                    continue
                perms, path = parts[1], parts[5]
                if not perms.startswith("r"):
                    continue  # guard/PROT_NONE pages distort the base
                base_name = path.rsplit("/", 1)[-1]
                if "libpython3" in base_name or \
                        base_name.startswith("python3") or \
                        base_name == "python":
                    start = int(parts[0].split("-")[0], 16)
                    off = int(parts[2], 16)
                    if best is None or start < best[1]:
                        best = (path, start, off)
    except OSError:
        return (None, 0, 0)
    return best if best else (None, 0, 0)


def _python_module_of_self():
    return _python_module_of("self")


@dataclass
class PyProcess:
    pid: int
    runtime_addr: int
    usable: bool


class PythonUnwinder:
    """Per-agent singleton; resolves python stacks for sampled tids."""

    def __init__(self, processes=None) -> None:
        self.offsets = calibrate()
        self.processes = processes
        self._self_py_fileid = ""
        path, _, _ = _python_module_of_self()
        if path:
            try:
                self._self_py_fileid = file_id(path)
            except OSError:
                pass
        self._procs: LRU[int, Optional[PyProcess]] = LRU(
            2048, ttl_seconds=300)
        self._str_cache: LRU[Tuple[int, int], str] = LRU(65536)
        self._ptid_cache: LRU[Tuple[int, int], int] = LRU(
            16384, ttl_seconds=60)
        # (pid, code addr) -> (name, filename)
        self._code_cache: LRU[Tuple[int, int], tuple] = LRU(65536)
        self.stacks_resolved = 0
        self.resolve_failures = 0

    @property
    def available(self) -> bool:
        return self.offsets is not None and bool(self._self_py_fileid)

    # -- process detection -------------------------------------------------

    def _probe_process(self, pid: int) -> Optional[PyProcess]:
        """Find _PyRuntime in the target iff it runs our exact
        interpreter build."""
        try:
            path, base, file_off = _python_module_of(pid)
            if path is None:
                return None
            if file_id(path) != self._self_py_fileid:
                return None  # different build: offsets not trustworthy
            addr = _runtime_addr_for(path, base, file_off)
            if addr is None:
                return None
            return PyProcess(pid=pid, runtime_addr=addr, usable=True)
        except (OSError, ValueError):
            return None

    def _process(self, pid: int) -> Optional[PyProcess]:
        cached = self._procs.get(pid, default="MISS")
        if cached != "MISS":
            return cached
        info = self._probe_process(pid)
        self._procs.put(pid, info)
        return info

    # -- remote reads ------------------------------------------------------

    def _word(self, pid: int, addr: int) -> Optional[int]:
        if addr == 0 or addr > (1 << 48):
            return None
        try:
            data = read_process_memory(pid, addr, 8)
        except OSError:
            return None
        if len(data) < 8:
            return None
        return struct.unpack("<Q", data)[0]

    def _struct_words(self, pid: int, addr: int,
                      offsets: Tuple[int, ...]) -> Optional[List[int]]:
        """One remote read covering all requested field offsets."""
        if addr == 0 or addr > (1 << 48):
            return None
        span = max(offsets) + 8
        try:
            data = read_process_memory(pid, addr, span)
        except OSError:
            return None
        if len(data) < span:
            return None
        return [struct.unpack_from("<Q", data, o)[0] for o in offsets]

    def _string(self, pid: int, addr: int) -> str:
        if addr == 0:
            return ""
        key = (pid, addr)
        cached = self._str_cache.get(key)
        if cached is not None:
            return cached
        off = self.offsets
        try:
            header = read_process_memory(
                pid, addr, off.unicode_data + 0)
            (length,) = struct.unpack_from("<Q", header, off.unicode_length)
            if length > 512:
                length = 512
            data = read_process_memory(pid, addr + off.unicode_data,
                                       int(length))
            s = data.decode("utf-8", "replace")
        except (OSError, struct.error):
            s = ""
        self._str_cache.put(key, s)
        return s

    # -- stack walk --------------------------------------------------------

    def _find_tstate(self, pid: int, runtime_addr: int, tid: int,
                     sp: int) -> Optional[int]:
        off = self.offsets
        interp = self._word(pid, runtime_addr + off.runtime_interp_head)
        if not interp:
            return None
        field_offs = (off.tstate_next, off.tstate_thread_id) + (
            (off.tstate_native_id,) if off.tstate_native_id >= 0 else ())
        tstate = self._word(pid, interp + off.interp_tstate_head)
        tstates: List[Tuple[int, int]] = []  # (tstate, pthread_t)
        hops = 0
        while tstate and hops < 512:
            fields = self._struct_words(pid, tstate, field_offs)
            if fields is None:
                break
            if off.tstate_native_id >= 0:
                if (fields[2] & 0xFFFFFFFF) == tid:
                    return tstate
            tstates.append((tstate, fields[1]))
            tstate = fields[0]
            hops += 1
        if not tstates:
            return None
        if len(tstates) == 1:
            return tstates[0][0]
        if off.tstate_native_id >= 0:
            return None  # 3.11+: exact match required, none found
        # 3.10: thread_id is pthread_self() == the address of glibc's
        # struct pthread, which stores the kernel tid at the calibrated
        # offset — exact matching without native_thread_id.
        if off.pthread_tid >= 0:
            # cache: (pid, pthread_addr) -> kernel tid
            for ts, ptid in tstates:
                if not ptid:
                    continue
                key = (pid, ptid)
                ktid = self._ptid_cache.get(key)
                if ktid is None:
                    try:
                        data = read_process_memory(
                            pid, ptid + off.pthread_tid, 4)
                        ktid = struct.unpack("<I", data)[0]
                    except OSError:
                        continue
                    self._ptid_cache.put(key, ktid)
                if ktid == tid:
                    return ts
            # A sampled thread with no tstate (OMP/IO worker): no frames.
            if tid != pid:
                return None
        # Fallbacks: main thread's tstate was created first and sits at
        # the TAIL of the head-linked list; otherwise pick the tstate
        # whose pthread struct sits just above the sampled SP (same
        # thread stack region).
        if tid == pid:
            return tstates[-1][0]
        best = None
        for ts, ptid in tstates:
            if ptid and sp and 0 < ptid - sp < (8 << 20):
                if best is None or ptid < best[0]:
                    best = (ptid, ts)
        return best[1] if best else None

    def stack_for(self, pid: int, tid: int, sp: int,
                  max_frames: int = 64) -> List[Frame]:
        """Python frames (leaf-first) for the sampled thread, or []."""
        if not self.available:
            return []
        info = self._process(pid)
        if info is None or not info.usable:
            return []
        off = self.offsets
        tstate = self._find_tstate(pid, info.runtime_addr, tid, sp)
        if tstate is None:
            self.resolve_failures += 1
            return []
        frame = self._word(pid, tstate + off.tstate_frame)
        out: List[Frame] = []
        hops = 0
        mapping = MappingFile(path="<python>")
        frame_offs = (off.frame_back, off.frame_code)
        while frame and hops < max_frames:
            fields = self._struct_words(pid, frame, frame_offs)
            if fields is None:
                break
            back, code = fields
            if code:
                cached = self._code_cache.get((pid, code))
                if cached is None:
                    cw = self._struct_words(
                        pid, code, (off.code_name, off.code_filename))
                    if cw is not None:
                        name = self._string(pid, cw[0] or 0)
                        filename = self._string(pid, cw[1] or 0)
                        cached = (name, filename)
                    else:
                        cached = ("", "")
                    self._code_cache.put((pid, code), cached)
                name, filename = cached
                if name:
                    out.append(Frame(
                        kind=FrameType.PYTHON, address=0, mapping=mapping,
                        function_name=name, source_file=filename))
            frame = back
            hops += 1
        if out:
            self.stacks_resolved += 1
        else:
            self.resolve_failures += 1
        return out
