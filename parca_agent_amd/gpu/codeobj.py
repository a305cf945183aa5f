"""GPU code-object registry.

The gfx950 analog of the reference's cubin handling (reference:
parcagpu/parcagpu.go:231-277 handleCubinLoaded → StoreCubin → upload as
executable `cubin-%016x`): AMD code objects are ELF (e_machine=EM_AMDGPU),
loaded from file URIs or from target-process memory. The registry parses
them, computes a FileID, indexes kernel symbols for agent-side PC
symbolization, and reports each one once as an executable so the
debuginfo uploader can ship it.
"""

from __future__ import annotations

import ctypes
import ctypes.util
import errno
import logging
import os
import re
from dataclasses import dataclass, field
from typing import Callable, Dict, Optional
from urllib.parse import unquote, urlparse

from ..elf import ELFFile, SymbolIndex, file_id_from_bytes
from ..model import MappingFile

log = logging.getLogger("parca_agent_amd.gpu.codeobj")

_libc = ctypes.CDLL(None, use_errno=True)


class _iovec(ctypes.Structure):
    _fields_ = [("iov_base", ctypes.c_void_p),
                ("iov_len", ctypes.c_size_t)]


_eperm_warned = False


def read_process_memory(pid: int, addr: int, size: int) -> bytes:
    """process_vm_readv — the ReadCubinFromProcess analog. Called on hot
    paths (interpreter unwinding); keep it allocation-light."""
    buf = ctypes.create_string_buffer(size)
    local = _iovec(ctypes.cast(buf, ctypes.c_void_p), size)
    remote = _iovec(ctypes.c_void_p(addr), size)
    n = _libc.process_vm_readv(
        pid, ctypes.byref(local), 1, ctypes.byref(remote), 1, 0)
    if n < 0:
        err = ctypes.get_errno()
        if err == errno.EPERM:
            # Yama ptrace_scope >= 1 without CAP_SYS_PTRACE denies
            # non-ancestor reads; warn ONCE so operators see why
            # interpreter unwinding degrades. Processes carrying the
            # rocprofiler tool opt in via PR_SET_PTRACER_ANY (tool.cc).
            global _eperm_warned
            if not _eperm_warned:
                _eperm_warned = True
                log.warning(
                    "process_vm_readv denied (EPERM) for pid %d: grant "
                    "the agent CAP_SYS_PTRACE or lower "
                    "kernel.yama.ptrace_scope for interpreter unwinding "
                    "of processes not running the GPU tool", pid)
        raise OSError(err, os.strerror(err))
    return buf.raw[:n]


_URI_PARAM_RE = re.compile(r"(\w+)=(\d+|0x[0-9a-fA-F]+)")


def parse_code_object_uri(uri: str):
    """Parse rocprofiler code-object URIs:
    `file:///path#offset=4096&size=12345` or
    `memory://<pid>#offset=0x7f..&size=4096`."""
    parsed = urlparse(uri)
    params = {}
    if parsed.fragment:
        for m in _URI_PARAM_RE.finditer(parsed.fragment):
            params[m.group(1)] = int(m.group(2), 0)
    if parsed.scheme == "file":
        return ("file", unquote(parsed.path), params.get("offset", 0),
                params.get("size", 0))
    if parsed.scheme == "memory":
        pid = int(parsed.netloc) if parsed.netloc else 0
        return ("memory", str(pid), params.get("offset", 0),
                params.get("size", 0))
    return (parsed.scheme or "unknown", parsed.path, 0, 0)


@dataclass
class CodeObjectInfo:
    code_object_id: int
    pid: int
    load_base: int
    load_size: int
    load_delta: int
    uri: str
    file_id: str = ""
    build_id: str = ""
    gfx_arch: str = ""
    symbols: Optional[SymbolIndex] = None
    error: Optional[str] = None
    unloaded: bool = False
    data: Optional[bytes] = None  # retained until uploaded

    @property
    def mapping_file(self) -> MappingFile:
        # Executable identity `codeobj-<fileid>` mirrors the reference's
        # `cubin-%016x` naming (parcagpu.go:262).
        return MappingFile(
            file_id=self.file_id,
            path=f"codeobj-{self.file_id[:16] or self.code_object_id}",
            build_id=self.build_id)

    def vaddr_for_offset(self, code_object_offset: int) -> int:
        """PC-sample code_object_offset (relative to load_base) -> the ELF
        vaddr the code object was linked at: vaddr = pc - load_delta =
        offset + load_base - load_delta."""
        return code_object_offset + self.load_base - self.load_delta

    def symbolize(self, code_object_offset: int) -> str:
        if self.symbols is None:
            return ""
        sym = self.symbols.lookup(self.vaddr_for_offset(code_object_offset))
        return sym.name if sym else ""


class CodeObjectRegistry:
    def __init__(self, on_executable: Optional[Callable] = None) -> None:
        # (pid, code_object_id) -> info: code object ids are unique per
        # process (rocprofiler scope), not across the node.
        self._objects: Dict[tuple, CodeObjectInfo] = {}
        self.on_executable = on_executable
        self._reported: set = set()

    def load(self, pid: int, ev) -> CodeObjectInfo:
        """Register a CodeObjectLoad event (gpu/events.py dataclass)."""
        info = CodeObjectInfo(
            code_object_id=ev.code_object_id, pid=pid,
            load_base=ev.load_base, load_size=ev.load_size,
            load_delta=ev.load_delta, uri=ev.uri)
        try:
            data = self._fetch(pid, ev)
            if data:
                info.data = data
                info.file_id = file_id_from_bytes(data)
                elf = ELFFile.from_bytes(data, path=info.uri)
                info.build_id = elf.build_id() or ""
                info.symbols = SymbolIndex(elf.symbols())
                info.gfx_arch = _gfx_arch_from_flags(elf)
        except Exception as e:  # never let a bad code object kill the drain
            info.error = str(e)
            log.debug("code object %d load failed: %s",
                      ev.code_object_id, e)
        self._objects[(pid, ev.code_object_id)] = info
        self._maybe_report(info)
        return info

    def _fetch(self, pid: int, ev) -> Optional[bytes]:
        kind, where, offset, size = parse_code_object_uri(ev.uri)
        if kind == "file":
            with open(where, "rb") as fh:
                fh.seek(offset)
                return fh.read(size or None)
        if kind == "memory" or ev.memory_base:
            base = ev.memory_base or offset
            length = ev.memory_size or size
            if base and length:
                return read_process_memory(pid, base, length)
        return None

    def _maybe_report(self, info: CodeObjectInfo) -> None:
        if self.on_executable is None or not info.file_id:
            # No uploader wired (or unidentifiable): free the raw bytes
            # NOW — Tensile-scale code-object archives run to hundreds
            # of MB and were being retained for the life of the agent.
            info.data = None
            return
        if info.file_id in self._reported:
            info.data = None  # already shipped; free the bytes
            return
        self._reported.add(info.file_id)
        try:
            self.on_executable(info)
        except Exception:
            log.warning("code object executable report failed", exc_info=True)
        finally:
            info.data = None

    def unload(self, pid: int, code_object_id: int) -> None:
        info = self._objects.get((pid, code_object_id))
        if info:
            info.unloaded = True

    def get(self, pid: int, code_object_id: int) -> Optional[CodeObjectInfo]:
        return self._objects.get((pid, code_object_id))

    def drop_process(self, pid: int) -> None:
        for key in [k for k in self._objects if k[0] == pid]:
            del self._objects[key]

    def __len__(self) -> int:
        return len(self._objects)


def _gfx_arch_from_flags(elf: ELFFile) -> str:
    """Best-effort gfx arch; EM_AMDGPU e_flags low byte is the mach id
    (gfx950 = 0x4f per LLVM EF_AMDGPU_MACH)."""
    # e_flags is not stored on ELFFile; derive from sections heuristically.
    note = elf.section(".note")
    return "amdgcn" if elf.e_machine == 224 else ""
