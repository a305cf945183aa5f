"""Core trace data model — the libpf.Trace/Frame analog.

Shapes mirror the fork API the reference consumes (SURVEY.md §2.9:
libpf.Trace{Frames}, Frame{Type, AddressOrLineno, FunctionName, ...},
samples.TraceEventMeta) so the reporter semantics carry over 1:1, but the
types are defined fresh for this agent.
"""

from __future__ import annotations

import enum
import hashlib
from dataclasses import dataclass, field
from typing import Dict, Optional, Tuple


class FrameType(enum.Enum):
    NATIVE = "native"
    KERNEL = "kernel"
    PYTHON = "python"
    PERL = "perl"
    RUBY = "ruby"
    JVM = "jvm"
    PHP = "php"
    GPU_PC = "amdgpu_pc"      # GPU program-counter frame (CUDAPCFrame analog)
    GPU_KERNEL = "amdgpu"     # GPU kernel name pseudo-frame
    JIT = "jit"               # perf-map-resolved JIT frame (node/JVM/...)
    ERROR = "error"
    UNKNOWN = "unknown"


class TraceOrigin(enum.Enum):
    # Mirrors support.TraceOrigin* (SURVEY.md §2.9).
    SAMPLING = "sampling"
    OFF_CPU = "off_cpu"
    MEMORY = "memory"
    GPU_KERNEL = "gpu_kernel"   # reference: TraceOriginCuda
    GPU_PC = "gpu_pc"           # reference: TraceOriginGpuPC
    PROBE = "probe"


@dataclass(frozen=True, slots=True)
class MappingFile:
    """Identity of the file backing a mapping (libpf.FrameMappingFile)."""

    file_id: str = ""       # agent-computed stable hash (elf.file_id)
    path: str = ""
    build_id: str = ""      # GNU build-id hex, "" if absent

    @property
    def id_label(self) -> str:
        return self.build_id or self.file_id


@dataclass(frozen=True, slots=True)
class Frame:
    kind: FrameType
    # File-relative address for native frames, line number for interpreted
    # frames, kernel-relative addr for kernel frames (AddressOrLineno).
    address: int = 0
    mapping: Optional[MappingFile] = None
    function_name: str = ""
    source_file: str = ""
    source_line: int = 0


@dataclass(frozen=True, slots=True)
class Trace:
    """Leaf-first frame list; hash is the stack-dedup key (TraceHash)."""

    frames: Tuple[Frame, ...]
    custom_labels: Tuple[Tuple[str, str], ...] = ()

    _hash_cache: Optional[bytes] = field(default=None, compare=False,
                                         repr=False, hash=False)

    def trace_hash(self) -> bytes:
        """16-byte stable hash over frame identities (traceutil.HashTrace).
        Memoized: the reporter calls this once per sample and cached Trace
        objects are reused across identical stacks."""
        if self._hash_cache is not None:
            return self._hash_cache
        h = hashlib.blake2b(digest_size=16)
        for f in self.frames:
            h.update(f.kind.value.encode())
            h.update(f.address.to_bytes(8, "little"))
            if f.mapping is not None:
                h.update(f.mapping.id_label.encode())
            h.update(f.function_name.encode())
            h.update(b"\x00")
        for k, v in self.custom_labels:
            # surrogatepass: label bytes from bad remote-memory decodes
            # must still hash deterministically (sanitization happens at
            # report time, not here).
            h.update(k.encode("utf-8", "surrogatepass"))
            h.update(b"=")
            h.update(v.encode("utf-8", "surrogatepass"))
        digest = h.digest()
        object.__setattr__(self, "_hash_cache", digest)
        return digest


@dataclass(slots=True)
class TraceEventMeta:
    """Per-event metadata (samples.TraceEventMeta analog)."""

    timestamp_ns: int = 0
    comm: str = ""
    process_name: str = ""
    executable_path: str = ""
    pid: int = 0
    tid: int = 0
    cpu: int = -1
    origin: TraceOrigin = TraceOrigin.SAMPLING
    value: int = 1
    env_vars: Dict[str, str] = field(default_factory=dict)
    # GPU-origin extras
    gpu_id: int = -1
    kernel_name: str = ""
    # Origin-data overrides: the memory origin spans FOUR sample types
    # (alloc/inuse x space/objects — reference OriginData dispatch,
    # parca_reporter.go:398-425); producers set these to pick one and
    # to carry the heap-sampler period.
    sample_type: Optional["SampleType"] = None
    period: int = 0


# Sample-type table per origin (reference: parca_reporter.go:389-455).
@dataclass(frozen=True, slots=True)
class SampleType:
    sample_type: str
    sample_unit: str
    period_type: str
    period_unit: str
    temporality: str = "delta"


def sample_type_for(origin: TraceOrigin, merge_gpu: bool = False) -> SampleType:
    if origin == TraceOrigin.SAMPLING:
        return SampleType("samples", "count", "cpu", "nanoseconds")
    if origin == TraceOrigin.OFF_CPU:
        # Reference period metadata is samples/count, not wallclock
        # (parca_reporter.go:394-397 writeSample(..., "wallclock",
        # "nanoseconds", "samples", "count")).
        return SampleType("wallclock", "nanoseconds", "samples", "count")
    if origin == TraceOrigin.GPU_KERNEL:
        if merge_gpu:
            return SampleType("gpu_time", "nanoseconds", "gpu_time", "nanoseconds")
        return SampleType("gpu_kernel_time", "nanoseconds",
                          "gpu_kernel_time", "nanoseconds")
    if origin == TraceOrigin.GPU_PC:
        if merge_gpu:
            return SampleType("gpu_time", "nanoseconds", "gpu_time", "nanoseconds")
        return SampleType("gpu_pcsample", "count", "gpu_pcsample", "nanoseconds")
    if origin == TraceOrigin.MEMORY:
        return SampleType("inuse_space", "bytes", "space", "bytes")
    if origin == TraceOrigin.PROBE:
        return SampleType("probe", "nanoseconds", "probe", "nanoseconds")
    raise ValueError(f"no sample type for origin {origin}")


def make_kernel_frame(address: int) -> Frame:
    return Frame(kind=FrameType.KERNEL, address=address,
                 mapping=MappingFile(path="[kernel.kallsyms]"))


def make_native_frame(mapping: MappingFile, address: int,
                      function_name: str = "", source_file: str = "",
                      source_line: int = 0) -> Frame:
    return Frame(kind=FrameType.NATIVE, address=address, mapping=mapping,
                 function_name=function_name, source_file=source_file,
                 source_line=source_line)
