"""GPU ring event decoding: Python mirrors of csrc/rocprof/ring.h layouts.

The u32 event-type discriminator dispatch mirrors the reference's
parcagpu reader (reference: parcagpu/parcagpu.go:149-213). Layouts are
verified against the C++ sizeof()s in tests/test_gpu_ring.py.
"""

from __future__ import annotations

import struct
from dataclasses import dataclass
from typing import Tuple

import numpy as np

EV_KERNEL_DISPATCH = 1
EV_CODE_OBJECT_LOAD = 2
EV_CODE_OBJECT_UNLOAD = 3
EV_KERNEL_SYMBOL = 4
EV_PC_SAMPLE_BATCH = 5
EV_GPU_CONFIG = 6
EV_ERROR = 7
EV_LAUNCH_STACK = 8

KERNEL_DISPATCH_FMT = struct.Struct("<6Q2I3I3I2I")
CODE_OBJECT_LOAD_FMT = struct.Struct("<QQQqQQII")
CODE_OBJECT_UNLOAD_FMT = struct.Struct("<Q")
KERNEL_SYMBOL_FMT = struct.Struct("<QQQII")
PC_SAMPLE_BATCH_HEADER_FMT = struct.Struct("<II")
GPU_CONFIG_FMT = struct.Struct("<4IQd")
ERROR_FMT = struct.Struct("<II")
LAUNCH_STACK_FMT = struct.Struct("<QQII")

# numpy dtype of one PCSample (ring.h PCSample, 64 bytes).
PC_SAMPLE_DTYPE = np.dtype([
    ("code_object_id", "<u8"),
    ("code_object_offset", "<u8"),
    ("timestamp", "<u8"),
    ("exec_mask", "<u8"),
    ("dispatch_id", "<u8"),
    ("correlation_id", "<u8"),
    ("hw_id", "<u8"),
    ("wave_in_group", "<u4"),
    ("flags", "<u4"),
])


@dataclass
class KernelDispatch:
    correlation_id: int
    dispatch_id: int
    kernel_id: int
    start_ns: int
    end_ns: int
    tid: int
    gpu_index: int
    pid: int
    grid: Tuple[int, int, int]
    workgroup: Tuple[int, int, int]
    private_segment_size: int
    group_segment_size: int

    @property
    def duration_ns(self) -> int:
        return max(self.end_ns - self.start_ns, 0)


@dataclass
class CodeObjectLoad:
    code_object_id: int
    load_base: int
    load_size: int
    load_delta: int
    memory_base: int
    memory_size: int
    storage_type: int
    uri: str


@dataclass
class KernelSymbol:
    kernel_id: int
    code_object_id: int
    kernel_object: int
    name: str


@dataclass
class GpuConfig:
    gpu_index: int
    method: int
    unit: int
    interval: int
    ns_per_sample: float


@dataclass
class LaunchStack:
    correlation_id: int
    tid: int
    pid: int
    ips: Tuple[int, ...]


@dataclass
class RingError:
    code: int
    message: str


def decode_kernel_dispatch(payload: bytes) -> KernelDispatch:
    v = KERNEL_DISPATCH_FMT.unpack_from(payload)
    return KernelDispatch(
        correlation_id=v[0], dispatch_id=v[1], kernel_id=v[2],
        start_ns=v[3], end_ns=v[4], tid=v[5], gpu_index=v[6], pid=v[7],
        grid=(v[8], v[9], v[10]), workgroup=(v[11], v[12], v[13]),
        private_segment_size=v[14], group_segment_size=v[15])


def decode_code_object_load(payload: bytes) -> CodeObjectLoad:
    v = CODE_OBJECT_LOAD_FMT.unpack_from(payload)
    uri_len = v[7]
    base = CODE_OBJECT_LOAD_FMT.size
    uri = payload[base : base + uri_len].decode("utf-8", "replace")
    return CodeObjectLoad(
        code_object_id=v[0], load_base=v[1], load_size=v[2], load_delta=v[3],
        memory_base=v[4], memory_size=v[5], storage_type=v[6], uri=uri)


def decode_code_object_unload(payload: bytes) -> int:
    return CODE_OBJECT_UNLOAD_FMT.unpack_from(payload)[0]


def decode_kernel_symbol(payload: bytes) -> KernelSymbol:
    v = KERNEL_SYMBOL_FMT.unpack_from(payload)
    base = KERNEL_SYMBOL_FMT.size
    name = payload[base : base + v[3]].decode("utf-8", "replace")
    return KernelSymbol(kernel_id=v[0], code_object_id=v[1],
                        kernel_object=v[2], name=name)


def decode_pc_sample_batch(payload: bytes) -> Tuple[int, np.ndarray]:
    """Returns (gpu_index, structured array of PC samples) — zero-copy."""
    gpu_index, count = PC_SAMPLE_BATCH_HEADER_FMT.unpack_from(payload)
    base = PC_SAMPLE_BATCH_HEADER_FMT.size
    arr = np.frombuffer(payload, dtype=PC_SAMPLE_DTYPE, count=count,
                        offset=base)
    return gpu_index, arr


def decode_gpu_config(payload: bytes) -> GpuConfig:
    v = GPU_CONFIG_FMT.unpack_from(payload)
    return GpuConfig(gpu_index=v[0], method=v[1], unit=v[2], interval=v[4],
                     ns_per_sample=v[5])


def decode_error(payload: bytes) -> RingError:
    code, msg_len = ERROR_FMT.unpack_from(payload)
    base = ERROR_FMT.size
    return RingError(code=code,
                     message=payload[base : base + msg_len].decode(
                         "utf-8", "replace"))


def decode_launch_stack(payload: bytes) -> LaunchStack:
    v = LAUNCH_STACK_FMT.unpack_from(payload)
    n = v[3]
    base = LAUNCH_STACK_FMT.size
    ips = struct.unpack_from(f"<{n}Q", payload, base) if n else ()
    return LaunchStack(correlation_id=v[0], tid=v[1], pid=v[2], ips=ips)


# -- encoding (test + synthetic-event support) ----------------------------


def encode_kernel_dispatch(ev: KernelDispatch) -> bytes:
    return KERNEL_DISPATCH_FMT.pack(
        ev.correlation_id, ev.dispatch_id, ev.kernel_id, ev.start_ns,
        ev.end_ns, ev.tid, ev.gpu_index, ev.pid, *ev.grid, *ev.workgroup,
        ev.private_segment_size, ev.group_segment_size)


def encode_code_object_load(ev: CodeObjectLoad) -> bytes:
    uri = ev.uri.encode()
    return CODE_OBJECT_LOAD_FMT.pack(
        ev.code_object_id, ev.load_base, ev.load_size, ev.load_delta,
        ev.memory_base, ev.memory_size, ev.storage_type, len(uri)) + uri


def encode_kernel_symbol(ev: KernelSymbol) -> bytes:
    name = ev.name.encode()
    return KERNEL_SYMBOL_FMT.pack(ev.kernel_id, ev.code_object_id,
                                  ev.kernel_object, len(name), 0) + name


def encode_pc_sample_batch(gpu_index: int, samples: np.ndarray) -> bytes:
    assert samples.dtype == PC_SAMPLE_DTYPE
    return PC_SAMPLE_BATCH_HEADER_FMT.pack(gpu_index, len(samples)) + \
        samples.tobytes()


def encode_gpu_config(ev: GpuConfig) -> bytes:
    return GPU_CONFIG_FMT.pack(ev.gpu_index, ev.method, ev.unit, 0,
                               ev.interval, ev.ns_per_sample)


def encode_launch_stack(ev: LaunchStack) -> bytes:
    return LAUNCH_STACK_FMT.pack(ev.correlation_id, ev.tid, ev.pid,
                                 len(ev.ips)) + \
        struct.pack(f"<{len(ev.ips)}Q", *ev.ips)


def encode_error(ev: RingError) -> bytes:
    msg = ev.message.encode()
    return ERROR_FMT.pack(ev.code, len(msg)) + msg
