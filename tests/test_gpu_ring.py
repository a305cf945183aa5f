"""GPU ring + event-layout tests (golden-bytes strategy, SURVEY.md §4:
'GPU paths testable by injecting synthetic ring records')."""

import os
import struct

import numpy as np
import pytest

from parca_agent_amd.gpu import events as ev


def _native():
    from parca_agent_amd.native import gpu
    return gpu()


@pytest.fixture
def ring(tmp_path):
    g = _native()
    path = str(tmp_path / "test.ring")
    prod = g.TestRingProducer(path, 1 << 16)
    cons = g.RingConsumer(path)
    return g, prod, cons


def test_struct_sizes_match_cpp():
    g = _native()
    assert ev.KERNEL_DISPATCH_FMT.size == g.SIZEOF_KERNEL_DISPATCH
    assert ev.CODE_OBJECT_LOAD_FMT.size == g.SIZEOF_CODE_OBJECT_LOAD
    assert ev.CODE_OBJECT_UNLOAD_FMT.size == g.SIZEOF_CODE_OBJECT_UNLOAD
    assert ev.KERNEL_SYMBOL_FMT.size == g.SIZEOF_KERNEL_SYMBOL
    assert ev.PC_SAMPLE_DTYPE.itemsize == g.SIZEOF_PC_SAMPLE
    assert ev.PC_SAMPLE_BATCH_HEADER_FMT.size == g.SIZEOF_PC_SAMPLE_BATCH_HEADER
    assert ev.GPU_CONFIG_FMT.size == g.SIZEOF_GPU_CONFIG
    assert ev.ERROR_FMT.size == g.SIZEOF_ERROR
    assert ev.LAUNCH_STACK_FMT.size == g.SIZEOF_LAUNCH_STACK


def test_event_type_constants_match():
    g = _native()
    assert ev.EV_KERNEL_DISPATCH == g.EV_KERNEL_DISPATCH
    assert ev.EV_CODE_OBJECT_LOAD == g.EV_CODE_OBJECT_LOAD
    assert ev.EV_PC_SAMPLE_BATCH == g.EV_PC_SAMPLE_BATCH
    assert ev.EV_GPU_CONFIG == g.EV_GPU_CONFIG
    assert ev.EV_LAUNCH_STACK == g.EV_LAUNCH_STACK


def test_kernel_dispatch_roundtrip(ring):
    g, prod, cons = ring
    d = ev.KernelDispatch(
        correlation_id=42, dispatch_id=7, kernel_id=3,
        start_ns=1000, end_ns=5500, tid=999, gpu_index=2, pid=1234,
        grid=(256, 1, 1), workgroup=(64, 1, 1),
        private_segment_size=0, group_segment_size=1024)
    prod.write(g.EV_KERNEL_DISPATCH, ev.encode_kernel_dispatch(d))
    [(rtype, payload)] = cons.drain()
    assert rtype == ev.EV_KERNEL_DISPATCH
    out = ev.decode_kernel_dispatch(payload)
    assert out == d
    assert out.duration_ns == 4500


def test_code_object_and_symbol_roundtrip(ring):
    g, prod, cons = ring
    load = ev.CodeObjectLoad(
        code_object_id=9, load_base=0x7f0000000000, load_size=0x4000,
        load_delta=0x7f0000000000 - 0x1000, memory_base=0, memory_size=0,
        storage_type=1, uri="file:///tmp/app#offset=4096&size=16384")
    prod.write(g.EV_CODE_OBJECT_LOAD, ev.encode_code_object_load(load))
    sym = ev.KernelSymbol(kernel_id=5, code_object_id=9,
                          kernel_object=0xdead, name="my_kernel.kd")
    prod.write(g.EV_KERNEL_SYMBOL, ev.encode_kernel_symbol(sym))
    recs = cons.drain()
    assert len(recs) == 2
    out_load = ev.decode_code_object_load(recs[0][1])
    assert out_load == load
    out_sym = ev.decode_kernel_symbol(recs[1][1])
    assert out_sym == sym


def test_pc_sample_batch_roundtrip(ring):
    g, prod, cons = ring
    samples = np.zeros(10, dtype=ev.PC_SAMPLE_DTYPE)
    samples["code_object_id"] = 9
    samples["code_object_offset"] = np.arange(10) * 64
    samples["exec_mask"] = (1 << 64) - 1
    samples["dispatch_id"] = 7
    prod.write(g.EV_PC_SAMPLE_BATCH, ev.encode_pc_sample_batch(3, samples))
    [(rtype, payload)] = cons.drain()
    gpu_index, out = ev.decode_pc_sample_batch(payload)
    assert gpu_index == 3
    assert len(out) == 10
    np.testing.assert_array_equal(out["code_object_offset"],
                                  samples["code_object_offset"])


def test_launch_stack_roundtrip(ring):
    g, prod, cons = ring
    st = ev.LaunchStack(correlation_id=42, tid=999, pid=1234,
                        ips=(0x1000, 0x2000, 0x3000))
    prod.write(g.EV_LAUNCH_STACK, ev.encode_launch_stack(st))
    [(_, payload)] = cons.drain()
    assert ev.decode_launch_stack(payload) == st


def test_gpu_config_roundtrip(ring):
    g, prod, cons = ring
    cfg = ev.GpuConfig(gpu_index=1, method=1, unit=3, interval=10000,
                       ns_per_sample=1e7)
    prod.write(g.EV_GPU_CONFIG, ev.encode_gpu_config(cfg))
    [(_, payload)] = cons.drain()
    assert ev.decode_gpu_config(payload) == cfg


def test_ring_fills_and_drops(tmp_path):
    g = _native()
    path = str(tmp_path / "small.ring")
    prod = g.TestRingProducer(path, 1 << 12)  # 4 KiB
    payload = b"x" * 256
    writes = sum(1 for _ in range(64) if prod.write(7, payload))
    assert writes < 64
    assert prod.dropped == 64 - writes
    cons = g.RingConsumer(path)
    recs = cons.drain()
    assert len(recs) == writes
    # After drain the ring has room again.
    assert prod.write(7, payload)


def test_ring_wraparound(tmp_path):
    g = _native()
    path = str(tmp_path / "wrap.ring")
    prod = g.TestRingProducer(path, 1 << 12)
    cons = g.RingConsumer(path)
    payload = bytes(range(256)) * 3  # 768 bytes
    for round_ in range(50):
        assert prod.write(5, payload)
        [(rtype, out)] = cons.drain()
        assert rtype == 5
        assert out == payload, f"corruption at round {round_}"
