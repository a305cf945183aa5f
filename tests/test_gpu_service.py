"""End-to-end GPU service test with a synthetic ring (no GPU needed):
fake tool-side producer writes code-object load, kernel symbols, launch
stacks, dispatch timings and PC samples through the real C++ ring; the
service must emit correctly-shaped gpu_kernel_time and gpu_pcsample
traces (reference flow: SURVEY.md §3.3)."""

import os
import subprocess
import textwrap

import numpy as np
import pytest

from parca_agent_amd.elf import ELFFile, SymbolIndex
from parca_agent_amd.gpu import events as ev
from parca_agent_amd.gpu.service import GPUProfilerService
from parca_agent_amd.model import FrameType, TraceOrigin
from parca_agent_amd.reporter import Reporter


class CollectingDestination:
    def __init__(self):
        self.samples = []

    def write_batch(self, batch):
        self.samples.extend(batch)

    def close(self):
        pass


@pytest.fixture(scope="module")
def fake_code_object(tmp_path_factory):
    """A real ELF shared object standing in for a gfx950 code object —
    the parser is ISA-agnostic and symbols resolve identically."""
    d = tmp_path_factory.mktemp("co")
    src = d / "kern.c"
    src.write_text(textwrap.dedent("""
        void my_gemm_kernel(void) {}
        void my_softmax_kernel(void) {}
    """))
    path = d / "kern.so"
    subprocess.run(["gcc", "-shared", "-fPIC", "-O0", str(src),
                    "-o", str(path)], check=True)
    return str(path)


def _make_service(tmp_path, monkeypatch=None):
    from parca_agent_amd.native import gpu as native_gpu

    g = native_gpu()
    dest = CollectingDestination()
    rep = Reporter([dest])
    # bucket_shift=0 (byte buckets): test functions are tiny and unaligned;
    # the production default of 64-byte buckets is far below kernel sizes.
    svc = GPUProfilerService(rep, shm_dir=str(tmp_path),
                             use_device_bucketize=False, bucket_shift=0)
    pid = os.getpid()  # must be alive so the ring is not reaped
    path = os.path.join(str(tmp_path), f"parca_gpu_{pid}.ring")
    prod = g.TestRingProducer(path, 1 << 20)
    return g, dest, rep, svc, prod, pid


def test_kernel_time_and_pc_flow(tmp_path, fake_code_object):
    g, dest, rep, svc, prod, pid = _make_service(tmp_path)
    size = os.path.getsize(fake_code_object)
    with ELFFile.open(fake_code_object) as elf:
        sym = next(s for s in elf.symbols()
                   if s.name == "my_gemm_kernel")

    load_base = 0x7F00_0000_0000
    prod.write(g.EV_CODE_OBJECT_LOAD, ev.encode_code_object_load(
        ev.CodeObjectLoad(
            code_object_id=1, load_base=load_base, load_size=size,
            load_delta=load_base,  # vaddr == code_object_offset
            memory_base=0, memory_size=0, storage_type=1,
            uri=f"file://{fake_code_object}#offset=0&size={size}")))
    prod.write(g.EV_KERNEL_SYMBOL, ev.encode_kernel_symbol(
        ev.KernelSymbol(kernel_id=55, code_object_id=1, kernel_object=0,
                        name="my_gemm_kernel.kd")))
    prod.write(g.EV_GPU_CONFIG, ev.encode_gpu_config(
        ev.GpuConfig(gpu_index=0, method=1, unit=3, interval=10000,
                     ns_per_sample=1e7)))
    # Host launch stack then its timing.
    prod.write(g.EV_LAUNCH_STACK, ev.encode_launch_stack(
        ev.LaunchStack(correlation_id=100, tid=42, pid=pid,
                       ips=(0xdead0000,))))
    prod.write(g.EV_KERNEL_DISPATCH, ev.encode_kernel_dispatch(
        ev.KernelDispatch(
            correlation_id=100, dispatch_id=1, kernel_id=55,
            start_ns=1_000_000, end_ns=3_500_000, tid=42, gpu_index=0,
            pid=pid, grid=(1024, 1, 1), workgroup=(256, 1, 1),
            private_segment_size=0, group_segment_size=0)))
    # PC samples inside my_gemm_kernel.
    samples = np.zeros(7, dtype=ev.PC_SAMPLE_DTYPE)
    samples["code_object_id"] = 1
    samples["code_object_offset"] = sym.value  # vaddr==offset by delta
    samples["exec_mask"] = (1 << 64) - 1
    prod.write(g.EV_PC_SAMPLE_BATCH, ev.encode_pc_sample_batch(0, samples))

    svc.drain_once()
    svc.flush_pc()
    rep.flush()

    by_origin = {}
    for s in dest.samples:
        by_origin.setdefault(s.sample_type.sample_type, []).append(s)

    # gpu_kernel_time trace: kernel pseudo-frame + host frame.
    [kt] = by_origin["gpu_kernel_time"]
    assert kt.value == 2_500_000
    assert kt.trace.frames[0].kind == FrameType.GPU_KERNEL
    assert kt.trace.frames[0].function_name == "my_gemm_kernel"
    assert kt.trace.frames[0].mapping.file_id  # code object identity
    assert len(kt.trace.frames) == 2  # + host stack frame
    assert kt.labels["gpu"] == "0"

    # gpu_pcsample trace: bucketed, symbolized, period from GpuConfig.
    [pc] = by_origin["gpu_pcsample"]
    assert pc.value == 7
    assert pc.period == int(1e7)
    assert pc.trace.frames[0].kind == FrameType.GPU_PC
    assert pc.trace.frames[0].function_name == "my_gemm_kernel"
    # exec_mask was all-64-lanes: the wave-occupancy label reflects it.
    assert pc.labels["avg_active_lanes"] == "64"

    assert svc.metrics.pc_samples == 7
    assert svc.metrics.kernels_reported == 1


def test_dispatch_without_stack_emits_kernel_only(tmp_path, fake_code_object):
    """Timing with no launch stack (capture rate-limited tool-side) is
    reported immediately with kernel-only attribution."""
    g, dest, rep, svc, prod, pid = _make_service(tmp_path)
    prod.write(g.EV_KERNEL_SYMBOL, ev.encode_kernel_symbol(
        ev.KernelSymbol(kernel_id=9, code_object_id=1, kernel_object=0,
                        name="orphan_kernel")))
    prod.write(g.EV_KERNEL_DISPATCH, ev.encode_kernel_dispatch(
        ev.KernelDispatch(
            correlation_id=1, dispatch_id=1, kernel_id=9,
            start_ns=0, end_ns=1000, tid=1, gpu_index=0, pid=pid,
            grid=(1, 1, 1), workgroup=(64, 1, 1),
            private_segment_size=0, group_segment_size=0)))
    svc.drain_once()
    rep.flush()
    [kt] = dest.samples
    assert kt.trace.frames[0].function_name == "orphan_kernel"
    assert len(kt.trace.frames) == 1


def test_ring_error_event_counted(tmp_path):
    g, dest, rep, svc, prod, pid = _make_service(tmp_path)
    prod.write(g.EV_ERROR, ev.encode_error(
        ev.RingError(code=2, message="rocprofiler pc buffer dropped")))
    svc.drain_once()
    assert svc.metrics.tool_errors == 1


def test_kernel_batch_aggregation(tmp_path, fake_code_object):
    """Identical (kernel, stack, gpu, tid) dispatches within one batch
    sum their durations into one reported sample."""
    g, dest, rep, svc, prod, pid = _make_service(tmp_path)
    prod.write(g.EV_KERNEL_SYMBOL, ev.encode_kernel_symbol(
        ev.KernelSymbol(kernel_id=3, code_object_id=1, kernel_object=0,
                        name="hot_kernel")))
    for i in range(10):
        prod.write(g.EV_KERNEL_DISPATCH, ev.encode_kernel_dispatch(
            ev.KernelDispatch(
                correlation_id=1000 + i, dispatch_id=i, kernel_id=3,
                start_ns=i * 100, end_ns=i * 100 + 50, tid=7, gpu_index=0,
                pid=pid, grid=(1, 1, 1), workgroup=(64, 1, 1),
                private_segment_size=0, group_segment_size=0)))
    # A different kernel stays separate.
    prod.write(g.EV_KERNEL_SYMBOL, ev.encode_kernel_symbol(
        ev.KernelSymbol(kernel_id=4, code_object_id=1, kernel_object=0,
                        name="other_kernel")))
    prod.write(g.EV_KERNEL_DISPATCH, ev.encode_kernel_dispatch(
        ev.KernelDispatch(
            correlation_id=2000, dispatch_id=99, kernel_id=4,
            start_ns=0, end_ns=7, tid=7, gpu_index=0, pid=pid,
            grid=(1, 1, 1), workgroup=(64, 1, 1),
            private_segment_size=0, group_segment_size=0)))
    svc.drain_once()
    rep.flush()
    by_name = {}
    for s in dest.samples:
        by_name.setdefault(s.trace.frames[0].function_name, []).append(s)
    [hot] = by_name["hot_kernel"]
    assert hot.value == 10 * 50  # summed
    [other] = by_name["other_kernel"]
    assert other.value == 7


def test_duty_cycle_scaling(tmp_path, fake_code_object):
    """A duty-cycle advertisement (GpuConfig method=100) scales observed
    kernel durations to full-wall estimates."""
    g, dest, rep, svc, prod, pid = _make_service(tmp_path)
    prod.write(g.EV_GPU_CONFIG, ev.encode_gpu_config(
        ev.GpuConfig(gpu_index=0, method=100, unit=0, interval=1000,
                     ns_per_sample=5.0)))  # 1s on / 4s off
    prod.write(g.EV_KERNEL_SYMBOL, ev.encode_kernel_symbol(
        ev.KernelSymbol(kernel_id=1, code_object_id=1, kernel_object=0,
                        name="cycled_kernel")))
    prod.write(g.EV_KERNEL_DISPATCH, ev.encode_kernel_dispatch(
        ev.KernelDispatch(
            correlation_id=1, dispatch_id=1, kernel_id=1,
            start_ns=0, end_ns=1000, tid=1, gpu_index=0, pid=pid,
            grid=(1, 1, 1), workgroup=(64, 1, 1),
            private_segment_size=0, group_segment_size=0)))
    svc.drain_once()
    rep.flush()
    [kt] = dest.samples
    assert kt.value == 5000  # 1000 ns observed x duty factor 5


def test_daemon_multi_gpu_fanout(tmp_path, fake_code_object):
    """Daemon shape: one agent, one process ring carrying events from
    TWO GPUs (gpu_index 0 and 5). Samples must fan out with per-GPU
    labels, per-GPU PC periods from each GPU's own GpuConfig, and the
    daemon node-merge must emit one pprof with both GPUs represented
    (VERDICT.md missing#7)."""
    g, dest, rep, svc, prod, pid = _make_service(tmp_path)
    size = os.path.getsize(fake_code_object)
    with ELFFile.open(fake_code_object) as elf:
        sym = next(s for s in elf.symbols() if s.name == "my_gemm_kernel")

    load_base = 0x7F00_0000_0000
    prod.write(g.EV_CODE_OBJECT_LOAD, ev.encode_code_object_load(
        ev.CodeObjectLoad(
            code_object_id=1, load_base=load_base, load_size=size,
            load_delta=load_base, memory_base=0, memory_size=0,
            storage_type=1,
            uri=f"file://{fake_code_object}#offset=0&size={size}")))
    prod.write(g.EV_KERNEL_SYMBOL, ev.encode_kernel_symbol(
        ev.KernelSymbol(kernel_id=55, code_object_id=1, kernel_object=0,
                        name="my_gemm_kernel.kd")))
    # Two GPUs with different PC-sampling intervals.
    prod.write(g.EV_GPU_CONFIG, ev.encode_gpu_config(
        ev.GpuConfig(gpu_index=0, method=1, unit=3, interval=10000,
                     ns_per_sample=1e7)))
    prod.write(g.EV_GPU_CONFIG, ev.encode_gpu_config(
        ev.GpuConfig(gpu_index=5, method=1, unit=3, interval=20000,
                     ns_per_sample=2e7)))
    for gpu, corr in ((0, 100), (5, 101)):
        prod.write(g.EV_KERNEL_DISPATCH, ev.encode_kernel_dispatch(
            ev.KernelDispatch(
                correlation_id=corr, dispatch_id=corr, kernel_id=55,
                start_ns=1_000_000, end_ns=2_000_000 + gpu, tid=42,
                gpu_index=gpu, pid=pid, grid=(1024, 1, 1),
                workgroup=(256, 1, 1), private_segment_size=0,
                group_segment_size=0)))
        samples = np.zeros(3 + gpu, dtype=ev.PC_SAMPLE_DTYPE)
        samples["code_object_id"] = 1
        samples["code_object_offset"] = sym.value
        samples["exec_mask"] = (1 << 64) - 1
        prod.write(g.EV_PC_SAMPLE_BATCH,
                   ev.encode_pc_sample_batch(gpu, samples))

    svc.drain_once()
    svc.flush_pc()
    rep.flush()

    kt = [s for s in dest.samples
          if s.sample_type.sample_type == "gpu_kernel_time"]
    pc = [s for s in dest.samples
          if s.sample_type.sample_type == "gpu_pcsample"]
    assert {s.labels["gpu"] for s in kt} == {"0", "5"}
    assert {s.labels["gpu"] for s in pc} == {"0", "5"}
    by_gpu_pc = {s.labels["gpu"]: s for s in pc}
    assert by_gpu_pc["0"].period == int(1e7)
    assert by_gpu_pc["5"].period == int(2e7)  # per-GPU GpuConfig
    assert by_gpu_pc["0"].value == 3
    assert by_gpu_pc["5"].value == 8

    # Daemon-side node merge: one pprof, both GPUs labeled.
    from parca_agent_amd.gpu.merge import build_node_profile_from_batch
    from parca_agent_amd.pprof import decode_profile

    data = build_node_profile_from_batch(dest.samples, node="daemon-node")
    prof = decode_profile(data)
    gpus = {s["labels"].get("gpu") for s in prof.samples}
    assert gpus == {"0", "5"}
    kt_by_gpu = {}
    for s in prof.samples:
        if "view" not in s["labels"]:
            kt_by_gpu[s["labels"]["gpu"]] = s["values"][0]
    assert kt_by_gpu["0"] == 1_000_000
    assert kt_by_gpu["5"] == 1_000_005


def test_storm_drain_100k_events_per_sec(tmp_path):
    """VERDICT.md next#6 evidence: the batched native drain must sustain
    an 8-GPU-node event storm (>=100k dispatches/s — ~5x the round-1
    soak rate) through the FULL service pipeline with zero ring drops
    and exact duration accounting, while reporter flushes run
    concurrently on the same GIL."""
    import threading
    import time as _t

    from parca_agent_amd.native import gpu as native_gpu

    g = native_gpu()
    dest = CollectingDestination()
    rep = Reporter([dest])
    # Production-shape knobs: 32 MiB ring (the tool-side default) and a
    # tight poll so the drain keeps ahead of a >2M/s producer burst.
    svc = GPUProfilerService(rep, shm_dir=str(tmp_path),
                             use_device_bucketize=False,
                             poll_interval=0.02)
    pid = os.getpid()
    path = os.path.join(str(tmp_path), f"parca_gpu_{pid}.ring")
    prod = g.TestRingProducer(path, 1 << 25)
    prod.write(g.EV_KERNEL_SYMBOL, ev.encode_kernel_symbol(
        ev.KernelSymbol(kernel_id=1, code_object_id=1, kernel_object=0,
                        name="storm_kernel")))

    n_events = 400_000
    dur_ns = 1000
    stop_flush = threading.Event()

    def flusher():
        while not stop_flush.is_set():
            rep.flush()
            _t.sleep(0.05)

    svc.start()
    fl = threading.Thread(target=flusher)
    fl.start()
    payloads = [ev.encode_kernel_dispatch(ev.KernelDispatch(
        correlation_id=i & 0xFFFF, dispatch_id=i, kernel_id=1,
        start_ns=0, end_ns=dur_ns, tid=1 + (i & 3), gpu_index=i & 7,
        pid=pid, grid=(1, 1, 1), workgroup=(64, 1, 1),
        private_segment_size=0, group_segment_size=0))
        for i in range(16)]
    t0 = _t.perf_counter()
    written = 0
    for i in range(n_events):
        while not prod.write(g.EV_KERNEL_DISPATCH, payloads[i & 15]):
            _t.sleep(0.001)  # ring momentarily full: backpressure, no loss
        written += 1
    dt = _t.perf_counter() - t0
    deadline = _t.monotonic() + 30
    while svc.metrics.events_by_type.get(ev.EV_KERNEL_DISPATCH, 0) < \
            written and _t.monotonic() < deadline:
        _t.sleep(0.05)
    svc.stop()
    stop_flush.set()
    fl.join()
    rep.flush()

    rate = written / dt
    assert rate > 100_000, f"producer only reached {rate:.0f}/s"
    assert svc.metrics.ring_dropped == 0
    assert prod.dropped == 0
    got = svc.metrics.events_by_type.get(ev.EV_KERNEL_DISPATCH, 0)
    assert got == written, (got, written)
    total_ns = sum(s.value for s in dest.samples
                   if s.sample_type.sample_type == "gpu_kernel_time")
    assert total_ns == written * dur_ns


def test_lazy_device_accumulator_swap_and_fold(tmp_path, monkeypatch,
                                               fake_code_object):
    """The DeviceAccumulator is constructed lazily on the FIRST PC
    batch (keeping the agent HIP-free until then); host-gathered
    buckets from the race window must fold into it and subsequent
    batches must route to the device path."""
    import numpy as np_

    import parca_agent_amd.gpu.service as svc_mod

    class FakeDevice:
        instances = []

        def __init__(self, layout):
            self.layout = layout
            self._pending = {}
            self.calls = []
            FakeDevice.instances.append(self)

        def _pending_arrays(self, gpu):
            total = self.layout.total_buckets
            pair = self._pending.get(gpu)
            if pair is None or len(pair[0]) < total:
                pair = (np_.zeros(total, dtype=np_.uint64),
                        np_.zeros(total, dtype=np_.uint64))
                self._pending[gpu] = pair
            return pair

        def accumulate(self, pid, samples, gpu=-1):
            self.calls.append((pid, len(samples), gpu))

        def read(self, also_reset=True):
            return {g: (h.copy(), l.copy())
                    for g, (h, l) in self._pending.items()}

        def layout_changed(self):
            pass

        def drop_process(self, pid):
            pass

    monkeypatch.setattr(svc_mod, "DeviceAccumulator", FakeDevice)
    g, dest, rep, svc, prod, pid = _make_service(tmp_path)
    svc._device_pending = True  # as on a /dev/kfd host

    size = os.path.getsize(fake_code_object)
    prod.write(g.EV_CODE_OBJECT_LOAD, ev.encode_code_object_load(
        ev.CodeObjectLoad(
            code_object_id=1, load_base=0x7F00_0000_0000, load_size=size,
            load_delta=0x7F00_0000_0000, memory_base=0, memory_size=0,
            storage_type=1,
            uri=f"file://{fake_code_object}#offset=0&size={size}")))
    samples = np.zeros(4, dtype=ev.PC_SAMPLE_DTYPE)
    samples["code_object_id"] = 1
    samples["exec_mask"] = 1
    prod.write(g.EV_PC_SAMPLE_BATCH, ev.encode_pc_sample_batch(2, samples))
    svc.drain_once()

    assert isinstance(svc.accumulator, FakeDevice)
    assert svc._device_pending is False
    # first batch went to the device accumulator
    assert FakeDevice.instances[-1].calls == [(pid, 4, 2)]
    # second batch routes straight there
    prod.write(g.EV_PC_SAMPLE_BATCH, ev.encode_pc_sample_batch(5, samples))
    svc.drain_once()
    assert FakeDevice.instances[-1].calls[-1] == (pid, 4, 5)


def test_lazy_device_accumulator_falls_back_on_error(tmp_path, monkeypatch,
                                                     fake_code_object):
    import parca_agent_amd.gpu.service as svc_mod
    from parca_agent_amd.gpu.pcbuckets import HostAccumulator

    def boom(layout):
        raise RuntimeError("no HIP device")

    monkeypatch.setattr(svc_mod, "DeviceAccumulator", boom)
    g, dest, rep, svc, prod, pid = _make_service(tmp_path)
    svc._device_pending = True
    samples = np.zeros(2, dtype=ev.PC_SAMPLE_DTYPE)
    samples["code_object_id"] = 9
    prod.write(g.EV_PC_SAMPLE_BATCH, ev.encode_pc_sample_batch(0, samples))
    svc.drain_once()
    assert isinstance(svc.accumulator, HostAccumulator)
    assert svc._device_pending is False
    assert svc.metrics.pc_samples == 2


def test_gpu_samples_carry_native_labels(tmp_path, fake_code_object):
    """GPU kernel-time samples join the launcher thread's custom labels
    (include/parca_custom_labels.h table) on (pid, tid)."""
    from tests.test_nativelabels import NSLOTS, _mk_table

    g, dest, rep, svc, prod, pid = _make_service(tmp_path)
    tid = 777
    labeldir = tmp_path / "labels"
    labeldir.mkdir()
    (labeldir / f"parca_labels_{pid}").write_bytes(_mk_table({
        tid % NSLOTS: (tid, 2, [("endpoint", "/train")])}))
    svc.native_labels.directory = str(labeldir)

    prod.write(g.EV_KERNEL_SYMBOL, ev.encode_kernel_symbol(
        ev.KernelSymbol(kernel_id=5, code_object_id=1, kernel_object=0,
                        name="labeled_kernel")))
    for t, corr in ((tid, 1), (42, 2)):  # second tid: unlabeled
        prod.write(g.EV_KERNEL_DISPATCH, ev.encode_kernel_dispatch(
            ev.KernelDispatch(
                correlation_id=corr, dispatch_id=corr, kernel_id=5,
                start_ns=0, end_ns=1000, tid=t, gpu_index=0, pid=pid,
                grid=(1, 1, 1), workgroup=(64, 1, 1),
                private_segment_size=0, group_segment_size=0)))
    svc.drain_once()
    rep.flush()
    by_tid = {s.labels.get("thread_id"): s for s in dest.samples}
    assert by_tid[str(tid)].labels.get("endpoint") == "/train"
    assert "endpoint" not in by_tid["42"].labels
