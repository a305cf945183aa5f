"""Node merge tests: rank-profile folding, collective attribution, and a
real multi-process all-gather over torch.distributed gloo (world=2) —
the CPU stand-in for the RCCL/xGMI path whose transport framing is
identical (tested natively in tests/test_gpu_device.py)."""

import json
import multiprocessing as mp
import os

import pytest

from parca_agent_amd.gpu.merge import (
    RankProfile,
    build_rank_profile,
    collective_op,
    is_collective_kernel,
    merge_node_profile,
)
from parca_agent_amd.model import (
    Frame,
    FrameType,
    MappingFile,
    SampleType,
    Trace,
)
from parca_agent_amd.pprof import decode_profile
from parca_agent_amd.reporter.reporter import PendingSample


def _kernel_sample(name, value, mapping=None):
    return PendingSample(
        trace=Trace(frames=(Frame(kind=FrameType.GPU_KERNEL, address=0,
                                  mapping=mapping, function_name=name),)),
        labels={}, value=value, timestamp_ns=0,
        sample_type=SampleType("gpu_kernel_time", "nanoseconds",
                               "gpu_kernel_time", "nanoseconds"),
        period=0)


def _pc_sample(fid, addr, sym, count):
    m = MappingFile(file_id=fid, path=f"codeobj-{fid[:16]}")
    return PendingSample(
        trace=Trace(frames=(Frame(kind=FrameType.GPU_PC, address=addr,
                                  mapping=m, function_name=sym),)),
        labels={}, value=count, timestamp_ns=0,
        sample_type=SampleType("gpu_pcsample", "count",
                               "gpu_pcsample", "nanoseconds"),
        period=0)


def test_collective_detection():
    assert is_collective_kernel(
        "ncclDevKernel_AllReduce_Sum_bf16_RING_LL(ncclDevKernelArgs4K)")
    assert not is_collective_kernel("Cijk_Ailk_Bljk_BBS_BH")
    assert collective_op(
        "ncclDevKernel_AllReduce_Sum_bf16_RING_LL(...)") == "AllReduce"
    assert collective_op("ncclDevKernel_ReduceScatter_Sum_f32") == \
        "ReduceScatter"


def test_build_rank_profile():
    cobj = MappingFile(file_id="b" * 32, path="codeobj-bbbb")
    samples = [
        _kernel_sample("gemm_k", 1000, mapping=cobj),
        _kernel_sample("gemm_k", 500, mapping=cobj),
        _kernel_sample("ncclDevKernel_AllReduce_Sum_bf16_RING_LL", 2000),
        _pc_sample("a" * 32, 0x100, "gemm_k", 7),
    ]
    rp = build_rank_profile(3, 3, samples, boottime_ns=1)
    assert rp.kernel_times[("b" * 32, "gemm_k")] == (1500, 2)
    assert rp.collectives["AllReduce"] == (2000, 1)
    assert rp.pc_buckets[("a" * 32, 0x100, "gemm_k")] == 7
    # wire round trip
    rp2 = RankProfile.from_bytes(rp.to_bytes())
    assert rp2 == rp


def test_same_name_different_code_objects_stay_distinct():
    """Two code objects exporting the same demangled kernel name must
    not be merged into one key (code-object identity is preserved
    through the wire payload and into the merged pprof build_id)."""
    a = MappingFile(file_id="a" * 32, path="codeobj-aaaa")
    b = MappingFile(file_id="b" * 32, path="codeobj-bbbb")
    samples = [_kernel_sample("gemm_k", 100, mapping=a),
               _kernel_sample("gemm_k", 900, mapping=b)]
    rp = build_rank_profile(0, 0, samples, boottime_ns=1)
    assert rp.kernel_times[("a" * 32, "gemm_k")] == (100, 1)
    assert rp.kernel_times[("b" * 32, "gemm_k")] == (900, 1)
    prof = decode_profile(merge_node_profile([rp]).serialize_gzip())
    build_ids = set()
    for s in prof.samples:
        loc = prof.locations[s["location_ids"][0]]
        m = prof.mappings.get(loc["mapping_id"])
        if m:
            build_ids.add(prof.strings[m["build_id"]])
    assert {"a" * 32, "b" * 32} <= build_ids


def test_merge_node_profile_pprof():
    rps = []
    for rank in range(4):
        rp = RankProfile(rank=rank, gpu_index=rank)
        rp.kernel_times = {("d" * 32, "gemm_k"): (1000 * (rank + 1),
                                                  rank + 1),
                           ("", "ncclDevKernel_AllReduce_Sum_bf16"):
                               (500, 2)}
        rp.collectives = {"AllReduce": (500, 2)}
        rp.pc_buckets = {("c" * 32, 0x40, "gemm_k"): 3}
        rps.append(rp)
    prof = decode_profile(
        merge_node_profile(rps, node="node-1").serialize_gzip())
    assert prof.sample_types[0].type == "gpu_time"
    gemm = [s for s in prof.samples
            if prof.stack_names(s)[0] == "gemm_k"
            and "view" not in s["labels"]]
    assert len(gemm) == 4  # one per GPU
    by_gpu = {s["labels"]["gpu"]: s["values"] for s in gemm}
    assert by_gpu["0"] == [1000, 1]
    assert by_gpu["3"] == [4000, 4]
    # collective attribution pseudo-frame
    coll = [s for s in prof.samples
            if "rccl::AllReduce" in prof.stack_names(s)]
    assert len(coll) == 4
    pc = [s for s in prof.samples if s["labels"].get("view") == "pc_sample"]
    assert len(pc) == 4


def _dist_worker(rank, world, port, q):
    os.environ.update({
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
        "RANK": str(rank), "WORLD_SIZE": str(world),
    })
    import torch.distributed as dist

    from parca_agent_amd.gpu.merge import NodeMergeService, TorchDistTransport

    dist.init_process_group("gloo")
    svc = NodeMergeService(TorchDistTransport(), rank=rank, gpu_index=rank,
                           node="testnode")
    samples = [_kernel_sample(f"kern_rank{rank}", 100 * (rank + 1)),
               _kernel_sample("ncclDevKernel_AllReduce_Sum_bf16", 50)]
    merged = svc.merge(samples)
    q.put((rank, merged))
    dist.destroy_process_group()


def test_multiprocess_gloo_merge():
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_dist_worker, args=(r, world, 29855, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world):
        rank, merged = q.get(timeout=120)
        results[rank] = merged
    for p in procs:
        p.join(timeout=30)
        assert p.exitcode == 0
    assert results[1] is None
    prof = decode_profile(results[0])
    names = {prof.stack_names(s)[0] for s in prof.samples}
    assert "kern_rank0" in names
    assert "kern_rank1" in names
    comments = prof.strings
    assert any("node=testnode" in s for s in comments)


def test_daemon_node_profile_destination(tmp_path):
    from parca_agent_amd.gpu.merge import DaemonNodeProfileDestination

    dest = DaemonNodeProfileDestination(str(tmp_path / "node"), node="n1")
    cobj = MappingFile(file_id="e" * 32, path="codeobj")
    batch = [_kernel_sample("k0", 100, mapping=cobj)]
    batch[0].labels["gpu"] = "2"
    dest.write_batch(batch)
    dest.write_batch([])  # batch with no GPU samples: no file
    files = sorted(os.listdir(tmp_path / "node"))
    assert len(files) == 1 and files[0].endswith(".node_gpu.pb.gz")
    prof = decode_profile(open(tmp_path / "node" / files[0], "rb").read())
    assert {s["labels"].get("gpu") for s in prof.samples} == {"2"}


def test_rccl_microbench_world2_cpu(tmp_path):
    """BASELINE config 3's harness under torchrun world=2 on CPU/gloo
    (VERDICT.md next#4): rank 0 must emit one merged node profile with
    per-rank, per-collective attribution (synthetic RCCL timings stand
    in for device kernels off-GPU)."""
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = tmp_path / "node_profile.pb.gz"
    env = dict(os.environ)
    env["PYTHONPATH"] = repo
    env["PARCA_BENCH_CHILD"] = "1"  # skip the tool re-exec on CPU
    res = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29613",
         os.path.join(repo, "tools", "rccl_microbench.py"),
         "--size-mb", "1", "--iters", "4", "--warmup", "1",
         "--out", str(out)],
        env=env, cwd=repo, capture_output=True, text=True, timeout=180)
    assert res.returncode == 0, res.stderr[-2000:]
    line = [ln for ln in res.stdout.splitlines()
            if ln.startswith("{")][-1]
    doc = json.loads(line)
    assert doc["world"] == 2
    assert doc["collective"] == "all_reduce"

    prof = decode_profile(open(out, "rb").read())
    coll = [s for s in prof.samples
            if "rccl::AllReduce" in prof.stack_names(s)]
    assert coll, "no per-collective attribution samples"
    ranks = {s["labels"].get("rank") for s in coll}
    assert ranks == {"0", "1"}
    by_rank = {s["labels"]["rank"]: s["values"][0] for s in coll}
    assert by_rank["0"] == 4 * 1000
    assert by_rank["1"] == 4 * 2000
