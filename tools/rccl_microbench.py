#!/usr/bin/env python3
"""RCCL all-reduce microbenchmark with per-collective profile attribution
(BASELINE.json config 3).

Each rank runs a bf16 all-reduce loop over xGMI under the rocprofiler
interception tool; the in-rank drain collects RCCL device-kernel timings;
ranks merge summaries with NodeMergeService (torch.distributed transport,
RCCL on GPU) and rank 0 writes one node-level pprof where every sample is
attributed to its collective (rccl::AllReduce etc.) and GPU.

Launch:
  torchrun --nnodes=1 --nproc-per-node 8 --master-addr 127.0.0.1 \\
      tools/rccl_microbench.py --size-mb 256 --iters 50
"""

import argparse
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def _ensure_tool_env() -> None:
    if os.environ.get("PARCA_BENCH_CHILD") == "1":
        return
    from parca_agent_amd.agent import tool_env

    env = dict(os.environ)
    env.update(tool_env(defer_start=False, pc_sampling=False,
                        launch_stacks=False))
    env["PARCA_BENCH_CHILD"] = "1"
    os.execve(sys.executable, [sys.executable] + sys.argv, env)


_ensure_tool_env()

import torch  # noqa: E402


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--size-mb", type=int, default=256)
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--out", default="gpurun_out/rccl_node_profile.pb.gz")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    on_gpu = torch.cuda.is_available()
    device = torch.device(f"cuda:{local_rank}" if on_gpu else "cpu")
    if world > 1:
        torch.distributed.init_process_group(
            "nccl" if on_gpu else "gloo")
        if on_gpu:
            torch.cuda.set_device(device)
    elif on_gpu:
        # Single-rank RCCL still exercises the collective path.
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29511")
        torch.distributed.init_process_group("nccl", rank=0, world_size=1)

    n = args.size_mb * (1 << 20) // 2  # bf16 elements
    x = torch.ones(n, dtype=torch.bfloat16, device=device)

    # In-rank GPU event drain (embedded mode).
    from parca_agent_amd.gpu.merge import NodeMergeService, TorchDistTransport
    from parca_agent_amd.gpu.service import GPUProfilerService
    from parca_agent_amd.reporter import Reporter

    class Dest:
        def __init__(self):
            self.samples = []

        def write_batch(self, batch):
            self.samples.extend(batch)

        def close(self):
            pass

    dest = Dest()
    rep = Reporter([dest])
    svc = GPUProfilerService(rep, use_device_bucketize=False)
    svc.start()

    for _ in range(args.warmup):
        torch.distributed.all_reduce(x)
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        torch.distributed.all_reduce(x)
    if on_gpu:
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0

    # Flush tool buffers and drain the tail.
    from parca_agent_amd.agent import InProcessToolControl

    try:
        InProcessToolControl().flush()
    except OSError:
        pass
    time.sleep(1.5)
    svc.stop()
    if not on_gpu:
        # CPU/gloo runs have no RCCL device kernels; feed synthetic
        # per-rank collective timings through the REAL reporter so the
        # attribution + merge pipeline is exercised end to end (the CPU
        # stand-in the round-end 8-GPU run replaces with real ones).
        from parca_agent_amd.model import (Frame, FrameType, MappingFile,
                                           Trace, TraceEventMeta,
                                           TraceOrigin)

        cobj = MappingFile(file_id=f"{rank:032x}", path="codeobj-synth")
        for i in range(args.iters):
            rep.report_trace_event(
                Trace(frames=(Frame(
                    kind=FrameType.GPU_KERNEL, address=0, mapping=cobj,
                    function_name="ncclDevKernel_AllReduce_Sum_bf16_RING"
                                  "(synthetic)"),)),
                TraceEventMeta(pid=os.getpid(), tid=os.getpid(),
                               origin=TraceOrigin.GPU_KERNEL,
                               value=1000 * (rank + 1), gpu_id=local_rank))
    rep.flush()

    merger = NodeMergeService(TorchDistTransport(), rank=rank,
                              gpu_index=local_rank, node=os.uname().nodename)
    merged = merger.merge(dest.samples)
    if rank == 0 and merged is not None:
        os.makedirs(os.path.dirname(args.out) or ".", exist_ok=True)
        with open(args.out, "wb") as fh:
            fh.write(merged)
        from parca_agent_amd.gpu.merge import RankProfile
        bus_gb = args.size_mb / 1024 * 2 * (world - 1) / max(world, 1)
        print(json.dumps({
            "collective": "all_reduce",
            "size_mb": args.size_mb,
            "iters": args.iters,
            "world": world,
            "sec_per_iter": dt / args.iters,
            "busbw_GBps": bus_gb / (dt / args.iters) if world > 1 else 0.0,
            "gpu_samples_collected": len(dest.samples),
            "profile": args.out,
        }))
    if torch.distributed.is_initialized():
        torch.distributed.destroy_process_group()
    return 0


if __name__ == "__main__":
    sys.exit(main())
