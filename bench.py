#!/usr/bin/env python3
"""Profiling-overhead benchmark (BASELINE.json headline metric).

Metric: profiling CPU overhead % at 19 Hz + dropped-sample rate on
1/2/4/8 MI355X. The workload is a Llama-3-8B-shaped training loop
(random-init weights, synthetic batches, bf16) — BASELINE.json config 4 —
run twice inside each rank: a baseline phase with all profiling inactive,
then a profiled phase with the full stack live (19 Hz perf CPU sampler,
rocprofiler kernel-dispatch tracing + gfx950 PC sampling in-process, shm
ring drain, CDNA4 bucketize kernel, Arrow reporter). Overhead is the
profiling-attributable CPU time as a percentage of the node's total CPU
capacity during the profiled phase — the same quantity behind the
reference's "<1 % CPU" claim — measured from rusage deltas;
workload slowdown % and dropped-sample rate are reported alongside.

Launch (driver contract): `python bench.py --gpus N --steps K --warmup W`;
for N>1 launched under torchrun with one rank per GPU over RCCL.
"""

import argparse
import json
import os
import resource
import sys
import time

# Re-exec with the rocprofiler tool injected BEFORE torch initializes HIP.
# The tool registers its services but stays dormant (defer-start) so the
# baseline phase is clean; the profiled phase starts it at runtime.
def _ensure_tool_env() -> None:
    if os.environ.get("PARCA_BENCH_CHILD") == "1":
        return
    if os.environ.get("PARCA_BENCH_NO_GPU_TOOL") == "1":
        return
    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    from parca_agent_amd.agent import tool_env

    env = dict(os.environ)
    shm_dir = env.get("PARCA_GPU_SHM_DIR", "/dev/shm")
    env.update(tool_env(shm_dir=shm_dir, defer_start=True,
                        ring_bytes=8 << 20, pc_sampling=True,
                        launch_stacks=True))
    env["PARCA_BENCH_CHILD"] = "1"
    os.execve(sys.executable, [sys.executable] + sys.argv, env)


_ensure_tool_env()

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch  # noqa: E402


def build_model(device, dtype, model_name):
    """Random-init Llama-3-8B (full size on MI355X 288 GB; a narrow proxy
    on CPU/dev boxes so the bench default finishes in minutes)."""
    from transformers import LlamaConfig, LlamaForCausalLM

    if model_name == "llama3-8b":
        cfg = LlamaConfig(
            vocab_size=128256, hidden_size=4096, intermediate_size=14336,
            num_hidden_layers=32, num_attention_heads=32,
            num_key_value_heads=8, max_position_embeddings=8192,
            rope_theta=500000.0, attn_implementation="sdpa")
    else:  # tiny proxy for smoke/CPU runs
        cfg = LlamaConfig(
            vocab_size=2048, hidden_size=256, intermediate_size=688,
            num_hidden_layers=4, num_attention_heads=8,
            num_key_value_heads=4, max_position_embeddings=2048)
    torch.manual_seed(1234)
    model = LlamaForCausalLM(cfg).to(device=device, dtype=dtype)
    # No gradient checkpointing: 288 GB HBM fits bs1/seq2048 activations
    # outright, and recompute would distort the overhead measurement.
    model.train()
    return model


def run_steps(model, opt, batch, n_steps, device, distributed):
    for _ in range(n_steps):
        out = model(input_ids=batch, labels=batch)
        out.loss.backward()
        if distributed:
            # DDP handles gradient all-reduce in backward.
            pass
        opt.step()
        opt.zero_grad(set_to_none=True)


def sync(device):
    if device.type == "cuda":
        torch.cuda.synchronize(device)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--seq-len", type=int, default=2048)
    ap.add_argument("--batch", type=int, default=1, help="per-GPU batch")
    ap.add_argument("--model", default=None,
                    help="llama3-8b (default on GPU) or tiny")
    ap.add_argument("--device", default=None)
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    distributed = world > 1

    on_gpu = torch.cuda.is_available()
    if args.device:
        device = torch.device(args.device)
    else:
        device = torch.device(f"cuda:{local_rank}" if on_gpu else "cpu")
    dtype = torch.bfloat16
    model_name = args.model or ("llama3-8b" if on_gpu else "tiny")

    if distributed:
        torch.distributed.init_process_group(
            backend="nccl" if on_gpu else "gloo")
        if on_gpu:
            torch.cuda.set_device(device)

    model = build_model(device, dtype, model_name)
    if distributed:
        model = torch.nn.parallel.DistributedDataParallel(
            model, device_ids=[local_rank] if on_gpu else None)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-4, foreach=True)

    torch.manual_seed(4321 + rank)
    vocab = model.module.config.vocab_size if distributed else \
        model.config.vocab_size
    batch = torch.randint(0, vocab, (args.batch, args.seq_len), device=device)

    def barrier():
        if distributed:
            torch.distributed.barrier()

    # ---- warmup (untimed) ----
    run_steps(model, opt, batch, args.warmup, device, distributed)
    sync(device)
    barrier()

    # ---- phase A: baseline, no profiling ----
    ru0 = resource.getrusage(resource.RUSAGE_SELF)
    t0 = time.perf_counter()
    run_steps(model, opt, batch, args.steps, device, distributed)
    sync(device)
    barrier()
    t_base = time.perf_counter() - t0
    ru1 = resource.getrusage(resource.RUSAGE_SELF)
    cpu_base = (ru1.ru_utime + ru1.ru_stime) - (ru0.ru_utime + ru0.ru_stime)

    # ---- bring up the full profiler ----
    from parca_agent_amd.agent import Agent, InProcessToolControl
    from parca_agent_amd.flags import Flags

    flags = Flags()
    flags.local_store.directory = os.environ.get(
        "PARCA_BENCH_STORE", f"gpurun_out/bench_profiles_rank{rank}")
    flags.rocm.enable = True
    agent = Agent(flags, enable_cpu=True, enable_gpu=True)
    tool = None
    if on_gpu and os.environ.get("PARCA_BENCH_CHILD") == "1":
        try:
            tool = InProcessToolControl()
            tool.start()
        except OSError as e:
            print(f"# tool control unavailable: {e}", file=sys.stderr)
    agent.start()

    # settle one step so lazy init (ring attach etc.) is off the clock
    run_steps(model, opt, batch, 1, device, distributed)
    sync(device)
    barrier()

    # ---- phase B: profiled ----
    ru2 = resource.getrusage(resource.RUSAGE_SELF)
    t1 = time.perf_counter()
    run_steps(model, opt, batch, args.steps, device, distributed)
    sync(device)
    barrier()
    t_prof = time.perf_counter() - t1
    ru3 = resource.getrusage(resource.RUSAGE_SELF)
    cpu_prof = (ru3.ru_utime + ru3.ru_stime) - (ru2.ru_utime + ru2.ru_stime)

    if tool is not None:
        tool.flush()
    agent.stop()
    stats = agent.stats()
    tool_stats = tool.stats() if tool is not None else \
        {"written": 0, "dropped": 0, "launch_stacks": 0}

    # profiling-attributable CPU seconds: rusage delta between phases
    # (workload CPU is constant by construction: same steps, same shapes).
    ncpu = os.cpu_count() or 1
    prof_cpu_seconds = max(cpu_prof - cpu_base, 0.0)
    overhead_pct = 100.0 * prof_cpu_seconds / (t_prof * ncpu)
    slowdown_pct = 100.0 * (t_prof - t_base) / t_base if t_base > 0 else 0.0

    ring_total = tool_stats["written"] + tool_stats["dropped"]
    ring_drop_pct = 100.0 * tool_stats["dropped"] / ring_total \
        if ring_total else 0.0
    dropped_pct = max(stats.dropped_sample_pct, ring_drop_pct)

    # MAX over ranks for every reported scalar.
    metrics_t = torch.tensor(
        [overhead_pct, slowdown_pct, dropped_pct, t_prof, t_base],
        dtype=torch.float64)
    if distributed:
        metrics_t = metrics_t.to(device if on_gpu else "cpu")
        torch.distributed.all_reduce(
            metrics_t, op=torch.distributed.ReduceOp.MAX)
        metrics_t = metrics_t.cpu()
    overhead_pct, slowdown_pct, dropped_pct, t_prof, t_base = \
        metrics_t.tolist()

    if rank == 0:
        result = {
            "metric": "profiling_cpu_overhead_pct",
            "value": round(overhead_pct, 4),
            "unit": "percent",
            "n_gpus": world if on_gpu else 0,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(1000.0 * t_prof / args.steps, 3),
            "higher_is_better": False,
            "scaling": "weak",
            "vs_baseline": round(overhead_pct / 1.0, 4),
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": model_name,
                "global_batch": args.batch * world,
                "seq_len": args.seq_len,
                "parallelism": f"dp{world}",
                "sampling_hz": 19,
                "slowdown_pct": round(slowdown_pct, 4),
                "dropped_sample_pct": round(dropped_pct, 6),
                "baseline_ms_per_step": round(1000.0 * t_base / args.steps, 3),
                "cpu_samples": stats.cpu_samples,
                "gpu_ring_events": tool_stats["written"],
                "gpu_launch_stacks": tool_stats["launch_stacks"],
                "pc_samples": stats.pc_samples,
                "kernels_reported": stats.kernels_reported,
            },
        }
        print(json.dumps(result))

    if distributed:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
