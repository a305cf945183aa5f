#!/usr/bin/env python3
"""Profiling-overhead benchmark (BASELINE.json headline metric).

Metric: profiling CPU overhead % at 19 Hz + dropped-sample rate on
1/2/4/8 MI355X. The workload is a Llama-3-8B-shaped training loop
(random-init weights, synthetic batches, bf16) — BASELINE.json config 4.
Deployment mirrors production exactly: ONE agent daemon per node
(19 Hz system-wide perf sampling + shm-ring drain + Arrow reporter),
with the rocprofiler interception tool injected into every workload rank
(kernel-dispatch timing, gfx950 PC sampling where supported, rate-limited
host launch stacks).

Each rank runs the training loop twice: a baseline phase (tool dormant,
no agent) and a profiled phase (tool active + agent daemon). The headline
value is profiling-attributable CPU time as a percentage of the node's
total CPU capacity during the profiled phase — the quantity behind the
reference's "<1 % CPU" claim: agent-daemon CPU (from
/proc/<pid>/stat) plus the in-workload tool cost (rusage delta between
phases, summed over ranks). Workload slowdown % and the dropped-sample
rate are reported alongside in config.

Driver contract: `python bench.py --gpus N --steps K --warmup W`; for
N>1 launched under torchrun, one rank per GPU over RCCL.
"""

import argparse
import json
import os
import resource
import signal
import subprocess
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
    defaults = tool_env(shm_dir=shm_dir, defer_start=True,
                        ring_bytes=32 << 20, pc_sampling=True,
                        launch_stacks=True)
    # Ambient PARCA_GPU_* overrides win (slowdown-decomposition runs).
    for k, v in defaults.items():
        env.setdefault(k, v)
    env["PARCA_GPU_DEFER_START"] = "1"
    env["PARCA_BENCH_CHILD"] = "1"
    os.execve(sys.executable, [sys.executable] + sys.argv, env)


_ensure_tool_env()

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

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
    # Construct directly on the target device: 8 ranks materializing an
    # 8B fp32 model on host RAM first would transiently need ~256 GB.
    with torch.device(device):
        model = LlamaForCausalLM(cfg)
    model = model.to(dtype=dtype)
    # No gradient checkpointing: 288 GB HBM fits bs1/seq2048 activations
    # outright, and recompute would distort the overhead measurement.
    model.train()
    return model


def run_steps(model, opt, batch, n_steps):
    for _ in range(n_steps):
        out = model(input_ids=batch, labels=batch)
        out.loss.backward()
        opt.step()
        opt.zero_grad(set_to_none=True)


def proc_cpu_seconds(pid: int) -> float:
    """utime+stime of a process from /proc/<pid>/stat, in seconds."""
    with open(f"/proc/{pid}/stat") as fh:
        data = fh.read()
    rest = data[data.rindex(")") + 2:].split()
    ticks = int(rest[11]) + int(rest[12])  # utime, stime (fields 14, 15)
    return ticks / os.sysconf("SC_CLK_TCK")


class AgentDaemon:
    """One per node: the production deployment unit."""

    def __init__(self, store_dir: str, stats_file: str, freq: int = 19):
        env = dict(os.environ)
        env["PARCA_STATS_FILE"] = stats_file
        env["PYTHONPATH"] = REPO
        env.setdefault("MALLOC_ARENA_MAX", "2")
        env.setdefault("MALLOC_MMAP_THRESHOLD_", "1048576")
        self.stats_file = stats_file
        self.proc = subprocess.Popen(
            [sys.executable, "-m", "parca_agent_amd",
             "--telemetry-disable-panic-reporting", "true",
             "--analytics-opt-out", "true",
             "--http-address", "127.0.0.1:17099",
             "--local-store-directory", store_dir,
             "--profiling-cpu-sampling-frequency", str(freq)],
            env=env, cwd=REPO,
            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)

    def cpu_seconds(self) -> float:
        try:
            return proc_cpu_seconds(self.proc.pid)
        except (OSError, ValueError):
            return 0.0

    def rss_mb(self) -> float:
        try:
            with open(f"/proc/{self.proc.pid}/statm") as fh:
                pages = int(fh.read().split()[1])
            return pages * os.sysconf("SC_PAGE_SIZE") / (1 << 20)
        except (OSError, ValueError, IndexError):
            return 0.0

    def stop(self) -> dict:
        self.proc.send_signal(signal.SIGTERM)
        try:
            self.proc.wait(timeout=30)
        except subprocess.TimeoutExpired:
            self.proc.kill()
        try:
            with open(self.stats_file) as fh:
                return json.load(fh)
        except (OSError, ValueError):
            return {}


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

    def sync():
        if device.type == "cuda":
            torch.cuda.synchronize(device)

    def barrier():
        if distributed:
            torch.distributed.barrier()

    # ---- warmup (untimed) ----
    run_steps(model, opt, batch, args.warmup)
    sync()
    barrier()

    # ---- phase A: baseline, no profiling anywhere ----
    ru0 = resource.getrusage(resource.RUSAGE_SELF)
    t0 = time.perf_counter()
    run_steps(model, opt, batch, args.steps)
    sync()
    barrier()
    t_base = time.perf_counter() - t0
    ru1 = resource.getrusage(resource.RUSAGE_SELF)
    cpu_base = (ru1.ru_utime + ru1.ru_stime) - (ru0.ru_utime + ru0.ru_stime)

    # ---- bring up production-shaped profiling ----
    # One agent daemon on the node (local_rank 0 owns it); every rank
    # activates its in-process rocprofiler tool.
    from parca_agent_amd.agent import InProcessToolControl

    agent = None
    out_dir = os.environ.get("PARCA_BENCH_STORE",
                             os.path.join(REPO, "gpurun_out"))
    os.makedirs(out_dir, exist_ok=True)
    if local_rank == 0:
        agent = AgentDaemon(os.path.join(out_dir, "bench_profiles"),
                            os.path.join(out_dir, "agent_stats.json"))
    tool = None
    if on_gpu and os.environ.get("PARCA_BENCH_CHILD") == "1":
        try:
            tool = InProcessToolControl()
            tool.start()
        except OSError as e:
            print(f"# tool control unavailable: {e}", file=sys.stderr)

    # Settle: agent attach, ring discovery and one-time .eh_frame
    # stack-delta table builds (libtorch-scale) happen off the clock —
    # they are startup costs, not steady-state overhead.
    time.sleep(float(os.environ.get("PARCA_BENCH_SETTLE",
                                    "8" if agent is not None else "0.5")))
    run_steps(model, opt, batch, 1)
    sync()
    barrier()

    # ---- phase B: profiled ----
    agent_cpu0 = agent.cpu_seconds() if agent is not None else 0.0
    ru2 = resource.getrusage(resource.RUSAGE_SELF)
    t1 = time.perf_counter()
    run_steps(model, opt, batch, args.steps)
    sync()
    barrier()
    t_prof = time.perf_counter() - t1
    ru3 = resource.getrusage(resource.RUSAGE_SELF)
    agent_cpu1 = agent.cpu_seconds() if agent is not None else 0.0
    agent_rss_mb = agent.rss_mb() if agent is not None else 0.0
    cpu_prof = (ru3.ru_utime + ru3.ru_stime) - (ru2.ru_utime + ru2.ru_stime)

    if tool is not None:
        tool.flush()
    tool_stats = tool.stats() if tool is not None else \
        {"written": 0, "dropped": 0, "launch_stacks": 0}
    time.sleep(1.0)  # let the agent drain the tail before stopping it
    agent_stats = agent.stop() if agent is not None else {}

    # Per-rank in-workload tool cost; agent cost counted once per node.
    tool_cpu = max(cpu_prof - cpu_base, 0.0)
    agent_cpu = max(agent_cpu1 - agent_cpu0, 0.0)

    ncpu = os.cpu_count() or 1
    t = torch.tensor([tool_cpu, float(tool_stats["written"]),
                      float(tool_stats["dropped"]),
                      float(tool_stats["launch_stacks"])],
                     dtype=torch.float64)
    tm = torch.tensor([t_prof, t_base], dtype=torch.float64)
    if distributed:
        dev = device if on_gpu else torch.device("cpu")
        t = t.to(dev)
        tm = tm.to(dev)
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.SUM)
        torch.distributed.all_reduce(tm, op=torch.distributed.ReduceOp.MAX)
        t = t.cpu()
        tm = tm.cpu()
    tool_cpu_sum, ring_written, ring_dropped, launch_stacks = t.tolist()
    t_prof, t_base = tm.tolist()

    if rank == 0:
        overhead_pct = 100.0 * (agent_cpu + tool_cpu_sum) / (t_prof * ncpu)
        slowdown_pct = 100.0 * (t_prof - t_base) / t_base if t_base > 0 \
            else 0.0
        ring_total = ring_written + ring_dropped
        ring_drop_pct = 100.0 * ring_dropped / ring_total if ring_total \
            else 0.0
        cpu_total = agent_stats.get("cpu_samples", 0) + \
            agent_stats.get("cpu_samples_lost", 0)
        cpu_drop_pct = 100.0 * agent_stats.get("cpu_samples_lost", 0) / \
            cpu_total if cpu_total else 0.0
        dropped_pct = max(ring_drop_pct, cpu_drop_pct,
                          agent_stats.get("dropped_sample_pct", 0.0))

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
                "agent_cpu_seconds": round(agent_cpu, 4),
                "agent_rss_mb": round(agent_rss_mb, 1),
                "tool_cpu_seconds": round(tool_cpu_sum, 4),
                "node_cpus": ncpu,
                "cpu_samples": agent_stats.get("cpu_samples", 0),
                "gpu_ring_events": int(ring_written),
                "gpu_launch_stacks": int(launch_stacks),
                "pc_samples": agent_stats.get("pc_samples", 0),
                "kernels_reported": agent_stats.get("kernels_reported", 0),
            },
        }
        print(json.dumps(result))

    if distributed:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
