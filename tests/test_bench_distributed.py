"""bench.py distributed-path smoke on CPU (gloo, world_size 2).

The driver launches bench.py under torch.distributed.run for N>1 GPUs;
this guards that path — rendezvous, DDP wrap, all-reduce of the
metrics, single rank-0 JSON line — without a GPU (SURVEY.md §6 bench
contract)."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

perf = pytest.mark.skipif(
    not os.access("/proc/sys/kernel/perf_event_paranoid", os.R_OK),
    reason="no perf_event support")


@perf
def test_bench_two_ranks_cpu(tmp_path):
    env = dict(os.environ)
    env["PYTHONPATH"] = REPO
    env["MASTER_ADDR"] = "127.0.0.1"
    env["MASTER_PORT"] = "29571"
    env["PARCA_BENCH_SETTLE"] = "1"
    env["PARCA_BENCH_NO_GPU_TOOL"] = "1"  # no HIP runtime on CPU hosts
    env["PARCA_BENCH_STORE"] = str(tmp_path)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run",
         "--nnodes=1", "--nproc-per-node", "2",
         "--master-addr", "127.0.0.1", "--master-port", "29571",
         os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "1",
         "--model", "tiny", "--device", "cpu", "--seq-len", "128"],
        env=env, capture_output=True, text=True, timeout=240)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.splitlines()
             if ln.startswith("{") and '"metric"' in ln]
    assert len(lines) == 1, f"expected ONE rank-0 JSON line, got {lines}"
    r = json.loads(lines[0])
    assert r["metric"] == "profiling_cpu_overhead_pct"
    assert r["config"]["parallelism"] == "dp2"
    assert r["steps"] == 2
    assert r["ms_per_step"] > 0
    assert r["config"]["dropped_sample_pct"] >= 0
