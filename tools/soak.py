#!/usr/bin/env python3
"""Stability soak: run the agent daemon against a continuous GPU workload
for N seconds; report agent CPU/RSS trajectory and drop counters.
Evidence for sustained operation (reference analog: the 1-day demo soak
gate, RELEASE.md:38 — scaled to a CI-sized window).

Note: the soak saturates every core (agent + workload); live
perf tests (test_unwind e2e, daemon lifecycle) are load-sensitive
and can flake if the pytest suite runs concurrently on the same
host. Run the suite before or after, not during.
"""

import json
import os
import signal
import subprocess
import sys
import time

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def proc_stat(pid):
    with open(f"/proc/{pid}/stat") as fh:
        f = fh.read().rsplit(") ", 1)[1].split()
    cpu = (int(f[11]) + int(f[12])) / os.sysconf("SC_CLK_TCK")
    rss = int(f[21]) * os.sysconf("SC_PAGE_SIZE")
    return cpu, rss


WORKLOAD = r"""
import sys, time, torch
dur = float(sys.argv[1])
dev = "cuda" if torch.cuda.is_available() else "cpu"
size = 4096 if dev == "cuda" else 512
burst = 50 if dev == "cuda" else 2
a = torch.randn(size, size, device=dev, dtype=torch.bfloat16)
b = torch.randn(size, size, device=dev, dtype=torch.bfloat16)
t0 = time.time(); n = 0
while time.time() - t0 < dur:
    for _ in range(burst):
        a = torch.tanh(a @ b)
    if dev == "cuda":
        torch.cuda.synchronize()
    n += burst
print("iters", n)
"""


def main():
    duration = float(sys.argv[1]) if len(sys.argv) > 1 else 180.0
    nprocs = int(sys.argv[2]) if len(sys.argv) > 2 else 1
    out_dir = os.path.join(REPO, "gpurun_out")
    os.makedirs(out_dir, exist_ok=True)
    stats_file = os.path.join(out_dir, "soak_agent_stats.json")

    env = dict(os.environ)
    env["PYTHONPATH"] = REPO
    env["PARCA_STATS_FILE"] = stats_file
    env.setdefault("MALLOC_ARENA_MAX", "2")
    env.setdefault("MALLOC_MMAP_THRESHOLD_", "1048576")
    if os.environ.get("PARCA_TRACEMALLOC"):
        env["PARCA_TRACEMALLOC"] = "1"
    extra = os.environ.get("PARCA_SOAK_AGENT_ARGS", "").split()
    agent = subprocess.Popen(
        [sys.executable, "-m", "parca_agent_amd",
         "--telemetry-disable-panic-reporting", "true",
         "--analytics-opt-out", "true",
         "--http-address", "127.0.0.1:17088",
         "--local-store-directory",
         os.path.join(out_dir, "soak_profiles")] + extra,
        env=env, cwd=REPO,
        stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    time.sleep(8)

    from parca_agent_amd.agent import tool_env
    from parca_agent_amd.native import gpu as native_gpu

    wenv = dict(os.environ)
    try:
        has_gpu = native_gpu().hip_device_count() > 0
    except Exception:
        has_gpu = False
    if has_gpu:
        # Injecting the rocprofiler tool on a GPU-less machine hangs HIP
        # runtime init; CPU soaks exercise the sampler pipeline only.
        wenv.update(tool_env())
    workers = [subprocess.Popen(
        [sys.executable, "-c", WORKLOAD, str(duration)], env=wenv)
        for _ in range(nprocs)]
    work = workers[0]
    # A perl burner alongside: the agent's perl unwinder should produce
    # interpreter frames for it (counted in agent stats as perl_stacks).
    perl = None
    try:
        # prctl(PR_SET_PTRACER, PR_SET_PTRACER_ANY): lets the sibling
        # agent read this process on Yama ptrace_scope=1 hosts (GPU
        # workloads get this from the injected tool; plain interpreters
        # need it explicitly or the agent needs CAP_SYS_PTRACE).
        perl = subprocess.Popen(
            ["perl", "-e",
             "syscall(157, 0x59616d61, -1); "
             "sub burn { my $x=0; $x+=$_ for 1..5000; $x } "
             "my $d = time()+" + str(int(duration)) + "; "
             "burn() while time() < $d;"],
            stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL)
    except OSError:
        pass

    samples = []
    t0 = time.time()
    while work.poll() is None and time.time() - t0 < duration + 120:
        try:
            cpu, rss = proc_stat(agent.pid)
            samples.append({"t": round(time.time() - t0, 1),
                            "agent_cpu_s": round(cpu, 2),
                            "agent_rss_mb": round(rss / 1e6, 1)})
        except OSError:
            break
        time.sleep(10)
    for w in workers:
        w.wait()
    if perl is not None:
        perl.terminate()
        perl.wait()
    time.sleep(2)
    # Memory breakdown before shutdown: kernel-side RSS split plus the
    # agent's own component gauges (where does the plateau live?).
    breakdown = {}
    try:
        with open(f"/proc/{agent.pid}/status") as fh:
            for line in fh:
                if line.startswith(("VmRSS", "RssAnon", "RssFile",
                                    "RssShmem")):
                    k, v = line.split(":", 1)
                    breakdown[k] = v.strip()
    except OSError:
        pass
    try:
        import urllib.request

        met = urllib.request.urlopen(
            "http://127.0.0.1:17088/metrics", timeout=5).read().decode()
        for line in met.splitlines():
            if line.startswith(("parca_agent_dwarf_table_bytes",
                                "parca_agent_dwarf_stacks_truncated")):
                k, _, v = line.rpartition(" ")
                breakdown[k] = v
    except Exception:
        pass
    # Top RSS regions (who owns the anon memory?)
    try:
        regions = []
        with open(f"/proc/{agent.pid}/smaps") as fh:
            cur = None
            for line in fh:
                if "-" in line.split(" ", 1)[0] and len(line.split()) >= 5:
                    parts = line.split()
                    name = parts[5] if len(parts) > 5 else "[anon]"
                    cur = [name, 0]
                    regions.append(cur)
                elif line.startswith("Rss:") and cur is not None:
                    cur[1] = int(line.split()[1])
        regions.sort(key=lambda r: -r[1])
        breakdown["top_regions_kb"] = [f"{n} {kb}" for n, kb in
                                       regions[:12] if kb > 1024]
    except OSError:
        pass
    agent.send_signal(signal.SIGTERM)
    agent.wait(timeout=30)
    stats = {}
    try:
        stats = json.load(open(stats_file))
    except Exception:
        pass
    print(json.dumps({
        "duration_s": duration,
        "workload_rc": max(w.returncode for w in workers),
        "n_workers": nprocs,
        "trajectory": samples[::3] + samples[-1:],
        "rss_first_mb": samples[0]["agent_rss_mb"] if samples else 0,
        "rss_last_mb": samples[-1]["agent_rss_mb"] if samples else 0,
        "agent_cpu_total_s": samples[-1]["agent_cpu_s"] if samples else 0,
        "rss_breakdown": breakdown,
        "agent_stats": stats,
    }, indent=1))


if __name__ == "__main__":
    main()
