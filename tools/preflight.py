#!/usr/bin/env python3
"""Environment preflight: verify every capability the agent needs and
print one PASS/WARN/FAIL line per check. Exit code 1 if any FAIL.

Run on a new node before deploying (docs/OPERATIONS.md privileges
section); each check mirrors a failure mode met on real machines.
"""

import ctypes
import json
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

RESULTS = []


def check(name, status, detail=""):
    RESULTS.append((name, status, detail))
    print(f"[{status:4s}] {name}" + (f" — {detail}" if detail else ""))


def main() -> int:
    # perf_event_open
    try:
        paranoid = int(open("/proc/sys/kernel/perf_event_paranoid").read())
        if os.geteuid() == 0 or paranoid <= 1:
            check("perf_event system-wide sampling", "PASS",
                  f"paranoid={paranoid}, euid={os.geteuid()}")
        else:
            check("perf_event system-wide sampling", "FAIL",
                  f"paranoid={paranoid} needs root/CAP_PERFMON")
    except OSError as e:
        check("perf_event system-wide sampling", "FAIL", str(e))

    # native extensions
    try:
        from parca_agent_amd.native import sampler

        sampler()
        check("native sampler extension", "PASS")
    except Exception as e:
        check("native sampler extension", "FAIL",
              f"{e} (run: python build_native.py)")

    tool = os.path.join(REPO, "parca_agent_amd", "native",
                        "libparca_rocprof.so")
    check("rocprofiler tool library", "PASS" if os.path.exists(tool)
          else "FAIL", tool)

    # Yama / cross-process reads
    try:
        scope = int(open("/proc/sys/kernel/yama/ptrace_scope").read())
    except OSError:
        scope = 0
    caps = 0
    try:
        for line in open("/proc/self/status"):
            if line.startswith("CapEff:"):
                caps = int(line.split()[1], 16)
    except OSError:
        pass
    has_ptrace = bool(caps & (1 << 19))
    if scope == 0 or has_ptrace:
        check("cross-process memory reads (interp unwinding)", "PASS",
              f"yama={scope}, cap_sys_ptrace={has_ptrace}")
    else:
        check("cross-process memory reads (interp unwinding)", "WARN",
              f"yama={scope} without CAP_SYS_PTRACE: only processes "
              "running the GPU tool are readable (PR_SET_PTRACER_ANY)")

    # uprobes
    check("uprobe PMU (probes service)",
          "PASS" if os.path.exists(
              "/sys/bus/event_source/devices/uprobe/type") else "WARN",
          "probes disabled without it")

    # kallsyms visibility
    try:
        line = open("/proc/kallsyms").readline()
        hidden = line.split()[0] == "0000000000000000"
        check("kernel symbolization (/proc/kallsyms)",
              "WARN" if hidden else "PASS",
              "kptr_restrict hides addresses" if hidden else "")
    except OSError as e:
        check("kernel symbolization (/proc/kallsyms)", "WARN", str(e))

    # shm dir
    shm = os.environ.get("PARCA_GPU_SHM_DIR", "/dev/shm")
    check("GPU shm ring directory", "PASS" if os.access(shm, os.W_OK)
          else "FAIL", shm)

    # GPU presence + PC sampling (optional)
    try:
        from parca_agent_amd.native import gpu

        n = gpu().hip_device_count()
        check("HIP devices", "PASS" if n > 0 else "WARN", f"count={n}")
    except Exception as e:
        check("HIP devices", "WARN", f"gpu extension: {e}")

    # perl/python unwinder calibration artifacts
    try:
        from parca_agent_amd.interp.python import calibrate

        check("CPython offset calibration", "PASS" if calibrate()
              else "WARN")
    except Exception as e:
        check("CPython offset calibration", "WARN", str(e))
    off = os.path.join(REPO, "parca_agent_amd", "native",
                       "perl_offsets.json")
    check("Perl offsets", "PASS" if os.path.exists(off) else "WARN",
          "build_native.py --only perl")

    fails = [r for r in RESULTS if r[1] == "FAIL"]
    print(json.dumps({"pass": sum(r[1] == "PASS" for r in RESULTS),
                      "warn": sum(r[1] == "WARN" for r in RESULTS),
                      "fail": len(fails)}))
    return 1 if fails else 0


if __name__ == "__main__":
    sys.exit(main())
