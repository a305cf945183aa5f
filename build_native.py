#!/usr/bin/env python3
"""Build all native components in-tree.

Artifacts land in parca_agent_amd/native/ so they travel to the GPU box
with the gpurun snapshot (built .so files are git-ignored but NOT
gpurun-ignored). Components:

  _sampler.so          g++      perf_event_open CPU sampler (pybind11)
  _gpu.so              hipcc    shm-ring reader + CDNA4 PC-bucketing kernel +
                                RCCL node merge (pybind11, gfx950)
  libparca_rocprof.so  g++      rocprofiler-sdk interception tool, loaded into
                                target HIP processes via ROCP_TOOL_LIBRARIES

Usage: python build_native.py [--only sampler|gpu|rocprof] [--debug]
"""

from __future__ import annotations

import argparse
import os
import shlex
import subprocess
import sys
import sysconfig
from pathlib import Path

REPO = Path(__file__).resolve().parent
CSRC = REPO / "csrc"
OUT = REPO / "parca_agent_amd" / "native"
ROCM = Path(os.environ.get("ROCM_PATH", "/opt/rocm"))
GFX_ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

EXT_SUFFIX = sysconfig.get_config_var("EXT_SUFFIX") or ".so"


def _run(cmd: list[str]) -> None:
    print("+", " ".join(shlex.quote(c) for c in cmd), flush=True)
    subprocess.run(cmd, check=True, cwd=str(REPO))


def _pybind_includes() -> list[str]:
    import pybind11

    return [
        f"-I{pybind11.get_include()}",
        f"-I{sysconfig.get_path('include')}",
    ]


def _needs_build(target: Path, sources: list[Path]) -> bool:
    if not target.exists():
        return True
    t = target.stat().st_mtime
    return any(s.stat().st_mtime > t for s in sources)


def build_sampler(debug: bool = False, force: bool = False) -> Path:
    src = CSRC / "sampler" / "sampler.cc"
    out = OUT / f"_sampler{EXT_SUFFIX}"
    deps = [src, CSRC / "sampler" / "ehframe.cc"]
    if not force and not _needs_build(out, deps):
        return out
    opt = "-O1" if debug else "-O2"
    _run([
        "g++", opt, "-g", "-std=c++17", "-shared", "-fPIC",
        "-fvisibility=hidden", "-pthread",
        *_pybind_includes(),
        f"-I{CSRC}",
        str(src), "-o", str(out),
    ])
    return out


def build_rocprof_tool(debug: bool = False, force: bool = False) -> Path:
    src = CSRC / "rocprof" / "tool.cc"
    out = OUT / "libparca_rocprof.so"
    if not src.exists():
        return out
    if not force and not _needs_build(out, [src, CSRC / "rocprof" / "ring.h"]):
        return out
    opt = "-O1" if debug else "-O2"
    # hipcc (host-only, no --offload-arch needed: no device code) supplies
    # the HIP type definitions the rocprofiler-sdk headers depend on.
    _run([
        str(ROCM / "bin" / "hipcc"), opt, "-g", "-std=c++17", "-shared",
        "-fPIC", "-pthread",
        f"-I{CSRC}",
        str(src),
        f"-L{ROCM}/lib", "-lrocprofiler-sdk",
        f"-Wl,-rpath,{ROCM}/lib",
        "-o", str(out),
    ])
    return out


def build_gpu(debug: bool = False, force: bool = False) -> Path:
    srcs = [CSRC / "gpu" / "gpu_module.cc", CSRC / "gpu" / "bucketize.hip"]
    srcs = [s for s in srcs if s.exists()]
    out = OUT / f"_gpu{EXT_SUFFIX}"
    if not srcs:
        return out
    if not force and not _needs_build(out, srcs + [CSRC / "rocprof" / "ring.h"]):
        return out
    opt = "-O1" if debug else "-O3"
    hipcc = str(ROCM / "bin" / "hipcc")
    _run([
        hipcc, opt, "-g", "-std=c++17", "-shared", "-fPIC",
        f"--offload-arch={GFX_ARCH}",
        "-fvisibility=hidden",
        *_pybind_includes(),
        f"-I{CSRC}",
        *[str(s) for s in srcs],
        f"-L{ROCM}/lib", "-lrccl", "-lamdhip64",
        f"-Wl,-rpath,{ROCM}/lib",
        "-o", str(out),
    ])
    return out


def build_perl_offsets(debug: bool = False, force: bool = False) -> Path:
    """Extract perl struct offsets for the local build (interp/perl.py).

    Compiles tools/perl_offsets.c against the installed CORE headers and
    stores the JSON keyed by the perl binary's FileID. Skipped quietly
    when perl or its headers are absent — the perl unwinder then reports
    itself unavailable.
    """
    import json as _json
    import shutil
    import subprocess as _sp

    out = OUT / "perl_offsets.json"
    src = REPO / "tools" / "perl_offsets.c"
    perl = shutil.which("perl")
    if perl is None or not src.exists():
        return out
    if not force and not _needs_build(out, [src]):
        return out
    try:
        core = _sp.run(
            [perl, "-MConfig", "-e", "print $Config{archlibexp}"],
            capture_output=True, text=True, timeout=30).stdout + "/CORE"
        ccflags = _sp.run(
            [perl, "-MConfig", "-e", "print $Config{ccflags}"],
            capture_output=True, text=True, timeout=30).stdout.split()
        if not (Path(core) / "perl.h").exists():
            return out
        exe = OUT / "perl_offsets_bin"
        _run(["gcc", "-O0", *ccflags, f"-I{core}", str(src),
              "-o", str(exe)])
        offs = _json.loads(_sp.run([str(exe)], capture_output=True,
                                   text=True, timeout=30).stdout)
        from parca_agent_amd.elf import file_id as _file_id

        out.write_text(_json.dumps({
            "perl_path": perl,
            "file_id": _file_id(perl),
            "offsets": offs,
        }, indent=1))
        exe.unlink(missing_ok=True)
    except Exception as e:  # noqa: BLE001 - best-effort optional feature
        print(f"perl offsets skipped: {e}")
    return out


def build_heap_preload(debug: bool = False, force: bool = False) -> Path:
    """LD_PRELOAD allocation sampler for OOM heap profiles
    (csrc/heap/heap_preload.c; consumed by oom/heap.py)."""
    src = CSRC / "heap" / "heap_preload.c"
    out = OUT / "libparca_heap.so"
    if not src.exists():
        return out
    if not force and not _needs_build(out, [src]):
        return out
    opt = "-O1" if debug else "-O2"
    _run(["gcc", opt, "-g", "-shared", "-fPIC", "-pthread",
          "-fno-omit-frame-pointer", str(src), "-o", str(out), "-ldl"])
    return out


BUILDERS = {
    "sampler": build_sampler,
    "rocprof": build_rocprof_tool,
    "gpu": build_gpu,
    "perl": build_perl_offsets,
    "heap": build_heap_preload,
}


def build_all(debug: bool = False, force: bool = False) -> None:
    OUT.mkdir(parents=True, exist_ok=True)
    for name, fn in BUILDERS.items():
        fn(debug=debug, force=force)


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--only", choices=sorted(BUILDERS), default=None)
    ap.add_argument("--debug", action="store_true")
    ap.add_argument("--force", action="store_true")
    args = ap.parse_args()
    OUT.mkdir(parents=True, exist_ok=True)
    if args.only:
        BUILDERS[args.only](debug=args.debug, force=args.force)
    else:
        build_all(debug=args.debug, force=args.force)
    print("native build OK")
