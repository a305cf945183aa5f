"""Native extension loader.

Extensions are built in-tree by build_native.py (driven from
__graft_entry__.build()). On a GPU box the GPU extension must be present —
we fail loudly rather than falling back to a slow path silently.
"""

from __future__ import annotations

import importlib
import os
from types import ModuleType
from typing import Optional

_cache: dict = {}


def _load(name: str) -> ModuleType:
    mod = _cache.get(name)
    if mod is None:
        mod = importlib.import_module(f"parca_agent_amd.native.{name}")
        _cache[name] = mod
    return mod


def sampler() -> ModuleType:
    """The perf_event_open sampler extension (_sampler)."""
    return _load("_sampler")


def gpu() -> ModuleType:
    """The HIP/RCCL extension (_gpu). Required on GPU hosts."""
    try:
        return _load("_gpu")
    except ImportError as e:
        raise ImportError(
            "parca_agent_amd.native._gpu is not built. Run "
            "`python build_native.py --only gpu` (requires hipcc; "
            "cross-compiles for gfx950 without a GPU)."
        ) from e


def try_sampler() -> Optional[ModuleType]:
    try:
        return sampler()
    except ImportError:
        return None


def rocprof_tool_path() -> str:
    """Path of the rocprofiler interception library to inject into HIP
    processes via ROCP_TOOL_LIBRARIES."""
    return os.path.join(os.path.dirname(__file__), "libparca_rocprof.so")
