"""GPU trace fixer: attaches host launch stacks to kernel-dispatch
timings by correlation id.

The rocprofiler-native rebuild of the reference's CUPTI trace-fixer
(SURVEY.md §2.9 interpreter/gpu: traces awaiting times, times awaiting
traces, GC metrics cuda.* -> rocm.*), simplified by two properties of our
pipeline the CUPTI/eBPF one did not have:

1. Single ordered channel: launch stacks and dispatch completions travel
   through ONE shm ring written by one producer, so a captured stack is
   always observed BEFORE its dispatch completion. A dispatch with no
   pending stack therefore has none coming (stack capture is rate-limited
   tool-side) and is emitted immediately as kernel-only — no
   "times awaiting traces" holding period.
2. Graph launches: many dispatch completions can share one correlation id
   (one hipGraphLaunch). Stacks are matched WITHOUT popping and evicted
   only by age, so every replayed kernel inherits the launch stack.
"""

from __future__ import annotations

import time
from dataclasses import dataclass
from typing import Callable, Dict, List, Optional, Tuple

from .events import KernelDispatch, LaunchStack


@dataclass
class FixerMetrics:
    traces_awaiting_times: int = 0   # pending stacks (gauge)
    traces_cleared: int = 0          # stacks evicted by age/cap
    matched: int = 0                 # dispatches that got a stack
    unmatched: int = 0               # dispatches emitted kernel-only


@dataclass
class CompletedKernel:
    dispatch: KernelDispatch
    stack: Optional[LaunchStack]  # None when no stack was captured


class GpuTraceFixer:
    """Per-process correlation state machine."""

    def __init__(self, max_age_seconds: float = 10.0,
                 max_pending: int = 65536,
                 clock: Callable[[], float] = time.monotonic) -> None:
        self.max_age = max_age_seconds
        self.max_pending = max_pending
        self._clock = clock
        # correlation id -> (stack, arrival time); kept until aged out so
        # graph replays keep matching.
        self._stacks: Dict[int, Tuple[LaunchStack, float]] = {}
        self.metrics = FixerMetrics()

    # -- ingestion ---------------------------------------------------------

    def add_stack(self, stack: LaunchStack) -> List[CompletedKernel]:
        if len(self._stacks) >= self.max_pending:
            # Evict the oldest entry to stay bounded.
            oldest = min(self._stacks, key=lambda k: self._stacks[k][1])
            del self._stacks[oldest]
            self.metrics.traces_cleared += 1
        self._stacks[stack.correlation_id] = (stack, self._clock())
        return []

    def add_times(self, dispatches: List[KernelDispatch]
                  ) -> List[CompletedKernel]:
        """Kernel timings arrived (the caller batches ×100 like
        parcagpu.go:96). Every dispatch completes NOW — with its stack if
        one was captured, kernel-only otherwise."""
        out: List[CompletedKernel] = []
        for d in dispatches:
            entry = self._stacks.get(d.correlation_id)
            if entry is not None:
                self.metrics.matched += 1
                out.append(CompletedKernel(dispatch=d, stack=entry[0]))
            else:
                self.metrics.unmatched += 1
                out.append(CompletedKernel(dispatch=d, stack=None))
        return out

    # -- GC (MaybeClearAll analog) ----------------------------------------

    def clear_stale(self) -> List[CompletedKernel]:
        """Evict aged stacks. Returns [] (kept for interface stability
        with callers that flush pending output on GC)."""
        now = self._clock()
        stale = [k for k, (_, ts) in self._stacks.items()
                 if now - ts > self.max_age]
        for k in stale:
            del self._stacks[k]
            self.metrics.traces_cleared += 1
        self.metrics.traces_awaiting_times = len(self._stacks)
        return []

    def pending_counts(self) -> Tuple[int, int]:
        """(times awaiting stacks, stacks awaiting times): the first is
        structurally zero in this design."""
        return 0, len(self._stacks)
