"""GPU trace fixer: marries kernel-dispatch timings with host launch
stacks by correlation id.

The rocprofiler-native rebuild of the reference's CUPTI trace-fixer state
machine (SURVEY.md §2.9 interpreter/gpu: InterceptTrace holds CUDA traces
until timing arrives, AddTimes marries timing batches with retained
traces, MaybeClearAll GCs both sides with eviction metrics
cuda.times_awaiting_traces / traces_awaiting_times / *_cleared,
metrics/all.go:1388-1441 — ours are rocm.*).

Differences by design: correlation ids come from rocprofiler (one per HIP
launch API call), launch stacks are captured in-process by the tool
library rather than by an eBPF uprobe, and both sides arrive over the
same shm ring so there is exactly one producer clock.
"""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Tuple

from .events import KernelDispatch, LaunchStack


@dataclass
class FixerMetrics:
    times_awaiting_traces: int = 0
    traces_awaiting_times: int = 0
    times_cleared: int = 0
    traces_cleared: int = 0
    matched: int = 0


@dataclass
class CompletedKernel:
    dispatch: KernelDispatch
    stack: Optional[LaunchStack]  # None when launch stacks are disabled


class GpuTraceFixer:
    """Per-process correlation state machine."""

    def __init__(self, max_age_seconds: float = 10.0,
                 max_pending: int = 65536,
                 clock: Callable[[], float] = time.monotonic) -> None:
        self.max_age = max_age_seconds
        self.max_pending = max_pending
        self._clock = clock
        # correlation id -> (stack, arrival)
        self._stacks: Dict[int, Tuple[LaunchStack, float]] = {}
        # correlation id -> (dispatch, arrival); dispatches whose stack has
        # not arrived yet (stack events can trail the async completion).
        self._times: Dict[int, Tuple[KernelDispatch, float]] = {}
        self.metrics = FixerMetrics()
        self.expect_stacks = True

    # -- ingestion ---------------------------------------------------------

    def add_stack(self, stack: LaunchStack) -> List[CompletedKernel]:
        """A host launch stack arrived (InterceptTrace analog)."""
        pending = self._times.pop(stack.correlation_id, None)
        if pending is not None:
            self.metrics.matched += 1
            return [CompletedKernel(dispatch=pending[0], stack=stack)]
        if len(self._stacks) < self.max_pending:
            self._stacks[stack.correlation_id] = (stack, self._clock())
        else:
            self.metrics.traces_cleared += 1
        return []

    def add_times(self, dispatches: List[KernelDispatch]
                  ) -> List[CompletedKernel]:
        """A batch of kernel timings arrived (AddTimes analog; the caller
        batches ×100 like parcagpu.go:96)."""
        out: List[CompletedKernel] = []
        for d in dispatches:
            if not self.expect_stacks:
                out.append(CompletedKernel(dispatch=d, stack=None))
                continue
            entry = self._stacks.pop(d.correlation_id, None)
            if entry is not None:
                self.metrics.matched += 1
                out.append(CompletedKernel(dispatch=d, stack=entry[0]))
            elif len(self._times) < self.max_pending:
                self._times[d.correlation_id] = (d, self._clock())
            else:
                self.metrics.times_cleared += 1
        return out

    # -- GC (MaybeClearAll analog) ----------------------------------------

    def clear_stale(self) -> List[CompletedKernel]:
        """Evict aged entries; stale timings are emitted WITHOUT a stack
        (kernel-only attribution beats dropping the sample), stale stacks
        are dropped. Returns emitted kernels."""
        now = self._clock()
        out: List[CompletedKernel] = []
        stale_times = [k for k, (_, ts) in self._times.items()
                       if now - ts > self.max_age]
        for k in stale_times:
            d, _ = self._times.pop(k)
            out.append(CompletedKernel(dispatch=d, stack=None))
            self.metrics.times_cleared += 1
        stale_stacks = [k for k, (_, ts) in self._stacks.items()
                        if now - ts > self.max_age]
        for k in stale_stacks:
            self._stacks.pop(k)
            self.metrics.traces_cleared += 1
        self.metrics.times_awaiting_traces = len(self._times)
        self.metrics.traces_awaiting_times = len(self._stacks)
        return out

    def pending_counts(self) -> Tuple[int, int]:
        return len(self._times), len(self._stacks)
