"""GPU trace fixer tests (reference semantics: interpreter/gpu trace
fixer — times awaiting traces / traces awaiting times)."""

from parca_agent_amd.gpu.events import KernelDispatch, LaunchStack
from parca_agent_amd.gpu.fixer import GpuTraceFixer
from parca_agent_amd.gpu.pcbuckets import (
    BucketLayout,
    HostAccumulator,
    _popcount64,
)

import numpy as np


def _dispatch(corr, start=100, end=200):
    return KernelDispatch(
        correlation_id=corr, dispatch_id=corr, kernel_id=1,
        start_ns=start, end_ns=end, tid=1, gpu_index=0, pid=10,
        grid=(1, 1, 1), workgroup=(64, 1, 1),
        private_segment_size=0, group_segment_size=0)


def _stack(corr):
    return LaunchStack(correlation_id=corr, tid=1, pid=10,
                       ips=(0x1000, 0x2000))


def test_stack_then_time():
    f = GpuTraceFixer()
    assert f.add_stack(_stack(1)) == []
    done = f.add_times([_dispatch(1)])
    assert len(done) == 1
    assert done[0].stack.correlation_id == 1
    assert f.metrics.matched == 1


def test_dispatch_without_stack_emits_immediately():
    """Ring FIFO order means a dispatch with no pending stack will never
    get one (stack capture is rate-limited tool-side): emit kernel-only
    right away — no holding period."""
    f = GpuTraceFixer()
    done = f.add_times([_dispatch(2)])
    assert len(done) == 1
    assert done[0].stack is None
    assert f.metrics.unmatched == 1


def test_batch_mixed():
    f = GpuTraceFixer()
    f.add_stack(_stack(1))
    f.add_stack(_stack(3))
    done = f.add_times([_dispatch(1), _dispatch(2), _dispatch(3)])
    assert len(done) == 3
    with_stack = {d.dispatch.correlation_id for d in done
                  if d.stack is not None}
    assert with_stack == {1, 3}


def test_graph_launch_stack_reuse():
    """Many dispatch completions sharing one correlation id (one
    hipGraphLaunch) must all inherit the launch stack: matching does not
    pop."""
    f = GpuTraceFixer()
    f.add_stack(_stack(7))
    done = f.add_times([_dispatch(7), _dispatch(7), _dispatch(7)])
    assert all(d.stack is not None for d in done)
    assert f.metrics.matched == 3


def test_clear_stale_drops_aged_stacks():
    now = [0.0]
    f = GpuTraceFixer(max_age_seconds=5.0, clock=lambda: now[0])
    f.add_stack(_stack(2))
    now[0] = 6.0
    assert f.clear_stale() == []
    assert f.metrics.traces_cleared == 1
    assert f.pending_counts() == (0, 0)
    # After eviction, a late dispatch goes kernel-only.
    [done] = f.add_times([_dispatch(2)])
    assert done.stack is None


def test_pending_cap_evicts_oldest():
    now = [0.0]
    f = GpuTraceFixer(max_pending=10, clock=lambda: now[0])
    for i in range(20):
        now[0] += 0.001
        f.add_stack(_stack(i))
    assert f.pending_counts()[1] == 10
    assert f.metrics.traces_cleared == 10
    # The newest survive.
    [done] = f.add_times([_dispatch(19)])
    assert done.stack is not None


# -- bucket layout / host accumulator -------------------------------------


def test_bucket_layout():
    lay = BucketLayout(bucket_shift=6)
    assert lay.add(10, 1, 1000)   # 16 buckets
    assert lay.add(10, 2, 64)     # 1 bucket
    assert lay.add(20, 1, 128)    # 2 buckets (same co id, other pid)
    assert not lay.add(10, 1, 1000)  # duplicate
    assert lay.total_buckets == 16 + 1 + 2
    assert lay.bucket_range(0) == (0, 16)
    assert lay.bucket_range(1) == (16, 17)
    assert lay.key_of_slot(2) == (20, 1)


def test_host_accumulator():
    from parca_agent_amd.gpu.events import PC_SAMPLE_DTYPE

    lay = BucketLayout(bucket_shift=6)
    lay.add(10, 7, 640)  # 10 buckets
    acc = HostAccumulator(lay)

    samples = np.zeros(5, dtype=PC_SAMPLE_DTYPE)
    samples["code_object_id"] = 7
    samples["code_object_offset"] = [0, 63, 64, 128, 9999]  # last: OOR
    samples["exec_mask"] = [0xF, 0xFF, 1, 1, 1]
    acc.accumulate(10, samples)

    [(gpu, (hist, lanes))] = acc.read().items()
    assert gpu == -1
    assert hist[0] == 2      # offsets 0 and 63 share bucket 0
    assert hist[1] == 1
    assert hist[2] == 1
    assert acc.out_of_range == 1
    assert lanes[0] == 4 + 8  # popcounts of 0xF and 0xFF

    # unknown code object
    bad = np.zeros(1, dtype=PC_SAMPLE_DTYPE)
    bad["code_object_id"] = 999
    acc.accumulate(10, bad)
    assert acc.unknown_code_object == 1

    # read resets
    [(_, (hist2, _l))] = acc.read().items()
    assert hist2.sum() == 0


def test_popcount64():
    arr = np.array([0, 1, 0xFF, (1 << 64) - 1, 0x8000000000000000],
                   dtype=np.uint64)
    np.testing.assert_array_equal(_popcount64(arr), [0, 1, 8, 64, 1])


def test_host_accumulator_multi_pid():
    from parca_agent_amd.gpu.events import PC_SAMPLE_DTYPE

    lay = BucketLayout(bucket_shift=6)
    lay.add(10, 7, 128)
    lay.add(20, 7, 128)  # same co id, different pid
    acc = HostAccumulator(lay)
    s = np.zeros(1, dtype=PC_SAMPLE_DTYPE)
    s["code_object_id"] = 7
    s["exec_mask"] = 1
    acc.accumulate(10, s)
    acc.accumulate(20, s)
    [(_, (hist, _l))] = acc.read().items()
    assert hist[0] == 1  # pid 10 slot
    assert hist[2] == 1  # pid 20 slot starts at bucket 2


def test_host_accumulator_per_gpu():
    from parca_agent_amd.gpu.events import PC_SAMPLE_DTYPE

    lay = BucketLayout(bucket_shift=6)
    lay.add(10, 7, 128)
    acc = HostAccumulator(lay)
    s = np.zeros(2, dtype=PC_SAMPLE_DTYPE)
    s["code_object_id"] = 7
    s["exec_mask"] = 1
    acc.accumulate(10, s, gpu=0)
    acc.accumulate(10, s[:1], gpu=5)
    out = acc.read()
    assert set(out) == {0, 5}
    assert out[0][0][0] == 2
    assert out[5][0][0] == 1
