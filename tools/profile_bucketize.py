#!/usr/bin/env python3
"""rocprofv3 target: run the CDNA4 bucketize kernel in a tight loop so
per-kernel stats/counters can be collected (profiles/ evidence)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from parca_agent_amd.gpu.events import PC_SAMPLE_DTYPE
from parca_agent_amd.gpu.pcbuckets import BucketLayout, DeviceAccumulator


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 4_000_000
    iters = int(sys.argv[2]) if len(sys.argv) > 2 else 20
    rng = np.random.default_rng(1)
    layout = BucketLayout(bucket_shift=6)
    layout.add(pid=1, code_object_id=1, load_size=1 << 20)  # 16K buckets: LDS path
    layout.add(pid=2, code_object_id=1, load_size=64 * 40000)  # 40K: global path
    dev = DeviceAccumulator(layout, device=0)
    samples = np.zeros(n, dtype=PC_SAMPLE_DTYPE)
    samples["code_object_id"] = 1
    samples["code_object_offset"] = rng.integers(0, 1 << 20, size=n)
    samples["exec_mask"] = rng.integers(1, 1 << 63, size=n, dtype=np.uint64)
    big = samples.copy()
    big["code_object_offset"] = rng.integers(0, 64 * 40000, size=n)
    for _ in range(iters):
        dev.accumulate(1, samples)   # LDS-staged variant
        dev.accumulate(2, big)       # global-atomics variant
    hist, _ = dev.read()[-1]
    print("total bucketed:", int(hist.sum()))


if __name__ == "__main__":
    main()
