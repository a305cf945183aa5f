"""PC-sample bucket accumulation.

Two backends with identical semantics:

- DeviceAccumulator: uploads raw sample batches to the MI355X and runs the
  CDNA4 LDS-staged histogram kernel (csrc/gpu/bucketize.hip) — host cost
  stays O(buckets), per the BASELINE.json north star.
- HostAccumulator: numpy bincount fallback for agents running on non-GPU
  hosts (and for CPU tests — both backends are verified against each
  other in tests/test_gpu_bucketize.py).

Bucket space: each (pid, code_object_id) gets a contiguous slot of
ceil(load_size / 2^bucket_shift) buckets. The layout is rebuilt when new
code objects appear (rare) — pending device state is flushed first.
"""

from __future__ import annotations

import logging
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

import numpy as np

log = logging.getLogger("parca_agent_amd.gpu.pcbuckets")

DEFAULT_BUCKET_SHIFT = 6  # 64-byte PC buckets: ~instruction-cluster granular


@dataclass
class BucketKey:
    pid: int
    code_object_id: int


class BucketLayout:
    """Maps (pid, code_object_id) -> [first_bucket, last_bucket) slots."""

    def __init__(self, bucket_shift: int = DEFAULT_BUCKET_SHIFT,
                 max_total_buckets: int = 1 << 22) -> None:
        self.bucket_shift = bucket_shift
        self.max_total = max_total_buckets
        self._slots: Dict[Tuple[int, int], int] = {}   # key -> slot index
        self._sizes: List[int] = []                    # buckets per slot
        self._keys: List[Tuple[int, int]] = []
        self._offsets: Optional[np.ndarray] = None     # prefix sums

    def add(self, pid: int, code_object_id: int, load_size: int) -> bool:
        key = (pid, code_object_id)
        if key in self._slots:
            return False
        n_buckets = max(1, (load_size + (1 << self.bucket_shift) - 1)
                        >> self.bucket_shift)
        if self.total_buckets + n_buckets > self.max_total:
            log.warning("bucket layout full; dropping code object %s", key)
            return False
        self._slots[key] = len(self._keys)
        self._keys.append(key)
        self._sizes.append(n_buckets)
        self._offsets = None
        return True

    @property
    def total_buckets(self) -> int:
        return int(sum(self._sizes))

    @property
    def n_slots(self) -> int:
        return len(self._keys)

    def offsets(self) -> np.ndarray:
        if self._offsets is None:
            self._offsets = np.zeros(len(self._sizes) + 1, dtype=np.uint32)
            np.cumsum(self._sizes, out=self._offsets[1:])
        return self._offsets

    def slot_of(self, pid: int, code_object_id: int) -> Optional[int]:
        return self._slots.get((pid, code_object_id))

    def key_of_slot(self, slot: int) -> Tuple[int, int]:
        return self._keys[slot]

    def bucket_range(self, slot: int) -> Tuple[int, int]:
        offs = self.offsets()
        return int(offs[slot]), int(offs[slot + 1])

    # Per-process slot-id table for the device kernel: the device searches
    # by raw code_object_id, so the table must be per-pid (ids collide
    # across processes). We therefore key device accumulation per pid.
    def device_tables(self, pid: int) -> Tuple[np.ndarray, np.ndarray,
                                               np.ndarray]:
        """(sorted co_ids, per-slot global offsets alignment, slot index
        remap) for one pid: returns (slot_ids, slot_offsets, global_base)
        where slot_offsets is LOCAL prefix (0-based) and global_base maps
        local slot -> global bucket offset."""
        entries = [(co, self._slots[(p, co)]) for (p, co) in self._keys
                   if p == pid]
        entries.sort()
        slot_ids = np.array([e[0] for e in entries], dtype=np.uint64)
        local_sizes = [self._sizes[e[1]] for e in entries]
        local_offsets = np.zeros(len(entries) + 1, dtype=np.uint32)
        np.cumsum(local_sizes, out=local_offsets[1:])
        global_base = np.array(
            [self.offsets()[e[1]] for e in entries], dtype=np.uint32)
        return slot_ids, local_offsets, global_base


class HostAccumulator:
    """Per-GPU histograms: the ``gpu`` accumulation key keeps the GPU
    dimension so an 8-GPU node's daemon agent fans samples out with
    per-GPU labels instead of folding all devices together."""

    def __init__(self, layout: BucketLayout) -> None:
        self.layout = layout
        self._hists: Dict[int, Tuple[np.ndarray, np.ndarray]] = {}
        self.unknown_code_object = 0
        self.out_of_range = 0

    def _arrays(self, gpu: int) -> Tuple[np.ndarray, np.ndarray]:
        total = self.layout.total_buckets
        pair = self._hists.get(gpu)
        if pair is None or len(pair[0]) < total:
            hist = np.zeros(total, dtype=np.uint64)
            lanes = np.zeros(total, dtype=np.uint64)
            if pair is not None:
                hist[: len(pair[0])] = pair[0]
                lanes[: len(pair[1])] = pair[1]
            pair = (hist, lanes)
            self._hists[gpu] = pair
        return pair

    def accumulate(self, pid: int, samples: np.ndarray,
                   gpu: int = -1) -> None:
        """samples: structured PC_SAMPLE_DTYPE array from one ring batch."""
        if len(samples) == 0:
            return
        hist, lane_hist = self._arrays(gpu)
        co_ids = samples["code_object_id"]
        offsets = samples["code_object_offset"]
        exec_masks = samples["exec_mask"]

        slot_ids, local_offsets, global_base = self.layout.device_tables(pid)
        if len(slot_ids) == 0:
            self.unknown_code_object += len(samples)
            return
        idx = np.searchsorted(slot_ids, co_ids)
        idx_clipped = np.minimum(idx, len(slot_ids) - 1)
        known = slot_ids[idx_clipped] == co_ids
        self.unknown_code_object += int((~known).sum())
        if not known.any():
            return
        sel_slot = idx_clipped[known]
        bucket_local = (offsets[known] >> self.layout.bucket_shift).astype(
            np.uint64)
        slot_size = (local_offsets[sel_slot + 1] -
                     local_offsets[sel_slot]).astype(np.uint64)
        in_range = bucket_local < slot_size
        self.out_of_range += int((~in_range).sum())
        if not in_range.any():
            return
        buckets = (global_base[sel_slot[in_range]].astype(np.uint64) +
                   bucket_local[in_range])
        np.add.at(hist, buckets, 1)
        lanes = _popcount64(exec_masks[known][in_range])
        np.add.at(lane_hist, buckets, lanes)

    def read(self, also_reset: bool = True
             ) -> Dict[int, Tuple[np.ndarray, np.ndarray]]:
        """gpu -> (hist, lane_hist) in the global bucket space."""
        out = {}
        for gpu in list(self._hists):
            h, l = self._arrays(gpu)
            out[gpu] = (h.copy(), l.copy())
            if also_reset:
                h[:] = 0
                l[:] = 0
        return out


def _popcount64(arr: np.ndarray) -> np.ndarray:
    """Vectorized 64-bit popcount."""
    x = arr.astype(np.uint64).copy()
    m1 = np.uint64(0x5555555555555555)
    m2 = np.uint64(0x3333333333333333)
    m4 = np.uint64(0x0F0F0F0F0F0F0F0F)
    h01 = np.uint64(0x0101010101010101)
    x = x - ((x >> np.uint64(1)) & m1)
    x = (x & m2) + ((x >> np.uint64(2)) & m2)
    x = (x + (x >> np.uint64(4))) & m4
    return (x * h01) >> np.uint64(56)


class DeviceAccumulator:
    """Per-(pid, gpu) CDNA4 device bucketizers sharing one BucketLayout.

    Device histograms are per-pid (code-object ids collide across
    processes) and per source GPU (the daemon agent fans an 8-GPU
    node's samples out with per-GPU labels); read() folds them into
    per-GPU arrays over the global bucket space.
    """

    def __init__(self, layout: BucketLayout, device: int = 0) -> None:
        from ..native import gpu as native_gpu

        self._native = native_gpu()
        if self._native.hip_device_count() == 0:
            raise RuntimeError("no HIP device for DeviceAccumulator")
        self.layout = layout
        self.device = device
        # (pid, gpu) -> DeviceBucketizer
        self._buckets: Dict[Tuple[int, int], object] = {}
        self._bases: Dict[int, np.ndarray] = {}     # pid -> global_base
        self._local_offsets: Dict[int, np.ndarray] = {}
        self._layout_gen: Dict[Tuple[int, int], int] = {}
        self._gen = 0
        self.unknown_code_object = 0
        self.out_of_range = 0
        # Global-bucket-space carryover from pre-rebuild device state,
        # per gpu. Global slot offsets are STABLE (the layout is
        # append-only), so folding early is safe.
        self._pending: Dict[int, Tuple[np.ndarray, np.ndarray]] = {}

    def layout_changed(self) -> None:
        self._gen += 1

    def _pending_arrays(self, gpu: int) -> Tuple[np.ndarray, np.ndarray]:
        total = self.layout.total_buckets
        pair = self._pending.get(gpu)
        if pair is None or len(pair[0]) < total:
            hist = np.zeros(total, dtype=np.uint64)
            lanes = np.zeros(total, dtype=np.uint64)
            if pair is not None:
                hist[: len(pair[0])] = pair[0]
                lanes[: len(pair[1])] = pair[1]
            pair = (hist, lanes)
            self._pending[gpu] = pair
        return pair

    def _fold_global(self, pid: int, gpu: int, h: np.ndarray,
                     l: np.ndarray) -> None:
        """Fold one pid's LOCAL-space histograms into the pending global
        arrays using that pid's current base/offset tables."""
        hist, lanes = self._pending_arrays(gpu)
        base = self._bases[pid]
        offs = self._local_offsets[pid]
        for i in range(len(base)):
            lo, hi = int(offs[i]), int(offs[i + 1])
            g = int(base[i])
            hist[g : g + (hi - lo)] += h[lo:hi]
            lanes[g : g + (hi - lo)] += l[lo:hi]

    def _bucketizer(self, pid: int, gpu: int):
        key = (pid, gpu)
        if self._layout_gen.get(key) != self._gen:
            # Flush existing device state into the global pending arrays
            # (with the OLD tables) before rebuilding for the new layout.
            old = self._buckets.pop(key, None)
            if old is not None:
                hist, lanes, overflow = old.read(True)
                self.unknown_code_object += int(overflow[0])
                self.out_of_range += int(overflow[1])
                self._fold_global(pid, gpu, hist.astype(np.uint64),
                                  lanes.astype(np.uint64))
            slot_ids, local_offsets, global_base = \
                self.layout.device_tables(pid)
            if len(slot_ids) == 0:
                return None
            self._buckets[key] = self._native.DeviceBucketizer(
                self.device, slot_ids, local_offsets,
                self.layout.bucket_shift)
            self._bases[pid] = global_base
            self._local_offsets[pid] = local_offsets
            self._layout_gen[key] = self._gen
        return self._buckets.get(key)

    def accumulate(self, pid: int, samples: np.ndarray,
                   gpu: int = -1) -> None:
        b = self._bucketizer(pid, gpu)
        if b is None:
            self.unknown_code_object += len(samples)
            return
        b.accumulate(np.ascontiguousarray(samples["code_object_id"]),
                     np.ascontiguousarray(samples["code_object_offset"]),
                     np.ascontiguousarray(samples["exec_mask"]))

    def read(self, also_reset: bool = True
             ) -> Dict[int, Tuple[np.ndarray, np.ndarray]]:
        """Fold all per-(pid, gpu) device histograms plus carryover into
        per-GPU arrays over the global bucket space and reset. Always
        destructive (the device state is drained into the result); the
        parameter exists for API parity with HostAccumulator."""
        del also_reset
        for (pid, gpu), b in list(self._buckets.items()):
            h, l, overflow = b.read(True)
            self.unknown_code_object += int(overflow[0])
            self.out_of_range += int(overflow[1])
            self._fold_global(pid, gpu, h.astype(np.uint64),
                              l.astype(np.uint64))
        out = {}
        for gpu in list(self._pending):
            hist, lanes = self._pending_arrays(gpu)
            out[gpu] = (hist.copy(), lanes.copy())
            hist[:] = 0
            lanes[:] = 0
        return out

    def drop_process(self, pid: int) -> None:
        for key in [k for k in self._buckets if k[0] == pid]:
            self._buckets.pop(key, None)
