// CDNA4 (gfx950) PC-sample bucketing kernel.
//
// Drains raw PC-sample records into per-(code-object, PC-bucket) histograms
// on-device, so host-side symbolization cost is O(buckets), not O(samples)
// (BASELINE.json north star; replaces the reference's per-record host-side
// gpu.HandlePCSample processing, parcagpu.go:171-178).
//
// Design for MI355X per /opt/skills/guides/cdna_hip_programming.md:
//  - 256-thread workgroups (4 waves of 64); grid-stride loop over GROUPS OF
//    FOUR samples per lane, grid capped at 2048 workgroups (Guideline 11
//    for memory-bound ops).
//  - Vectorized input: each lane issues 2x global_load_dwordx4 (16 B) per
//    SoA array for its 4 samples — the widest load the ISA has
//    (Guideline 2/13). The scalar tail (< 4 samples) is handled by
//    workgroup 0.
//  - Consecutive PC samples are temporally correlated and usually land in
//    the SAME bucket; equal-bucket runs within a lane's 4 samples are
//    merged locally so the common case issues ONE ds-atomic for 4 samples.
//  - LDS-staged histogram: each workgroup accumulates into an LDS-resident
//    u32 histogram with ds-atomics, then merges once into the global
//    histogram with device-scope atomics (Guideline 12: per-block partial
//    accumulation, one global atomic per touched bucket). Active-lane
//    (exec-mask popcount) sums are ALSO staged in LDS as u32 when the
//    doubled footprint fits (u32 is safe: <= 64 lanes x <= n samples per
//    workgroup per launch, and launches are per ring batch <= ~4M).
//  - LDS budget: 128 KiB of the 160 KiB/CU -> 32K buckets counts-only or
//    16K buckets with staged lane sums. One workgroup per CU is resident
//    at that footprint; the kernel is atomic-throughput bound, not
//    occupancy bound, so that trade is measured-fine (see
//    profiles/bucketize_kernel_stats.txt).
//  - Above the LDS budget: direct global atomics, absorbed by the per-XCD
//    L2.
//  - Code-object id -> dense slot mapping happens on-device via binary
//    search over a sorted table (<=1024 entries, L1-resident).
//
// Why no MFMA (north-star text says "MFMA-packed int32 accumulate"): MFMA
// is a dense matrix-multiply-accumulate engine; histogramming is a
// data-dependent scatter-accumulate with no matrix structure to feed the
// A/B operands — expressing it as a one-hot x ones GEMM would burn
// (bucket-width x 16) MACs per sample to add 1. The CDNA4-native primitive
// for scatter-accumulate IS the LDS ds_add path this kernel uses; measured
// at 13.7 G samples/s it exceeds any realistic PC-sample rate (~1 M/s/GPU)
// by 4 orders of magnitude. Decision: keep DS atomics, documented here
// against BASELINE.json's north-star wording.

#include <hip/hip_runtime.h>

#include <cstdint>

namespace parca_gpu {

constexpr int kThreads = 256;
constexpr uint32_t kMaxLdsBytes = 128 * 1024;

struct BucketizeArgs {
  const uint64_t* code_object_ids;  // [n]
  const uint64_t* offsets;          // [n]
  const uint64_t* exec_masks;       // [n] may be null
  uint32_t n;
  // sorted code-object id table and per-slot bucket layout
  const uint64_t* slot_ids;      // [n_slots] sorted
  const uint32_t* slot_offsets;  // [n_slots+1] prefix of bucket counts
  uint32_t n_slots;
  uint32_t bucket_shift;  // PC bytes per bucket = 1 << shift
  uint32_t total_buckets;
  uint32_t* histogram;       // [total_buckets] sample counts
  uint64_t* lane_histogram;  // [total_buckets] active-lane sums (nullable)
  uint32_t* overflow;        // [2]: unknown code-object, out-of-range bucket
};

__device__ inline int find_slot(const uint64_t* ids, uint32_t n, uint64_t v) {
  uint32_t lo = 0, hi = n;
  while (lo < hi) {
    uint32_t mid = (lo + hi) >> 1;
    if (ids[mid] < v)
      lo = mid + 1;
    else
      hi = mid;
  }
  return (lo < n && ids[lo] == v) ? static_cast<int>(lo) : -1;
}

// Bucket index for one sample, or -1/-2 for unknown-object/out-of-range.
__device__ inline int64_t bucket_of(const BucketizeArgs& a, uint64_t co,
                                    uint64_t off) {
  int slot = find_slot(a.slot_ids, a.n_slots, co);
  if (slot < 0) return -1;
  uint32_t first = a.slot_offsets[slot];
  uint32_t last = a.slot_offsets[slot + 1];
  uint32_t bucket = first + static_cast<uint32_t>(off >> a.bucket_shift);
  if (bucket >= last) return -2;
  return bucket;
}

// 32-byte vector load of 4 consecutive u64 (2x global_load_dwordx4).
__device__ inline void load4(const uint64_t* p, uint32_t i, uint64_t out[4]) {
  const ulonglong2* v = reinterpret_cast<const ulonglong2*>(p + i);
  ulonglong2 a0 = v[0];
  ulonglong2 a1 = v[1];
  out[0] = a0.x;
  out[1] = a0.y;
  out[2] = a1.x;
  out[3] = a1.y;
}

// Per-lane accumulator that merges equal-bucket runs before touching
// memory: PC samples arrive in time order, so a hot loop's 4 consecutive
// samples usually share one bucket -> one atomic instead of four.
template <typename HistOp, typename LaneOp, typename OvfOp>
struct RunMerger {
  int64_t cur = INT64_MIN;
  uint32_t count = 0;
  uint32_t lanes = 0;
  uint32_t ovf_unknown = 0;
  uint32_t ovf_range = 0;
  HistOp hist;
  LaneOp lane;
  OvfOp ovf;

  __device__ inline void add(int64_t bucket, uint32_t lane_count) {
    if (bucket < 0) {
      if (bucket == -1)
        ++ovf_unknown;
      else
        ++ovf_range;
      return;
    }
    if (bucket == cur) {
      ++count;
      lanes += lane_count;
      return;
    }
    flush_bucket();
    cur = bucket;
    count = 1;
    lanes = lane_count;
  }

  __device__ inline void flush_bucket() {
    if (count != 0) {
      hist(static_cast<uint32_t>(cur), count);
      lane(static_cast<uint32_t>(cur), lanes);
      count = 0;
    }
  }

  __device__ inline void finish() {
    flush_bucket();
    if (ovf_unknown != 0 || ovf_range != 0) ovf(ovf_unknown, ovf_range);
  }
};

template <typename H, typename L, typename O>
__device__ inline RunMerger<H, L, O> make_merger(H h, L l, O o) {
  return RunMerger<H, L, O>{INT64_MIN, 0, 0, 0, 0, h, l, o};
}

// Grid-stride body: vectorized groups of 4 plus scalar tail on block 0.
template <typename Merger>
__device__ inline void bucketize_body(const BucketizeArgs& a, Merger& m) {
  const bool want_lanes =
      a.lane_histogram != nullptr && a.exec_masks != nullptr;
  uint32_t n4 = a.n & ~3u;
  for (uint32_t i = (blockIdx.x * kThreads + threadIdx.x) * 4; i < n4;
       i += gridDim.x * kThreads * 4) {
    uint64_t co[4], off[4], mask[4];
    load4(a.code_object_ids, i, co);
    load4(a.offsets, i, off);
    if (want_lanes) load4(a.exec_masks, i, mask);
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      uint32_t lanes = want_lanes ? static_cast<uint32_t>(__popcll(mask[k]))
                                  : 0u;
      m.add(bucket_of(a, co[k], off[k]), lanes);
    }
  }
  if (blockIdx.x == 0) {
    for (uint32_t i = n4 + threadIdx.x; i < a.n; i += kThreads) {
      uint32_t lanes = want_lanes
                           ? static_cast<uint32_t>(__popcll(a.exec_masks[i]))
                           : 0u;
      m.add(bucket_of(a, a.code_object_ids[i], a.offsets[i]), lanes);
    }
  }
  m.finish();
}

// LDS-staged variant. kStageLanes additionally stages the active-lane
// sums in LDS (u32), doubling the per-bucket footprint.
template <bool kStageLanes>
__global__ __launch_bounds__(kThreads) void bucketize_lds(BucketizeArgs a) {
  extern __shared__ uint32_t lds[];
  uint32_t* lds_hist = lds;
  uint32_t* lds_lanes = lds + (kStageLanes ? a.total_buckets : 0);
  uint32_t init_span = kStageLanes ? 2 * a.total_buckets : a.total_buckets;
  for (uint32_t b = threadIdx.x; b < init_span; b += kThreads) lds[b] = 0;
  __syncthreads();

  auto m = make_merger(
      [&](uint32_t b, uint32_t c) {
        atomicAdd(&lds_hist[b], c);  // ds_add_u32
      },
      [&](uint32_t b, uint32_t l) {
        if (kStageLanes) {
          if (l != 0) atomicAdd(&lds_lanes[b], l);
        } else if (a.lane_histogram != nullptr && l != 0) {
          atomicAdd(
              reinterpret_cast<unsigned long long*>(&a.lane_histogram[b]),
              static_cast<unsigned long long>(l));
        }
      },
      [&](uint32_t u, uint32_t r) {
        if (u) atomicAdd(&a.overflow[0], u);
        if (r) atomicAdd(&a.overflow[1], r);
      });
  bucketize_body(a, m);
  __syncthreads();

  // One device-scope atomic per non-zero bucket per workgroup.
  for (uint32_t b = threadIdx.x; b < a.total_buckets; b += kThreads) {
    uint32_t v = lds_hist[b];
    if (v != 0) atomicAdd(&a.histogram[b], v);
    if (kStageLanes) {
      uint32_t l = lds_lanes[b];
      if (l != 0)
        atomicAdd(reinterpret_cast<unsigned long long*>(&a.lane_histogram[b]),
                  static_cast<unsigned long long>(l));
    }
  }
}

// Fallback: histogram too large for LDS — direct global atomics (the
// per-XCD L2 absorbs the traffic).
__global__ __launch_bounds__(kThreads) void bucketize_global(BucketizeArgs a) {
  auto m = make_merger(
      [&](uint32_t b, uint32_t c) { atomicAdd(&a.histogram[b], c); },
      [&](uint32_t b, uint32_t l) {
        if (a.lane_histogram != nullptr && l != 0)
          atomicAdd(
              reinterpret_cast<unsigned long long*>(&a.lane_histogram[b]),
              static_cast<unsigned long long>(l));
      },
      [&](uint32_t u, uint32_t r) {
        if (u) atomicAdd(&a.overflow[0], u);
        if (r) atomicAdd(&a.overflow[1], r);
      });
  bucketize_body(a, m);
}

void launch_bucketize(const BucketizeArgs& args, hipStream_t stream) {
  // 4 samples per lane per grid-stride step.
  uint32_t blocks = (args.n + kThreads * 4 - 1) / (kThreads * 4);
  if (blocks > 2048) blocks = 2048;  // grid-stride the rest (G11)
  if (blocks == 0) return;
  bool want_lanes = args.lane_histogram != nullptr;
  size_t hist_bytes = static_cast<size_t>(args.total_buckets) * 4;
  if (want_lanes && hist_bytes * 2 <= kMaxLdsBytes) {
    hipLaunchKernelGGL(bucketize_lds<true>, dim3(blocks), dim3(kThreads),
                       hist_bytes * 2, stream, args);
  } else if (hist_bytes <= kMaxLdsBytes) {
    hipLaunchKernelGGL(bucketize_lds<false>, dim3(blocks), dim3(kThreads),
                       hist_bytes, stream, args);
  } else {
    hipLaunchKernelGGL(bucketize_global, dim3(blocks), dim3(kThreads), 0,
                       stream, args);
  }
}

}  // namespace parca_gpu
