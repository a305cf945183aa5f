// CDNA4 (gfx950) PC-sample bucketing kernel.
//
// Drains raw PC-sample records into per-(code-object, PC-bucket) histograms
// on-device, so host-side symbolization cost is O(buckets), not O(samples)
// (BASELINE.json north star; replaces the reference's per-record host-side
// gpu.HandlePCSample processing, parcagpu.go:171-178).
//
// Design for MI355X per /opt/skills/guides/cdna_hip_programming.md:
//  - 256-thread workgroups (4 waves of 64); grid-stride loop, grid capped
//    at ~2048 workgroups (Guideline 11 for memory-bound ops).
//  - LDS-staged histogram: each workgroup accumulates into an LDS-resident
//    u32 histogram with ds-atomics (conflict-free in the common case where
//    neighbouring samples hit distinct buckets), then merges once into the
//    global histogram with device-scope atomics (Guideline 12: per-block
//    partial reduction first, one global atomic per touched bucket).
//  - When the histogram exceeds the LDS budget (160 KiB/CU; we cap our
//    use at 32K buckets = 128 KiB to keep 2 workgroups/CU resident),
//    falls back to direct global atomics, which the per-XCD L2 absorbs.
//  - Input samples are read as 4×u64 vectors per lane (code_object_id,
//    offset packed by the host into a contiguous SoA) — coalesced 8B/lane
//    loads (Guideline 2/13).
//  - Code-object id -> dense slot mapping happens on-device via binary
//    search over a sorted table (<=1024 entries, L1-resident).

#include <hip/hip_runtime.h>

#include <cstdint>

namespace parca_gpu {

constexpr int kThreads = 256;
constexpr uint32_t kMaxLdsBuckets = 32 * 1024;  // 128 KiB of u32 counts

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

__device__ inline void accumulate_global(const BucketizeArgs& a, uint32_t i) {
  uint64_t co = a.code_object_ids[i];
  int slot = find_slot(a.slot_ids, a.n_slots, co);
  if (slot < 0) {
    atomicAdd(&a.overflow[0], 1u);
    return;
  }
  uint32_t first = a.slot_offsets[slot];
  uint32_t last = a.slot_offsets[slot + 1];
  uint32_t bucket = first + static_cast<uint32_t>(
      a.offsets[i] >> a.bucket_shift);
  if (bucket >= last) {
    atomicAdd(&a.overflow[1], 1u);
    return;
  }
  atomicAdd(&a.histogram[bucket], 1u);
  if (a.lane_histogram != nullptr && a.exec_masks != nullptr) {
    uint64_t lanes = __popcll(a.exec_masks[i]);
    atomicAdd(reinterpret_cast<unsigned long long*>(&a.lane_histogram[bucket]),
              static_cast<unsigned long long>(lanes));
  }
}

// LDS-staged variant: whole histogram fits in LDS.
__global__ __launch_bounds__(kThreads) void bucketize_lds(BucketizeArgs a) {
  extern __shared__ uint32_t lds_hist[];
  for (uint32_t b = threadIdx.x; b < a.total_buckets; b += kThreads)
    lds_hist[b] = 0;
  __syncthreads();

  for (uint32_t i = blockIdx.x * kThreads + threadIdx.x; i < a.n;
       i += gridDim.x * kThreads) {
    uint64_t co = a.code_object_ids[i];
    int slot = find_slot(a.slot_ids, a.n_slots, co);
    if (slot < 0) {
      atomicAdd(&a.overflow[0], 1u);
      continue;
    }
    uint32_t first = a.slot_offsets[slot];
    uint32_t last = a.slot_offsets[slot + 1];
    uint32_t bucket =
        first + static_cast<uint32_t>(a.offsets[i] >> a.bucket_shift);
    if (bucket >= last) {
      atomicAdd(&a.overflow[1], 1u);
      continue;
    }
    atomicAdd(&lds_hist[bucket], 1u);  // compiles to ds_add on LDS
    if (a.lane_histogram != nullptr && a.exec_masks != nullptr) {
      atomicAdd(
          reinterpret_cast<unsigned long long*>(&a.lane_histogram[bucket]),
          static_cast<unsigned long long>(__popcll(a.exec_masks[i])));
    }
  }
  __syncthreads();

  // One device-scope atomic per non-zero bucket per workgroup.
  for (uint32_t b = threadIdx.x; b < a.total_buckets; b += kThreads) {
    uint32_t v = lds_hist[b];
    if (v != 0) atomicAdd(&a.histogram[b], v);
  }
}

// Fallback: histogram too large for LDS — direct global atomics.
__global__ __launch_bounds__(kThreads) void bucketize_global(BucketizeArgs a) {
  for (uint32_t i = blockIdx.x * kThreads + threadIdx.x; i < a.n;
       i += gridDim.x * kThreads) {
    accumulate_global(a, i);
  }
}

void launch_bucketize(const BucketizeArgs& args, hipStream_t stream) {
  uint32_t blocks = (args.n + kThreads - 1) / kThreads;
  if (blocks > 2048) blocks = 2048;  // grid-stride the rest (G11)
  if (blocks == 0) return;
  if (args.total_buckets <= kMaxLdsBuckets) {
    size_t lds_bytes = static_cast<size_t>(args.total_buckets) * 4;
    hipLaunchKernelGGL(bucketize_lds, dim3(blocks), dim3(kThreads), lds_bytes,
                       stream, args);
  } else {
    hipLaunchKernelGGL(bucketize_global, dim3(blocks), dim3(kThreads), 0,
                       stream, args);
  }
}

}  // namespace parca_gpu
