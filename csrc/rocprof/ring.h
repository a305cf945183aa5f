// Shared-memory event ring between the rocprofiler interception tool
// (running inside target HIP processes) and the agent.
//
// This is the MI355X-native analog of the reference's `cupti_events` BPF
// ringbuf (reference: parcagpu/parcagpu.go:69-82, event tag dispatch at
// 149-213): one file per process under PARCA_GPU_SHM_DIR (default
// /dev/shm), a single logical producer (the tool serializes its
// rocprofiler callback threads with a process-local mutex), a single
// consumer (the agent). Records are tagged with a u32 event type at
// offset 0, sizes are 8-byte aligned, and producer-side drops are counted
// in the header (the `rocm.errors.ringbuf_full` metric, mirroring
// `cuda.errors.ringbuf_full`, metrics/all.go:1334-1339).

#pragma once

#include <atomic>
#include <cstdint>
#include <cstring>

namespace parca {

constexpr uint32_t kRingMagic = 0xA3D9C4E1;
constexpr uint32_t kRingVersion = 1;

// Event type tags (u32 discriminator at record offset 0).
enum EventType : uint32_t {
  kEvKernelDispatch = 1,
  kEvCodeObjectLoad = 2,
  kEvCodeObjectUnload = 3,
  kEvKernelSymbol = 4,
  kEvPCSampleBatch = 5,
  kEvGpuConfig = 6,
  kEvError = 7,
  kEvLaunchStack = 8,
};

struct RingHeader {
  uint32_t magic;
  uint32_t version;
  uint64_t capacity;  // bytes in the data area (power of two)
  alignas(64) std::atomic<uint64_t> head;  // producer cursor (monotonic)
  alignas(64) std::atomic<uint64_t> tail;  // consumer cursor (monotonic)
  alignas(64) std::atomic<uint64_t> dropped;      // records dropped (full)
  std::atomic<uint64_t> written;                  // records written
  uint32_t pid;
  uint32_t reserved;
  uint8_t pad[40];
};
static_assert(sizeof(RingHeader) == 64 * 4, "ring header layout");

struct RecordHeader {
  uint32_t type;
  uint32_t size;  // total bytes incl. this header, 8-byte aligned
};

// -- event payloads (fixed little-endian C layouts; Python decodes with
//    struct — keep in sync with parca_agent_amd/gpu/events.py) -----------

struct KernelDispatchEvent {
  uint64_t correlation_id;   // internal correlation id
  uint64_t dispatch_id;
  uint64_t kernel_id;        // rocprofiler kernel symbol id
  uint64_t start_ns;         // agent-domain (system) timestamp
  uint64_t end_ns;
  uint64_t tid;              // launching thread
  uint32_t gpu_index;        // logical GPU index (agent node order)
  uint32_t pid;
  uint32_t grid[3];
  uint32_t workgroup[3];
  uint32_t private_segment_size;
  uint32_t group_segment_size;
};

struct CodeObjectLoadEvent {
  uint64_t code_object_id;
  uint64_t load_base;
  uint64_t load_size;
  int64_t load_delta;
  uint64_t memory_base;  // nonzero if storage_type == memory
  uint64_t memory_size;
  uint32_t storage_type;
  uint32_t uri_len;  // uri bytes follow the struct
};

struct CodeObjectUnloadEvent {
  uint64_t code_object_id;
};

struct KernelSymbolEvent {
  uint64_t kernel_id;
  uint64_t code_object_id;
  uint64_t kernel_object;
  uint32_t name_len;  // demangle-ready name bytes follow
  uint32_t reserved;
};

// One PC sample within a kEvPCSampleBatch record.
struct PCSample {
  uint64_t code_object_id;
  uint64_t code_object_offset;
  uint64_t timestamp;
  uint64_t exec_mask;
  uint64_t dispatch_id;
  uint64_t correlation_id;
  uint64_t hw_id;        // raw rocprofiler_pc_sampling_hw_id_v0_t bits
  uint32_t wave_in_group;
  uint32_t flags;        // bit0: stochastic, bit1: valid-issue info
};

struct PCSampleBatchHeader {
  uint32_t gpu_index;
  uint32_t count;  // PCSample entries following
};

struct GpuConfigEvent {
  uint32_t gpu_index;
  uint32_t method;    // rocprofiler_pc_sampling_method_t
  uint32_t unit;      // rocprofiler_pc_sampling_unit_t
  uint32_t reserved;
  uint64_t interval;  // configured sampling interval (unit-typed)
  // Nanoseconds per sample as computed by the tool: for TIME units the
  // interval itself, for CYCLES interval/clock-rate. The GpuConfig ->
  // NsPerSample semantics of the reference (parca_reporter.go:89-102).
  double ns_per_sample;
};

struct ErrorEvent {
  uint32_t code;
  uint32_t msg_len;  // message bytes follow
};

// Host callstack captured at kernel-launch enqueue; replaces the
// reference's cudaLaunchKernel uprobe stack capture (SURVEY.md §3.3 flow
// A). Frames are raw user-space IPs of the launching thread, leaf-first;
// the agent resolves them against /proc/<pid>/maps.
struct LaunchStackEvent {
  uint64_t correlation_id;
  uint64_t tid;
  uint32_t pid;
  uint32_t n_frames;  // u64 ips follow
};

// -- producer (tool side) -------------------------------------------------

inline uint64_t ring_align(uint64_t n) { return (n + 7) & ~uint64_t(7); }

class RingProducer {
 public:
  // mem points at a mapping of header+data. Initializes the header when
  // magic is unset (fresh file).
  RingProducer(void* mem, uint64_t capacity, uint32_t pid) {
    hdr_ = static_cast<RingHeader*>(mem);
    data_ = reinterpret_cast<uint8_t*>(mem) + sizeof(RingHeader);
    if (hdr_->magic != kRingMagic) {
      memset(hdr_, 0, sizeof(RingHeader));
      hdr_->capacity = capacity;
      hdr_->pid = pid;
      hdr_->version = kRingVersion;
      std::atomic_thread_fence(std::memory_order_release);
      hdr_->magic = kRingMagic;
    }
    cap_ = hdr_->capacity;
  }

  // Reserve-copy-publish. Returns false (and counts a drop) when full.
  bool write(uint32_t type, const void* payload_a, size_t size_a,
             const void* payload_b = nullptr, size_t size_b = 0) {
    uint64_t need = ring_align(sizeof(RecordHeader) + size_a + size_b);
    uint64_t head = hdr_->head.load(std::memory_order_relaxed);
    uint64_t tail = hdr_->tail.load(std::memory_order_acquire);
    if (head - tail + need > cap_) {
      hdr_->dropped.fetch_add(1, std::memory_order_relaxed);
      return false;
    }
    uint64_t off = head & (cap_ - 1);
    RecordHeader rh{type, static_cast<uint32_t>(need)};
    write_bytes(off, &rh, sizeof(rh));
    write_bytes((off + sizeof(rh)) & (cap_ - 1), payload_a, size_a);
    if (payload_b != nullptr && size_b > 0)
      write_bytes((off + sizeof(rh) + size_a) & (cap_ - 1), payload_b, size_b);
    hdr_->head.store(head + need, std::memory_order_release);
    hdr_->written.fetch_add(1, std::memory_order_relaxed);
    return true;
  }

  RingHeader* header() { return hdr_; }

 private:
  void write_bytes(uint64_t off, const void* src, size_t n) {
    const uint8_t* s = static_cast<const uint8_t*>(src);
    uint64_t first = cap_ - off;
    if (n <= first) {
      memcpy(data_ + off, s, n);
    } else {
      memcpy(data_ + off, s, first);
      memcpy(data_, s + first, n - first);
    }
  }

  RingHeader* hdr_;
  uint8_t* data_;
  uint64_t cap_;
};

}  // namespace parca
