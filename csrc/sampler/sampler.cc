// perf_event_open CPU sampler — the MI355X-native replacement for the
// reference's eBPF perf-event sampler (reference: fork tracer driven from
// main.go:496-607; behavior spec in SURVEY.md §2.9/§3.2).
//
// Design: one perf event per online CPU at a prime sampling frequency
// (19 Hz default, flags/flags.go:44-51 rationale), each with an mmap ring.
// A dedicated drain thread epoll-waits on all rings and decodes
// PERF_RECORD_SAMPLE (kernel+user callchain via frame pointers),
// PERF_RECORD_{COMM,MMAP2,EXIT,FORK,LOST}. Decoded events land in a
// mutex-guarded batch the Python side swaps out at its poll interval.
// Lost-record accounting feeds the dropped-sample-rate metric that
// BASELINE.json names.
//
// Optional DWARF mode additionally captures PERF_SAMPLE_REGS_USER +
// PERF_SAMPLE_STACK_USER so the agent-side .eh_frame unwinder can walk
// stacks of frame-pointer-less binaries (reference analog: the fork's
// eBPF stack-delta unwinder, SURVEY.md §7 stage 4).

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <atomic>
#include <cerrno>
#include <cstdint>
#include <malloc.h>
#include <cstring>
#include <memory>
#include <mutex>
#include <stdexcept>
#include <string>
#include <thread>
#include <vector>

#include <fcntl.h>
#include <linux/perf_event.h>
#include <poll.h>
#include <sys/epoll.h>
#include <sys/ioctl.h>
#include <sys/mman.h>
#include <sys/syscall.h>
#include <unistd.h>

#include "sampler/ehframe.cc"

namespace py = pybind11;

namespace {

// Callchain context markers (linux/perf_event.h): PERF_CONTEXT_HV=-32,
// KERNEL=-128, USER=-512, GUEST=-2048, GUEST_KERNEL=-2176, GUEST_USER=-2560,
// MAX=-4095. Anything >= (u64)-4095 is a marker, not an IP.
constexpr uint64_t kContextMin = static_cast<uint64_t>(-4095LL);
constexpr uint64_t kContextKernel = static_cast<uint64_t>(-128LL);
constexpr uint64_t kContextHV = static_cast<uint64_t>(-32LL);
constexpr uint64_t kContextGuestKernel = static_cast<uint64_t>(-2176LL);

long perf_event_open(struct perf_event_attr* attr, pid_t pid, int cpu,
                     int group_fd, unsigned long flags) {
  return syscall(SYS_perf_event_open, attr, pid, cpu, group_fd, flags);
}

struct SampleEvent {
  uint32_t pid = 0;
  uint32_t tid = 0;
  uint32_t cpu = 0;
  uint64_t time_ns = 0;
  std::vector<uint64_t> kernel_ips;
  std::vector<uint64_t> user_ips;
  // DWARF mode: user register file (ABI order per PERF_SAMPLE_REGS_USER
  // mask) + the copied stack bytes. Stacks live in ONE shared per-batch
  // arena (offset/len view), not a per-sample string: thousands of 8-16
  // KiB mallocs per second fragmented glibc arenas into hundreds of MB
  // of retained RSS on many-core nodes, while one arena block per batch
  // is mmap-sized and returns to the kernel on free. Python materializes
  // bytes lazily and only for the samples it actually unwinds.
  std::vector<uint64_t> regs;
  std::shared_ptr<std::vector<uint8_t>> stack_arena;
  uint64_t stack_off = 0;
  uint64_t stack_len = 0;
  uint64_t stack_dyn_size = 0;
};

struct ProcEvent {
  // kind: 0=comm, 1=mmap2, 2=exit, 3=fork
  int kind = 0;
  uint32_t pid = 0;
  uint32_t tid = 0;
  uint32_t ppid = 0;
  std::string comm;
  uint64_t addr = 0;
  uint64_t len = 0;
  uint64_t pgoff = 0;
  uint32_t prot = 0;
  std::string filename;
  uint64_t time_ns = 0;
};

struct Batch {
  std::vector<SampleEvent> samples;
  std::vector<ProcEvent> proc_events;
  // Shared stack-dump arena for this batch's samples.
  std::shared_ptr<std::vector<uint8_t>> arena;
};

class RingBuffer {
 public:
  RingBuffer() = default;
  RingBuffer(const RingBuffer&) = delete;
  RingBuffer& operator=(const RingBuffer&) = delete;

  bool init(int fd, size_t data_pages) {
    size_t page = sysconf(_SC_PAGESIZE);
    size_ = (1 + data_pages) * page;
    void* m = mmap(nullptr, size_, PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
    if (m == MAP_FAILED) return false;
    meta_ = static_cast<perf_event_mmap_page*>(m);
    data_ = static_cast<uint8_t*>(m) + page;
    data_size_ = data_pages * page;
    fd_ = fd;
    return true;
  }

  ~RingBuffer() {
    if (meta_) munmap(meta_, size_);
  }

  // Visit every pending record; cb(type, payload, size).
  template <typename Fn>
  void drain(Fn&& cb) {
    uint64_t head = __atomic_load_n(&meta_->data_head, __ATOMIC_ACQUIRE);
    uint64_t tail = meta_->data_tail;
    while (tail < head) {
      auto* hdr = reinterpret_cast<perf_event_header*>(
          data_ + (tail & (data_size_ - 1)));
      // Records can wrap the ring; copy into scratch when they do.
      size_t off = tail & (data_size_ - 1);
      const uint8_t* rec;
      if (off + hdr->size > data_size_) {
        scratch_.resize(hdr->size);
        size_t first = data_size_ - off;
        memcpy(scratch_.data(), data_ + off, first);
        memcpy(scratch_.data() + first, data_, hdr->size - first);
        rec = scratch_.data();
        hdr = reinterpret_cast<perf_event_header*>(scratch_.data());
      } else {
        rec = data_ + off;
      }
      cb(hdr->type, rec, hdr->size);
      tail += hdr->size;
    }
    __atomic_store_n(&meta_->data_tail, tail, __ATOMIC_RELEASE);
  }

  int fd() const { return fd_; }

 private:
  perf_event_mmap_page* meta_ = nullptr;
  uint8_t* data_ = nullptr;
  size_t data_size_ = 0;
  size_t size_ = 0;
  int fd_ = -1;
  std::vector<uint8_t> scratch_;
};

// Register mask for x86-64 user regs (PERF_SAMPLE_REGS_USER). Bit order
// follows arch/x86/include/uapi/asm/perf_regs.h: AX,BX,CX,DX,SI,DI,BP,SP,
// IP,FLAGS (0-9), segment regs (10-15, REJECTED by the kernel for
// sampling), R8-R15 (16-23). Registers are delivered in ascending bit
// order; Python indexes them via REGS_ORDER exported below.
constexpr uint64_t kX86RegsMask = 0x3FFULL | (0xFFULL << 16);

class PerfSampler {
 public:
  PerfSampler(int freq, bool dwarf_stacks, uint32_t stack_dump_size,
              int ring_pages, bool track_mmaps, int target_pid)
      : freq_(freq),
        dwarf_(dwarf_stacks),
        stack_dump_(stack_dump_size),
        ring_pages_(ring_pages),
        track_mmaps_(track_mmaps),
        target_pid_(target_pid) {
    if (ring_pages_ <= 0 || (ring_pages_ & (ring_pages_ - 1)) != 0)
      throw std::invalid_argument("ring_pages must be a power of two");
  }

  ~PerfSampler() { stop(); }

  void start() {
    if (running_.exchange(true)) return;
    int ncpu = sysconf(_SC_NPROCESSORS_ONLN);
    epfd_ = epoll_create1(EPOLL_CLOEXEC);
    if (epfd_ < 0) throw std::runtime_error("epoll_create1 failed");

    rings_.clear();
    fds_.clear();
    rings_.reserve(ncpu);

    for (int cpu = 0; cpu < ncpu; ++cpu) {
      struct perf_event_attr attr;
      memset(&attr, 0, sizeof(attr));
      attr.size = sizeof(attr);
      attr.type = PERF_TYPE_SOFTWARE;
      attr.config = PERF_COUNT_SW_CPU_CLOCK;
      attr.freq = 1;
      attr.sample_freq = freq_;
      attr.sample_type = PERF_SAMPLE_TID | PERF_SAMPLE_TIME |
                         PERF_SAMPLE_CALLCHAIN | PERF_SAMPLE_CPU;
      if (dwarf_) {
        attr.sample_type |= PERF_SAMPLE_REGS_USER | PERF_SAMPLE_STACK_USER;
        attr.sample_regs_user = kX86RegsMask;
        attr.sample_stack_user = stack_dump_;
      }
      attr.disabled = 1;
      attr.inherit = 0;
      attr.exclude_hv = 1;
      attr.sample_max_stack = 127;
      attr.wakeup_events = 1;
      if (track_mmaps_) {
        // Process-lifecycle events: with pid=-1/cpu=N each CPU's ring only
        // sees events that happened on that CPU, so they must be enabled on
        // every ring; Python dedupes (comm/mmap2 keyed by pid+payload).
        attr.comm = 1;
        attr.task = 1;
        attr.mmap = 1;
        attr.mmap2 = 1;
        attr.mmap_data = 0;
      }

      int fd = perf_event_open(&attr, target_pid_, cpu, -1, PERF_FLAG_FD_CLOEXEC);
      if (fd < 0) {
        if (errno == EACCES || errno == EPERM)
          throw std::runtime_error(
              "perf_event_open: permission denied (need root/CAP_PERFMON or "
              "kernel.perf_event_paranoid <= 1)");
        throw std::runtime_error(std::string("perf_event_open cpu ") +
                                 std::to_string(cpu) + ": " + strerror(errno));
      }
      auto ring = std::make_unique<RingBuffer>();
      if (!ring->init(fd, ring_pages_)) {
        close(fd);
        throw std::runtime_error("mmap of perf ring failed");
      }
      struct epoll_event ev;
      ev.events = EPOLLIN;
      ev.data.u32 = static_cast<uint32_t>(rings_.size());
      epoll_ctl(epfd_, EPOLL_CTL_ADD, fd, &ev);
      ioctl(fd, PERF_EVENT_IOC_ENABLE, 0);
      fds_.push_back(fd);
      rings_.push_back(std::move(ring));
    }

    drain_thread_ = std::thread([this] { drain_loop(); });
  }

  void stop() {
    if (!running_.exchange(false)) return;
    if (drain_thread_.joinable()) drain_thread_.join();
    for (int fd : fds_) {
      ioctl(fd, PERF_EVENT_IOC_DISABLE, 0);
      close(fd);
    }
    fds_.clear();
    rings_.clear();
    if (epfd_ >= 0) close(epfd_);
    epfd_ = -1;
  }

  // Swap out the accumulated batch. Returns (samples, proc_events).
  Batch take() {
    Batch out;
    {
      std::lock_guard<std::mutex> lk(mu_);
      out.samples.swap(batch_.samples);
      out.proc_events.swap(batch_.proc_events);
      out.arena = std::move(batch_.arena);
      batch_.arena.reset();
    }
    return out;
  }

  uint64_t lost_count() const { return lost_.load(); }
  uint64_t sample_count() const { return nsamples_.load(); }
  uint64_t unknown_count() const { return unknown_.load(); }
  int n_cpus() const { return static_cast<int>(rings_.size()); }

 private:
  void drain_loop() {
    std::vector<struct epoll_event> events(rings_.size());
    while (running_.load(std::memory_order_relaxed)) {
      int n = epoll_wait(epfd_, events.data(), events.size(), 100);
      if (n < 0) {
        if (errno == EINTR) continue;
        break;
      }
      Batch local;
      for (auto& ring : rings_) {
        ring->drain([&](uint32_t type, const uint8_t* rec, size_t size) {
          decode(type, rec, size, local);
        });
      }
      if (!local.samples.empty() || !local.proc_events.empty()) {
        std::lock_guard<std::mutex> lk(mu_);
        auto& b = batch_;
        b.samples.insert(b.samples.end(),
                         std::make_move_iterator(local.samples.begin()),
                         std::make_move_iterator(local.samples.end()));
        b.proc_events.insert(b.proc_events.end(),
                             std::make_move_iterator(local.proc_events.begin()),
                             std::make_move_iterator(local.proc_events.end()));
        // Backstop: never let an idle Python side grow the batch unbounded.
        if (b.samples.size() > 1 << 20) {
          lost_ += b.samples.size();
          b.samples.clear();
        }
      }
    }
  }

  void decode(uint32_t type, const uint8_t* rec, size_t size, Batch& out) {
    const uint8_t* p = rec + sizeof(perf_event_header);
    const uint8_t* end = rec + size;
    auto rd64 = [&]() {
      uint64_t v;
      memcpy(&v, p, 8);
      p += 8;
      return v;
    };
    auto rd32 = [&]() {
      uint32_t v;
      memcpy(&v, p, 4);
      p += 4;
      return v;
    };

    switch (type) {
      case PERF_RECORD_SAMPLE: {
        SampleEvent ev;
        ev.pid = rd32();
        ev.tid = rd32();
        ev.time_ns = rd64();
        ev.cpu = rd32();
        rd32();  // res
        uint64_t nr = rd64();
        if (nr > 512 || p + nr * 8 > end) {
          unknown_++;
          return;
        }
        bool in_kernel = true;  // callchain starts with a context marker
        for (uint64_t i = 0; i < nr; ++i) {
          uint64_t ip = rd64();
          if (ip >= kContextMin) {
            in_kernel = (ip == kContextKernel || ip == kContextHV ||
                         ip == kContextGuestKernel);
            continue;
          }
          (in_kernel ? ev.kernel_ips : ev.user_ips).push_back(ip);
        }
        if (dwarf_ && p < end) {
          uint64_t abi = rd64();
          if (abi != 0 /* PERF_SAMPLE_REGS_ABI_NONE */) {
            int nregs = __builtin_popcountll(kX86RegsMask);
            ev.regs.resize(nregs);
            for (int i = 0; i < nregs; ++i) ev.regs[i] = rd64();
          }
          if (p < end) {
            uint64_t stack_size = rd64();
            if (stack_size > 0 && p + stack_size <= end) {
              const uint8_t* stack_start = p;
              p += stack_size;
              uint64_t dyn = rd64();
              uint64_t keep = std::min<uint64_t>(dyn, stack_size);
              if (keep > 0) {
                if (!out.arena)
                  out.arena = std::make_shared<std::vector<uint8_t>>();
                ev.stack_arena = out.arena;
                ev.stack_off = out.arena->size();
                ev.stack_len = keep;
                out.arena->insert(out.arena->end(), stack_start,
                                  stack_start + keep);
              }
              ev.stack_dyn_size = dyn;
            }
          }
        }
        nsamples_++;
        out.samples.push_back(std::move(ev));
        break;
      }
      case PERF_RECORD_LOST: {
        rd64();  // id
        lost_ += rd64();
        break;
      }
      case PERF_RECORD_COMM: {
        ProcEvent ev;
        ev.kind = 0;
        ev.pid = rd32();
        ev.tid = rd32();
        ev.comm.assign(reinterpret_cast<const char*>(p));
        out.proc_events.push_back(std::move(ev));
        break;
      }
      case PERF_RECORD_EXIT:
      case PERF_RECORD_FORK: {
        ProcEvent ev;
        ev.kind = (type == PERF_RECORD_EXIT) ? 2 : 3;
        ev.pid = rd32();
        ev.ppid = rd32();
        ev.tid = rd32();
        rd32();  // ptid
        ev.time_ns = rd64();
        out.proc_events.push_back(std::move(ev));
        break;
      }
      case PERF_RECORD_MMAP2: {
        ProcEvent ev;
        ev.kind = 1;
        ev.pid = rd32();
        ev.tid = rd32();
        ev.addr = rd64();
        ev.len = rd64();
        ev.pgoff = rd64();
        rd64();  // maj/min or build-id part
        rd64();  // ino
        rd64();  // ino_generation
        ev.prot = rd32();
        rd32();  // flags
        ev.filename.assign(reinterpret_cast<const char*>(p));
        // Only executable mappings matter for unwinding/symbolization.
        if (ev.prot & 0x4 /* PROT_EXEC */) out.proc_events.push_back(std::move(ev));
        break;
      }
      default:
        unknown_++;
    }
  }

  int freq_;
  bool dwarf_;
  uint32_t stack_dump_;
  int ring_pages_;
  bool track_mmaps_;
  int target_pid_;

  std::vector<std::unique_ptr<RingBuffer>> rings_;
  std::vector<int> fds_;
  int epfd_ = -1;
  std::atomic<bool> running_{false};
  std::thread drain_thread_;

  std::mutex mu_;
  Batch batch_;
  std::atomic<uint64_t> lost_{0}, nsamples_{0}, unknown_{0};
};

// -- off-CPU sampling -----------------------------------------------------
//
// Native replacement for the fork's eBPF off-CPU profiling
// (tracer.StartOffCPUProfiling, main.go:534-539): per-CPU
// PERF_COUNT_SW_CONTEXT_SWITCHES events sample every Nth switch-out with
// the blocking callchain (N = 1/threshold, the probabilistic knob), and
// attr.context_switch=1 adds lightweight PERF_RECORD_SWITCH records for
// every switch so blocked DURATIONS can be paired per-tid in Python.

struct SwitchEvent {
  uint32_t pid = 0;
  uint32_t tid = 0;
  uint32_t cpu = 0;
  bool is_out = false;
  bool preempt = false;
  uint64_t time_ns = 0;
};

class OffCpuSampler {
 public:
  OffCpuSampler(uint64_t sample_period, int ring_pages)
      : period_(sample_period), ring_pages_(ring_pages) {}

  ~OffCpuSampler() { stop(); }

  void start() {
    if (running_.exchange(true)) return;
    int ncpu = sysconf(_SC_NPROCESSORS_ONLN);
    epfd_ = epoll_create1(EPOLL_CLOEXEC);
    for (int cpu = 0; cpu < ncpu; ++cpu) {
      struct perf_event_attr attr;
      memset(&attr, 0, sizeof(attr));
      attr.size = sizeof(attr);
      attr.type = PERF_TYPE_SOFTWARE;
      attr.config = PERF_COUNT_SW_CONTEXT_SWITCHES;
      attr.sample_period = period_;
      attr.sample_type = PERF_SAMPLE_TID | PERF_SAMPLE_TIME |
                         PERF_SAMPLE_CALLCHAIN | PERF_SAMPLE_CPU;
      attr.context_switch = 1;
      attr.sample_id_all = 1;  // tid/time on SWITCH records
      attr.disabled = 1;
      attr.exclude_hv = 1;
      attr.sample_max_stack = 127;
      attr.wakeup_events = 64;
      int fd = perf_event_open(&attr, -1, cpu, -1, PERF_FLAG_FD_CLOEXEC);
      if (fd < 0) {
        if (cpu == 0)
          throw std::runtime_error(std::string("off-cpu event: ") +
                                   strerror(errno));
        continue;
      }
      auto ring = std::make_unique<RingBuffer>();
      if (!ring->init(fd, ring_pages_)) {
        close(fd);
        continue;
      }
      struct epoll_event ev;
      ev.events = EPOLLIN;
      ev.data.u32 = static_cast<uint32_t>(rings_.size());
      epoll_ctl(epfd_, EPOLL_CTL_ADD, fd, &ev);
      ioctl(fd, PERF_EVENT_IOC_ENABLE, 0);
      fds_.push_back(fd);
      rings_.push_back(std::move(ring));
    }
    drain_thread_ = std::thread([this] { drain_loop(); });
  }

  void stop() {
    if (!running_.exchange(false)) return;
    if (drain_thread_.joinable()) drain_thread_.join();
    for (int fd : fds_) {
      ioctl(fd, PERF_EVENT_IOC_DISABLE, 0);
      close(fd);
    }
    fds_.clear();
    rings_.clear();
    if (epfd_ >= 0) close(epfd_);
    epfd_ = -1;
  }

  std::pair<std::vector<SampleEvent>, std::vector<SwitchEvent>> take() {
    std::lock_guard<std::mutex> lk(mu_);
    auto out = std::make_pair(std::move(samples_), std::move(switches_));
    samples_.clear();
    switches_.clear();
    return out;
  }

  uint64_t lost() const { return lost_.load(); }

 private:
  void drain_loop() {
    std::vector<struct epoll_event> events(rings_.size() + 1);
    while (running_.load(std::memory_order_relaxed)) {
      int n = epoll_wait(epfd_, events.data(), events.size(), 100);
      if (n < 0 && errno != EINTR) break;
      std::vector<SampleEvent> ls;
      std::vector<SwitchEvent> lw;
      for (auto& ring : rings_) {
        ring->drain(
            [&](uint32_t type, const uint8_t* rec, size_t size) {
              decode(type, rec, size, ls, lw);
            });
      }
      if (!ls.empty() || !lw.empty()) {
        std::lock_guard<std::mutex> lk(mu_);
        samples_.insert(samples_.end(),
                        std::make_move_iterator(ls.begin()),
                        std::make_move_iterator(ls.end()));
        switches_.insert(switches_.end(), lw.begin(), lw.end());
        if (switches_.size() > 1 << 21) switches_.clear();
        if (samples_.size() > 1 << 18) samples_.clear();
      }
    }
  }

  void decode(uint32_t type, const uint8_t* rec, size_t size,
              std::vector<SampleEvent>& ls, std::vector<SwitchEvent>& lw) {
    auto* hdr = reinterpret_cast<const perf_event_header*>(rec);
    const uint8_t* p = rec + sizeof(perf_event_header);
    const uint8_t* end = rec + size;
    auto rd64 = [&]() {
      uint64_t v;
      memcpy(&v, p, 8);
      p += 8;
      return v;
    };
    auto rd32 = [&]() {
      uint32_t v;
      memcpy(&v, p, 4);
      p += 4;
      return v;
    };
    if (type == PERF_RECORD_SAMPLE) {
      SampleEvent ev;
      ev.pid = rd32();
      ev.tid = rd32();
      ev.time_ns = rd64();
      ev.cpu = rd32();
      rd32();
      uint64_t nr = rd64();
      if (nr > 512 || p + nr * 8 > end) return;
      bool in_kernel = true;
      for (uint64_t i = 0; i < nr; ++i) {
        uint64_t ip = rd64();
        if (ip >= kContextMin) {
          in_kernel = (ip == kContextKernel || ip == kContextHV ||
                       ip == kContextGuestKernel);
          continue;
        }
        (in_kernel ? ev.kernel_ips : ev.user_ips).push_back(ip);
      }
      ls.push_back(std::move(ev));
    } else if (type == 14 /* PERF_RECORD_SWITCH */ ||
               type == 15 /* PERF_RECORD_SWITCH_CPU_WIDE */) {
      // sample_id trailer (sample_id_all): TID(8) TIME(8) CPU(8) at end
      // for our sample_type (TID|TIME|CALLCHAIN|CPU -> trailer has
      // TID, TIME, CPU, then IDENTIFIER absent).
      SwitchEvent ev;
      ev.is_out = (hdr->misc & 0x2000 /* PERF_RECORD_MISC_SWITCH_OUT */);
      ev.preempt =
          (hdr->misc & 0x4000 /* PERF_RECORD_MISC_SWITCH_OUT_PREEMPT */);
      const uint8_t* body = p;
      if (type == 15) body += 8;  // next_prev pid/tid, unused
      // trailer layout: {u32 pid,tid; u64 time; u32 cpu,res}
      if (body + 24 > end) return;
      memcpy(&ev.pid, body, 4);
      memcpy(&ev.tid, body + 4, 4);
      memcpy(&ev.time_ns, body + 8, 8);
      memcpy(&ev.cpu, body + 16, 4);
      lw.push_back(ev);
    } else if (type == PERF_RECORD_LOST) {
      p += 8;
      lost_ += rd64();
    }
  }

  uint64_t period_;
  int ring_pages_;
  std::vector<std::unique_ptr<RingBuffer>> rings_;
  std::vector<int> fds_;
  int epfd_ = -1;
  std::atomic<bool> running_{false};
  std::thread drain_thread_;
  std::mutex mu_;
  std::vector<SampleEvent> samples_;
  std::vector<SwitchEvent> switches_;
  std::atomic<uint64_t> lost_{0};
};

// -- uprobes --------------------------------------------------------------
//
// Paired entry/return uprobes via the perf "uprobe" PMU — the native
// replacement for the reference's uprobe BPF program
// (probes/bpf/probe.bpf.c:85-154). One perf event per (probe, cpu);
// events are tagged by ring index, pairing/depth-tracking happens in
// Python (probes/service.py) using the per-event timestamps.

struct ProbeFire {
  uint32_t probe_id = 0;
  uint32_t pid = 0;
  uint32_t tid = 0;
  uint32_t cpu = 0;
  uint64_t time_ns = 0;
  bool is_return = false;
};

int read_uprobe_pmu_type() {
  FILE* f = fopen("/sys/bus/event_source/devices/uprobe/type", "r");
  if (!f) return -1;
  int t = -1;
  if (fscanf(f, "%d", &t) != 1) t = -1;
  fclose(f);
  return t;
}

int read_retprobe_bit() {
  // format/retprobe is "config:N"
  FILE* f = fopen("/sys/bus/event_source/devices/uprobe/format/retprobe",
                  "r");
  if (!f) return 0;
  int bit = 0;
  if (fscanf(f, "config:%d", &bit) != 1) bit = 0;
  fclose(f);
  return bit;
}

class UprobeGroup {
 public:
  struct Spec {
    std::string path;
    uint64_t offset;
    bool retprobe;
    uint32_t probe_id;
    uint64_t max_rate;  // fires/sec per probe pair; 0 = unlimited
  };

  explicit UprobeGroup(
      const std::vector<std::tuple<std::string, uint64_t, bool, uint32_t,
                                   uint64_t>>& specs) {
    for (auto& [path, offset, ret, id, rate] : specs)
      specs_.push_back(Spec{path, offset, ret, id, rate});
  }

  ~UprobeGroup() { stop(); }

  void start() {
    if (running_.exchange(true)) return;
    int pmu_type = read_uprobe_pmu_type();
    if (pmu_type < 0)
      throw std::runtime_error("uprobe PMU unavailable "
                               "(/sys/bus/event_source/devices/uprobe)");
    int ret_bit = read_retprobe_bit();
    int ncpu = sysconf(_SC_NPROCESSORS_ONLN);
    epfd_ = epoll_create1(EPOLL_CLOEXEC);

    for (size_t si = 0; si < specs_.size(); ++si) {
      const Spec& spec = specs_[si];
      for (int cpu = 0; cpu < ncpu; ++cpu) {
        struct perf_event_attr attr;
        memset(&attr, 0, sizeof(attr));
        attr.size = sizeof(attr);
        attr.type = pmu_type;
        attr.config = spec.retprobe ? (1ULL << ret_bit) : 0;
        attr.config1 = reinterpret_cast<uint64_t>(spec.path.c_str());
        attr.config2 = spec.offset;
        attr.sample_period = 1;
        attr.sample_type = PERF_SAMPLE_TID | PERF_SAMPLE_TIME |
                           PERF_SAMPLE_CPU;
        attr.disabled = 1;
        attr.exclude_hv = 1;
        attr.wakeup_events = 1;
        int fd = perf_event_open(&attr, -1, cpu, -1, PERF_FLAG_FD_CLOEXEC);
        if (fd < 0) {
          if (cpu == 0)
            throw std::runtime_error(
                std::string("uprobe attach failed for ") + spec.path + ":" +
                std::to_string(spec.offset) + ": " + strerror(errno));
          continue;
        }
        auto ring = std::make_unique<RingBuffer>();
        if (!ring->init(fd, 8)) {
          close(fd);
          continue;
        }
        struct epoll_event ev;
        ev.events = EPOLLIN;
        ev.data.u32 = static_cast<uint32_t>(rings_.size());
        epoll_ctl(epfd_, EPOLL_CTL_ADD, fd, &ev);
        ioctl(fd, PERF_EVENT_IOC_ENABLE, 0);
        fds_.push_back(fd);
        ring_spec_.push_back(static_cast<uint32_t>(si));
        rings_.push_back(std::move(ring));
        // Entry and return probes of one pair throttle TOGETHER (a
        // one-sided gap would corrupt the agent's scope pairing).
        uint32_t pair = spec.probe_id >> 1;
        pair_fds_[pair].push_back(fd);
        if (spec.max_rate)
          pair_rate_[pair] = std::max(pair_rate_[pair], spec.max_rate);
      }
    }
    drain_thread_ = std::thread([this] { drain_loop(); });
  }

  void stop() {
    if (!running_.exchange(false)) return;
    if (drain_thread_.joinable()) drain_thread_.join();
    for (int fd : fds_) {
      ioctl(fd, PERF_EVENT_IOC_DISABLE, 0);
      close(fd);
    }
    fds_.clear();
    rings_.clear();
    ring_spec_.clear();
    if (epfd_ >= 0) close(epfd_);
    epfd_ = -1;
  }

  std::vector<ProbeFire> take() {
    std::vector<ProbeFire> out;
    std::lock_guard<std::mutex> lk(mu_);
    out.swap(fires_);
    return out;
  }

  uint64_t lost() const { return lost_.load(); }
  int n_events() const { return static_cast<int>(fds_.size()); }
  uint64_t throttles() const { return throttles_.load(); }
  uint64_t throttled_fires() const { return throttled_fires_.load(); }

 private:
  struct Throttle {
    uint64_t win_start = 0;
    uint64_t count = 0;
    uint64_t disabled_until = 0;
    bool disabled = false;
  };

  static uint64_t mono_ns() {
    struct timespec ts;
    clock_gettime(CLOCK_MONOTONIC, &ts);
    return uint64_t(ts.tv_sec) * 1000000000ull + ts.tv_nsec;
  }

  void drain_loop() {
    std::vector<struct epoll_event> events(rings_.size() + 1);
    while (running_.load(std::memory_order_relaxed)) {
      int n = epoll_wait(epfd_, events.data(), events.size(), 100);
      if (n < 0 && errno != EINTR) break;
      uint64_t now = mono_ns();
      // Re-enable probes whose cooldown expired (flood protection,
      // a >100k calls/s symbol must not saturate
      // the drain — the reference filters in-kernel, we gate the perf
      // events themselves with PERF_EVENT_IOC_DISABLE).
      for (auto& [pair, th] : throttle_) {
        if (th.disabled && now >= th.disabled_until) {
          for (int fd : pair_fds_[pair]) ioctl(fd, PERF_EVENT_IOC_ENABLE, 0);
          th.disabled = false;
          th.win_start = now;
          th.count = 0;
        }
      }
      std::vector<ProbeFire> local;
      for (size_t ri = 0; ri < rings_.size(); ++ri) {
        const Spec& spec = specs_[ring_spec_[ri]];
        uint32_t pair = spec.probe_id >> 1;
        uint64_t rate = spec.max_rate ? pair_rate_[pair] : 0;
        rings_[ri]->drain([&](uint32_t type, const uint8_t* rec,
                              size_t size) {
          if (type == PERF_RECORD_LOST) {
            const uint8_t* p = rec + sizeof(perf_event_header) + 8;
            uint64_t lost;
            memcpy(&lost, p, 8);
            lost_ += lost;
            return;
          }
          if (type != PERF_RECORD_SAMPLE) return;
          if (rate) {
            Throttle& th = throttle_[pair];
            if (th.disabled) {
              // in-flight events drained after the disable
              throttled_fires_++;
              return;
            }
            if (now - th.win_start > 1000000000ull) {
              th.win_start = now;
              th.count = 0;
            }
            if (++th.count > rate) {
              for (int fd : pair_fds_[pair])
                ioctl(fd, PERF_EVENT_IOC_DISABLE, 0);
              th.disabled = true;
              th.disabled_until = now + 1000000000ull;
              throttles_++;
              throttled_fires_++;
              return;
            }
          }
          const uint8_t* p = rec + sizeof(perf_event_header);
          ProbeFire f;
          memcpy(&f.pid, p, 4);
          memcpy(&f.tid, p + 4, 4);
          memcpy(&f.time_ns, p + 8, 8);
          memcpy(&f.cpu, p + 16, 4);
          f.probe_id = spec.probe_id;
          f.is_return = spec.retprobe;
          local.push_back(f);
        });
      }
      if (!local.empty()) {
        std::lock_guard<std::mutex> lk(mu_);
        fires_.insert(fires_.end(), local.begin(), local.end());
        if (fires_.size() > 1 << 20) {
          lost_ += fires_.size();
          fires_.clear();
        }
      }
    }
  }

  std::vector<Spec> specs_;
  std::vector<std::unique_ptr<RingBuffer>> rings_;
  std::vector<uint32_t> ring_spec_;
  std::vector<int> fds_;
  int epfd_ = -1;
  std::atomic<bool> running_{false};
  std::thread drain_thread_;
  std::mutex mu_;
  std::vector<ProbeFire> fires_;
  std::unordered_map<uint32_t, std::vector<int>> pair_fds_;
  std::unordered_map<uint32_t, uint64_t> pair_rate_;
  std::unordered_map<uint32_t, Throttle> throttle_;
  std::atomic<uint64_t> throttles_{0};
  std::atomic<uint64_t> throttled_fires_{0};
  std::atomic<uint64_t> lost_{0};
};

}  // namespace

PYBIND11_MODULE(_sampler, m) {
  m.doc() = "perf_event_open CPU sampler (native core)";
  // Order in which sampled user registers appear in SampleEvent.regs.
  m.attr("REGS_ORDER") = py::make_tuple(
      "ax", "bx", "cx", "dx", "si", "di", "bp", "sp", "ip", "flags",
      "r8", "r9", "r10", "r11", "r12", "r13", "r14", "r15");

  py::class_<SampleEvent>(m, "SampleEvent")
      .def_readonly("pid", &SampleEvent::pid)
      .def_readonly("tid", &SampleEvent::tid)
      .def_readonly("cpu", &SampleEvent::cpu)
      .def_readonly("time_ns", &SampleEvent::time_ns)
      .def_readonly("kernel_ips", &SampleEvent::kernel_ips)
      .def_readonly("user_ips", &SampleEvent::user_ips)
      .def_readonly("regs", &SampleEvent::regs)
      .def_property_readonly(
          "stack",
          [](const SampleEvent& s) -> py::bytes {
            if (!s.stack_arena || s.stack_len == 0) return py::bytes("", 0);
            return py::bytes(
                reinterpret_cast<const char*>(s.stack_arena->data() +
                                              s.stack_off),
                static_cast<Py_ssize_t>(s.stack_len));
          })
      .def_readonly("stack_dyn_size", &SampleEvent::stack_dyn_size);

  py::class_<ProcEvent>(m, "ProcEvent")
      .def_readonly("kind", &ProcEvent::kind)
      .def_readonly("pid", &ProcEvent::pid)
      .def_readonly("tid", &ProcEvent::tid)
      .def_readonly("ppid", &ProcEvent::ppid)
      .def_readonly("comm", &ProcEvent::comm)
      .def_readonly("addr", &ProcEvent::addr)
      .def_readonly("len", &ProcEvent::len)
      .def_readonly("pgoff", &ProcEvent::pgoff)
      .def_readonly("prot", &ProcEvent::prot)
      .def_readonly("filename", &ProcEvent::filename)
      .def_readonly("time_ns", &ProcEvent::time_ns);

  py::class_<parca_unwind::Unwinder>(m, "Unwinder")
      .def(py::init<>())
      .def(
          "add_module_from_eh_frame",
          [](parca_unwind::Unwinder& u, py::bytes eh_frame,
             uint64_t section_vaddr, size_t max_rows) {
            char* buf;
            Py_ssize_t len;
            PyBytes_AsStringAndSize(eh_frame.ptr(), &buf, &len);
            int mid;
            {
              py::gil_scoped_release rel;
              std::vector<parca_unwind::Row> rows;
              parca_unwind::parse_eh_frame(
                  reinterpret_cast<const uint8_t*>(buf), len, section_vaddr,
                  rows, max_rows);
              mid = u.add_module(std::move(rows));
              // Return the build's transient arena to the kernel: the
              // parse vector peaks at 24 B/row and glibc would retain
              // hundreds of MB otherwise (libtorch/rocblas-scale
              // .eh_frame sections).
              malloc_trim(0);
            }
            return mid;
          },
          py::arg("eh_frame"), py::arg("section_vaddr"),
          py::arg("max_rows") = 20 * 1000 * 1000)
      .def("set_mappings",
           [](parca_unwind::Unwinder& u, uint32_t pid,
              const std::vector<std::tuple<uint64_t, uint64_t, uint64_t,
                                           int>>& maps) {
             std::vector<parca_unwind::Mapping> ms;
             ms.reserve(maps.size());
             for (auto& [start, end, bias, mod] : maps)
               ms.push_back(parca_unwind::Mapping{start, end, bias, mod});
             u.set_mappings(pid, std::move(ms));
           })
      .def("drop_process", &parca_unwind::Unwinder::drop_process)
      .def(
          "unwind",
          [](const parca_unwind::Unwinder& u, uint32_t pid, uint64_t ip,
             uint64_t sp, uint64_t bp, py::bytes stack, int max_frames) {
            char* buf;
            Py_ssize_t len;
            PyBytes_AsStringAndSize(stack.ptr(), &buf, &len);
            return u.unwind(pid, ip, sp, bp,
                            reinterpret_cast<const uint8_t*>(buf), len,
                            max_frames);
          },
          py::arg("pid"), py::arg("ip"), py::arg("sp"), py::arg("bp"),
          py::arg("stack"), py::arg("max_frames") = 128)
      .def_property_readonly("n_modules",
                             &parca_unwind::Unwinder::n_modules)
      .def("module_rows", &parca_unwind::Unwinder::module_rows)
      .def("module_bytes", &parca_unwind::Unwinder::module_bytes)
      .def_property_readonly("total_bytes",
                             &parca_unwind::Unwinder::total_bytes)
      .def_property_readonly("stacks_truncated",
                             &parca_unwind::Unwinder::stacks_truncated);

  py::class_<SwitchEvent>(m, "SwitchEvent")
      .def_readonly("pid", &SwitchEvent::pid)
      .def_readonly("tid", &SwitchEvent::tid)
      .def_readonly("cpu", &SwitchEvent::cpu)
      .def_readonly("is_out", &SwitchEvent::is_out)
      .def_readonly("preempt", &SwitchEvent::preempt)
      .def_readonly("time_ns", &SwitchEvent::time_ns);

  py::class_<OffCpuSampler>(m, "OffCpuSampler")
      .def(py::init<uint64_t, int>(), py::arg("sample_period") = 100,
           py::arg("ring_pages") = 64)
      .def("start", &OffCpuSampler::start,
           py::call_guard<py::gil_scoped_release>())
      .def("stop", &OffCpuSampler::stop,
           py::call_guard<py::gil_scoped_release>())
      .def("take",
           [](OffCpuSampler& s) {
             std::pair<std::vector<SampleEvent>, std::vector<SwitchEvent>>
                 out;
             {
               py::gil_scoped_release rel;
               out = s.take();
             }
             return py::make_tuple(std::move(out.first),
                                   std::move(out.second));
           })
      .def_property_readonly("lost", &OffCpuSampler::lost);

  py::class_<ProbeFire>(m, "ProbeFire")
      .def_readonly("probe_id", &ProbeFire::probe_id)
      .def_readonly("pid", &ProbeFire::pid)
      .def_readonly("tid", &ProbeFire::tid)
      .def_readonly("cpu", &ProbeFire::cpu)
      .def_readonly("time_ns", &ProbeFire::time_ns)
      .def_readonly("is_return", &ProbeFire::is_return);

  py::class_<UprobeGroup>(m, "UprobeGroup")
      .def(py::init<const std::vector<
               std::tuple<std::string, uint64_t, bool, uint32_t,
                          uint64_t>>&>(),
           py::arg("specs"))
      .def("start", &UprobeGroup::start,
           py::call_guard<py::gil_scoped_release>())
      .def("stop", &UprobeGroup::stop,
           py::call_guard<py::gil_scoped_release>())
      .def("take",
           [](UprobeGroup& g) {
             std::vector<ProbeFire> out;
             {
               py::gil_scoped_release rel;
               out = g.take();
             }
             return out;
           })
      .def_property_readonly("lost", &UprobeGroup::lost)
      .def_property_readonly("n_events", &UprobeGroup::n_events)
      .def_property_readonly("throttles", &UprobeGroup::throttles)
      .def_property_readonly("throttled_fires",
                             &UprobeGroup::throttled_fires);

  py::class_<PerfSampler>(m, "PerfSampler")
      .def(py::init<int, bool, uint32_t, int, bool, int>(),
           py::arg("freq") = 19, py::arg("dwarf_stacks") = false,
           py::arg("stack_dump_size") = 16384, py::arg("ring_pages") = 64,
           py::arg("track_mmaps") = true, py::arg("target_pid") = -1)
      .def("start", &PerfSampler::start,
           py::call_guard<py::gil_scoped_release>())
      .def("stop", &PerfSampler::stop,
           py::call_guard<py::gil_scoped_release>())
      .def(
          "take",
          [](PerfSampler& s) {
            Batch b;
            {
              py::gil_scoped_release rel;
              b = s.take();
            }
            return py::make_tuple(std::move(b.samples),
                                  std::move(b.proc_events));
          })
      .def_property_readonly("lost", &PerfSampler::lost_count)
      .def_property_readonly("n_samples", &PerfSampler::sample_count)
      .def_property_readonly("unknown", &PerfSampler::unknown_count)
      .def_property_readonly("n_cpus", &PerfSampler::n_cpus);
}
