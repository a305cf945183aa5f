// ThreadSanitizer stress test for the shm SPSC ring (csrc/rocprof/ring.h).
//
// The reference runs its whole suite under the Go race detector
// (SURVEY.md §5.2); this is the native analog for the one lock-free
// structure two processes share: a producer thread hammers variable-size
// records while a consumer drains concurrently, under -fsanitize=thread.
// Built and executed by tests/test_native_races.py.

#include <atomic>
#include <cstdio>
#include <cstring>
#include <thread>
#include <vector>

#include "rocprof/ring.h"

using namespace parca;

namespace {

// Consumer mirroring the logic of RingConsumer::drain (gpu_module.cc).
struct Consumer {
  RingHeader* hdr;
  uint8_t* data;
  uint64_t cap;

  template <typename Fn>
  size_t drain(Fn&& cb) {
    uint64_t tail = hdr->tail.load(std::memory_order_relaxed);
    uint64_t head = hdr->head.load(std::memory_order_acquire);
    size_t n = 0;
    while (tail < head) {
      RecordHeader rh;
      copy_out(tail & (cap - 1), &rh, sizeof(rh));
      if (rh.size < sizeof(rh) || rh.size > cap) {
        tail = head;
        break;
      }
      std::vector<uint8_t> payload(rh.size - sizeof(rh));
      copy_out((tail + sizeof(rh)) & (cap - 1), payload.data(),
               payload.size());
      cb(rh.type, payload);
      tail += rh.size;
      ++n;
    }
    hdr->tail.store(tail, std::memory_order_release);
    return n;
  }

  void copy_out(uint64_t off, void* dst, size_t n) {
    uint64_t first = cap - off;
    if (n <= first) {
      memcpy(dst, data + off, n);
    } else {
      memcpy(dst, data + off, first);
      memcpy(static_cast<uint8_t*>(dst) + first, data, n - first);
    }
  }
};

}  // namespace

int main() {
  constexpr uint64_t kCap = 1 << 14;  // small: force wraps + fulls
  std::vector<uint8_t> mem(sizeof(RingHeader) + kCap, 0);
  RingProducer producer(mem.data(), kCap, 1234);
  auto* hdr = reinterpret_cast<RingHeader*>(mem.data());
  Consumer consumer{hdr, mem.data() + sizeof(RingHeader), kCap};

  constexpr int kRecords = 200000;
  std::atomic<bool> done{false};
  std::atomic<uint64_t> produced{0};
  std::atomic<uint64_t> checksum_in{0};

  std::thread prod([&] {
    uint8_t buf[512];
    for (int i = 0; i < kRecords; ++i) {
      uint32_t size = 8 + (i * 37) % 480;
      // Record sizes are 8-byte padded on the ring and pad bytes carry
      // stale ring content by design (Python decoders parse structured
      // payloads and ignore the tail), so the payload self-describes its
      // length for checksumming.
      memcpy(buf, &size, 4);
      for (uint32_t j = 4; j < size; ++j)
        buf[j] = static_cast<uint8_t>(i + j);
      uint64_t sum = 0;
      for (uint32_t j = 4; j < size; ++j) sum += buf[j];
      // Retry while full: the consumer is draining concurrently.
      int spins = 0;
      while (!producer.write(i % 7 + 1, buf, size)) {
        if (++spins > 1000000) {
          fprintf(stderr, "producer livelock\n");
          done = true;
          return;
        }
        std::this_thread::yield();
      }
      checksum_in += sum;
      produced++;
    }
    done = true;
  });

  uint64_t consumed = 0;
  uint64_t checksum_out = 0;
  bool corrupt = false;
  while (!done.load() || consumed < produced.load()) {
    consumer.drain([&](uint32_t type, const std::vector<uint8_t>& payload) {
      if (type < 1 || type > 7) corrupt = true;
      uint32_t size = 0;
      if (payload.size() >= 4) memcpy(&size, payload.data(), 4);
      if (size < 4 || size > payload.size()) {
        corrupt = true;
        size = 0;
      }
      uint64_t sum = 0;
      for (uint32_t j = 4; j < size; ++j) sum += payload[j];
      checksum_out += sum;
      consumed++;
    });
    std::this_thread::yield();
  }
  prod.join();

  // Note: `dropped` counts failed write ATTEMPTS (the producer retries
  // here; production callers do not), so it is informational only.
  uint64_t dropped = hdr->dropped.load();
  if (corrupt || consumed != static_cast<uint64_t>(kRecords) ||
      checksum_in.load() != checksum_out) {
    fprintf(stderr,
            "FAIL corrupt=%d consumed=%llu/%d sum_in=%llu sum_out=%llu "
            "dropped=%llu\n",
            corrupt, (unsigned long long)consumed, kRecords,
            (unsigned long long)checksum_in.load(),
            (unsigned long long)checksum_out, (unsigned long long)dropped);
    return 1;
  }
  printf("OK records=%llu wraps over %llu bytes, dropped=%llu\n",
         (unsigned long long)consumed,
         (unsigned long long)hdr->head.load(),
         (unsigned long long)dropped);
  return 0;
}
