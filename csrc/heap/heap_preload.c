/* parca heap sampler: LD_PRELOAD allocation profiler whose state
 * survives the process.
 *
 * The reference ships REAL allocation profiles at OOM kill via oomprof
 * (reference: oom/oomprof.go:16-125) — but only for Go processes, by
 * introspecting the Go runtime. This MI355X-native equivalent works for
 * any process (PyTorch ranks included): a byte-rate allocation sampler
 * interposed on the malloc family keeps a stack-aggregated profile in a
 * shared-memory file (/dev/shm/parca_heap_<pid>). shm files OUTLIVE the
 * process, so when the kernel OOM-kills the workload the agent's OOM
 * watcher (oom/watcher.py + oom/heap.py) reads the file post-mortem and
 * ships alloc/inuse space+objects profiles — "what allocated the
 * memory", not just "who died".
 *
 * Design:
 *  - every thread accumulates allocation bytes; each time the counter
 *    crosses `sample_rate` (default 512 KiB — the reference's memPeriod,
 *    parca_reporter.go memory origin), the triggering allocation is
 *    sampled: stack captured (frame-pointer walk with bounds checks,
 *    cheap and safe in signal-free context), folded into an
 *    open-addressing table keyed by stack hash.
 *  - sampled pointers go into a live-pointer table so free() can move
 *    their bytes from inuse to freed — giving real inuse_space.
 *  - /proc/self/maps text is snapshotted into the file (refreshed every
 *    256 samples) so the agent can symbolize post-mortem.
 *  - TLS reentrance guard; fork children re-open their own file.
 *
 * Build: gcc -O2 -shared -fPIC -o libparca_heap.so heap_preload.c -ldl
 */

#define _GNU_SOURCE
#include <dlfcn.h>
#include <fcntl.h>
#include <pthread.h>
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

#define PARCA_HEAP_MAGIC 0x48504150u /* "PAPH" */
#define MAX_FRAMES 24
#define N_ENTRIES 16384      /* stack aggregation slots (power of two) */
#define N_LIVE 65536         /* sampled live pointers (power of two) */
#define MAPS_CAP (256 * 1024)

typedef struct {
  uint64_t stack_hash;
  uint32_t n_frames;
  uint32_t _pad;
  uint64_t ips[MAX_FRAMES];
  uint64_t alloc_bytes; /* sampled bytes allocated from this stack */
  uint64_t alloc_count;
  uint64_t free_bytes; /* sampled bytes since freed */
  uint64_t free_count;
} heap_entry_t;

typedef struct {
  uint64_t ptr; /* 0 = empty */
  uint32_t entry;
  uint32_t _pad;
  uint64_t size;
} live_slot_t;

typedef struct {
  uint32_t magic;
  uint32_t version;
  uint32_t pid;
  uint32_t n_entries;
  uint64_t sample_rate;
  uint64_t samples;
  uint64_t dropped_entries; /* table-full stacks */
  uint64_t maps_len;
  uint8_t _pad[64 - 48];
  /* heap_entry_t entries[N_ENTRIES]; live_slot_t live[N_LIVE];
     char maps[MAPS_CAP]; */
} heap_header_t;

static heap_header_t *g_hdr;
static heap_entry_t *g_entries;
static live_slot_t *g_live;
static char *g_maps;
static uint64_t g_rate = 512 * 1024;
static char g_path[256];

static void *(*real_malloc)(size_t);
static void *(*real_calloc)(size_t, size_t);
static void *(*real_realloc)(void *, size_t);
static void (*real_free)(void *);
static int (*real_posix_memalign)(void **, size_t, size_t);
static void *(*real_aligned_alloc)(size_t, size_t);

static __thread uint64_t t_accum;
static __thread int t_busy; /* reentrance guard */

/* bootstrap arena for dlsym's own early allocations */
static char boot_buf[16384];
static size_t boot_used;

static void *boot_alloc(size_t n) {
  n = (n + 15) & ~(size_t)15;
  if (boot_used + n > sizeof(boot_buf)) return NULL;
  void *p = boot_buf + boot_used;
  boot_used += n;
  return p;
}

static int is_boot_ptr(void *p) {
  return (char *)p >= boot_buf && (char *)p < boot_buf + sizeof(boot_buf);
}

static void snapshot_maps(void) {
  if (!g_maps) return;
  int fd = open("/proc/self/maps", O_RDONLY);
  if (fd < 0) return;
  size_t off = 0;
  ssize_t n;
  while (off < MAPS_CAP - 1 &&
         (n = read(fd, g_maps + off, MAPS_CAP - 1 - off)) > 0)
    off += (size_t)n;
  close(fd);
  g_maps[off] = 0;
  __atomic_store_n(&g_hdr->maps_len, off, __ATOMIC_RELEASE);
}

static void heap_open(void) {
  const char *dir = getenv("PARCA_HEAP_DIR");
  if (!dir) dir = "/dev/shm";
  const char *rate = getenv("PARCA_HEAP_SAMPLE_RATE");
  if (rate) {
    uint64_t r = strtoull(rate, NULL, 10);
    if (r >= 4096) g_rate = r;
  }
  snprintf(g_path, sizeof(g_path), "%s/parca_heap_%d", dir, getpid());
  size_t total = sizeof(heap_header_t) + sizeof(heap_entry_t) * N_ENTRIES +
                 sizeof(live_slot_t) * N_LIVE + MAPS_CAP;
  int fd = open(g_path, O_CREAT | O_RDWR | O_TRUNC, 0644);
  if (fd < 0) return;
  if (ftruncate(fd, (off_t)total) != 0) {
    close(fd);
    return;
  }
  void *mem = mmap(NULL, total, PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
  close(fd);
  if (mem == MAP_FAILED) return;
  g_hdr = (heap_header_t *)mem;
  g_entries = (heap_entry_t *)(g_hdr + 1);
  g_live = (live_slot_t *)(g_entries + N_ENTRIES);
  g_maps = (char *)(g_live + N_LIVE);
  g_hdr->version = 1;
  g_hdr->pid = (uint32_t)getpid();
  g_hdr->n_entries = N_ENTRIES;
  g_hdr->sample_rate = g_rate;
  snapshot_maps();
  __atomic_store_n(&g_hdr->magic, PARCA_HEAP_MAGIC, __ATOMIC_RELEASE);
}

static void child_reinit(void) {
  /* the parent's mapping must not be scribbled on by the child */
  g_hdr = NULL;
  g_entries = NULL;
  g_live = NULL;
  g_maps = NULL;
  heap_open();
}

static void __attribute__((constructor)) heap_init(void) {
  real_malloc = (void *(*)(size_t))dlsym(RTLD_NEXT, "malloc");
  real_calloc = (void *(*)(size_t, size_t))dlsym(RTLD_NEXT, "calloc");
  real_realloc = (void *(*)(void *, size_t))dlsym(RTLD_NEXT, "realloc");
  real_free = (void (*)(void *))dlsym(RTLD_NEXT, "free");
  real_posix_memalign =
      (int (*)(void **, size_t, size_t))dlsym(RTLD_NEXT, "posix_memalign");
  real_aligned_alloc =
      (void *(*)(size_t, size_t))dlsym(RTLD_NEXT, "aligned_alloc");
  heap_open();
  pthread_atfork(NULL, NULL, child_reinit);
}

/* Frame-pointer stack walk: safe (bounds-checked against the thread
 * stack) and allocation-free. FP-less frames simply truncate. */
static int capture_stack(uint64_t *ips, int max) {
  void *fp = __builtin_frame_address(0);
  uintptr_t lo = (uintptr_t)&fp;
  uintptr_t hi = lo + (8u << 20);
  int n = 0;
  while (n < max) {
    uintptr_t f = (uintptr_t)fp;
    if (f < lo || f > hi - 16 || (f & 7)) break;
    uint64_t ret = *((uint64_t *)f + 1);
    if (ret < 0x1000) break;
    ips[n++] = ret;
    void *next = *(void **)f;
    if ((uintptr_t)next <= f) break;
    fp = next;
  }
  return n;
}

static void record_sample(void *ptr, size_t size, uint64_t est_bytes) {
  uint64_t ips[MAX_FRAMES];
  int n = capture_stack(ips, MAX_FRAMES);
  uint64_t h = 1469598103934665603ull;
  for (int i = 0; i < n; i++) {
    h ^= ips[i];
    h *= 1099511628211ull;
  }
  if (h == 0) h = 1;
  uint32_t idx = (uint32_t)(h & (N_ENTRIES - 1));
  heap_entry_t *e = NULL;
  for (uint32_t probe = 0; probe < 64; probe++) {
    heap_entry_t *cand = &g_entries[(idx + probe) & (N_ENTRIES - 1)];
    uint64_t cur = __atomic_load_n(&cand->stack_hash, __ATOMIC_ACQUIRE);
    if (cur == h) {
      e = cand;
      break;
    }
    if (cur == 0) {
      uint64_t expect = 0;
      if (__atomic_compare_exchange_n(&cand->stack_hash, &expect, h, 0,
                                      __ATOMIC_ACQ_REL, __ATOMIC_ACQUIRE)) {
        cand->n_frames = (uint32_t)n;
        memcpy(cand->ips, ips, sizeof(uint64_t) * (size_t)n);
        e = cand;
        break;
      }
      if (expect == h) {
        e = cand;
        break;
      }
    }
  }
  if (!e) {
    __atomic_fetch_add(&g_hdr->dropped_entries, 1, __ATOMIC_RELAXED);
    return;
  }
  __atomic_fetch_add(&e->alloc_bytes, est_bytes, __ATOMIC_RELAXED);
  __atomic_fetch_add(&e->alloc_count, 1, __ATOMIC_RELAXED);
  uint64_t ns = __atomic_fetch_add(&g_hdr->samples, 1, __ATOMIC_RELAXED);
  if ((ns & 255) == 255) snapshot_maps();

  /* remember the pointer so free() can migrate inuse -> freed */
  uint32_t eidx = (uint32_t)(e - g_entries);
  uint64_t p = (uint64_t)(uintptr_t)ptr;
  uint32_t li = (uint32_t)((p >> 4) & (N_LIVE - 1));
  for (uint32_t probe = 0; probe < 32; probe++) {
    live_slot_t *s = &g_live[(li + probe) & (N_LIVE - 1)];
    uint64_t expect = 0;
    if (__atomic_compare_exchange_n(&s->ptr, &expect, p, 0, __ATOMIC_ACQ_REL,
                                    __ATOMIC_ACQUIRE)) {
      s->entry = eidx;
      s->size = est_bytes;
      return;
    }
  }
}

static void note_alloc(void *ptr, size_t size) {
  if (!g_hdr || !ptr || t_busy) return;
  t_accum += size;
  if (t_accum < g_rate) return;
  uint64_t est = t_accum;
  t_accum = 0;
  t_busy = 1;
  record_sample(ptr, size, est);
  t_busy = 0;
}

static void note_free(void *ptr) {
  if (!g_hdr || !ptr) return;
  uint64_t p = (uint64_t)(uintptr_t)ptr;
  uint32_t li = (uint32_t)((p >> 4) & (N_LIVE - 1));
  for (uint32_t probe = 0; probe < 32; probe++) {
    live_slot_t *s = &g_live[(li + probe) & (N_LIVE - 1)];
    uint64_t cur = __atomic_load_n(&s->ptr, __ATOMIC_ACQUIRE);
    if (cur == 0) return;
    if (cur != p) continue;
    uint64_t expect = p;
    if (__atomic_compare_exchange_n(&s->ptr, &expect, (uint64_t)-1, 0,
                                    __ATOMIC_ACQ_REL, __ATOMIC_ACQUIRE)) {
      heap_entry_t *e = &g_entries[s->entry];
      __atomic_fetch_add(&e->free_bytes, s->size, __ATOMIC_RELAXED);
      __atomic_fetch_add(&e->free_count, 1, __ATOMIC_RELAXED);
      __atomic_store_n(&s->ptr, 0, __ATOMIC_RELEASE);
    }
    return;
  }
}

void *malloc(size_t size) {
  if (!real_malloc) return boot_alloc(size);
  void *p = real_malloc(size);
  note_alloc(p, size);
  return p;
}

void *calloc(size_t n, size_t size) {
  if (!real_calloc) {
    void *p = boot_alloc(n * size);
    if (p) memset(p, 0, n * size);
    return p;
  }
  void *p = real_calloc(n, size);
  note_alloc(p, n * size);
  return p;
}

void *realloc(void *old, size_t size) {
  if (!real_realloc) return boot_alloc(size);
  if (old && !is_boot_ptr(old)) note_free(old);
  void *p = is_boot_ptr(old) ? boot_alloc(size) : real_realloc(old, size);
  note_alloc(p, size);
  return p;
}

void free(void *p) {
  if (!p || is_boot_ptr(p)) return;
  if (real_free) {
    note_free(p);
    real_free(p);
  }
}

int posix_memalign(void **out, size_t align, size_t size) {
  if (!real_posix_memalign) return 12 /* ENOMEM */;
  int rc = real_posix_memalign(out, align, size);
  if (rc == 0) note_alloc(*out, size);
  return rc;
}

void *aligned_alloc(size_t align, size_t size) {
  if (!real_aligned_alloc) return NULL;
  void *p = real_aligned_alloc(align, size);
  note_alloc(p, size);
  return p;
}
