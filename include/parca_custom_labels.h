/* parca-agent-amd native custom labels — single-header C API.
 *
 * Programs publish per-thread labels ("service"="billing",
 * "endpoint"="/checkout", ...) that the agent attaches to every CPU
 * sample taken while the label is set. This is the reference's native
 * custom-labels capability (label write-out: parca_reporter.go:362-374;
 * unwinder-side counters: metrics/all.go:1442-1477) re-designed for a
 * userspace perf agent: the BPF path reads a TLS label set via fs_base,
 * which perf_event samples cannot capture, so here the library
 * publishes labels in a small shared-memory table keyed by tid
 * (/dev/shm/parca_labels_<pid>) and the agent joins on (pid, tid) per
 * sample with a seqlock-consistent read — no ptrace, no TLS layout
 * knowledge, works from any language that can mmap a file.
 *
 * Usage (C/C++):
 *   #include "parca_custom_labels.h"
 *   parca_label_set("endpoint", "/checkout");
 *   ... work ...
 *   parca_label_set("endpoint", "");   // empty value deletes
 *
 * All functions are async-signal-unsafe but thread-safe; each thread
 * owns one table slot (claimed by tid, linear probing). Label churn is
 * wait-free for readers via a per-slot seqlock. The file outlives no
 * one: it is unlinked-on-boot-stale by the agent and sized ~400 KiB.
 *
 * Header-only: every translation unit that includes this gets its own
 * static state, which is correct (they all map the same file) at the
 * cost of one mmap per TU that calls into it.
 */
#ifndef PARCA_CUSTOM_LABELS_H
#define PARCA_CUSTOM_LABELS_H

#include <fcntl.h>
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <sys/syscall.h>
#include <unistd.h>

#ifdef __cplusplus
extern "C" {
#endif

#if defined(__GNUC__) || defined(__clang__)
#define PARCA_LABELS_UNUSED __attribute__((unused))
#else
#define PARCA_LABELS_UNUSED
#endif

#define PARCA_LABELS_MAGIC 0x53424C4350ULL /* "PCLBS" */
#define PARCA_LABELS_VERSION 1u
#define PARCA_LABELS_NSLOTS 512u
#define PARCA_LABELS_MAX 8u
#define PARCA_LABELS_KEY_LEN 32u
#define PARCA_LABELS_VAL_LEN 64u

typedef struct {
  char key[PARCA_LABELS_KEY_LEN]; /* NUL-terminated; key[0]==0 => unused */
  char val[PARCA_LABELS_VAL_LEN]; /* NUL-terminated */
} parca_label_t;

typedef struct {
  uint32_t tid; /* 0 = free slot */
  uint32_t seq; /* seqlock: odd while the owner writes */
  uint32_t count;
  uint32_t _pad;
  parca_label_t labels[PARCA_LABELS_MAX];
} parca_label_slot_t;

typedef struct {
  uint64_t magic;
  uint32_t version;
  uint32_t nslots;
  uint32_t max_labels;
  uint32_t key_len;
  uint32_t val_len;
  uint32_t _pad;
  uint8_t _rsvd[32];
} parca_label_header_t;

static parca_label_header_t *parca_labels__hdr_;
static __thread parca_label_slot_t *parca_labels__slot_;

static PARCA_LABELS_UNUSED size_t parca_labels__size_(void) {
  return sizeof(parca_label_header_t) +
         (size_t)PARCA_LABELS_NSLOTS * sizeof(parca_label_slot_t);
}

static PARCA_LABELS_UNUSED parca_label_header_t *parca_labels__map_(void) {
  if (parca_labels__hdr_) return parca_labels__hdr_;
  char path[128];
  const char *dir = getenv("PARCA_LABELS_DIR");
  if (!dir || !dir[0]) dir = "/dev/shm";
  snprintf(path, sizeof(path), "%s/parca_labels_%d", dir, (int)getpid());
  int fd = open(path, O_RDWR | O_CREAT, 0644);
  if (fd < 0) return NULL;
  size_t sz = parca_labels__size_();
  struct stat st;
  if (fstat(fd, &st) != 0 || (size_t)st.st_size < sz) {
    if (ftruncate(fd, (off_t)sz) != 0) {
      close(fd);
      return NULL;
    }
  }
  void *m = mmap(NULL, sz, PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
  close(fd);
  if (m == MAP_FAILED) return NULL;
  parca_label_header_t *h = (parca_label_header_t *)m;
  if (h->magic != PARCA_LABELS_MAGIC) {
    /* First mapper in this process initializes; racing TUs/threads all
     * write identical bytes, and readers gate on the magic landing
     * last via the release fence below. */
    h->version = PARCA_LABELS_VERSION;
    h->nslots = PARCA_LABELS_NSLOTS;
    h->max_labels = PARCA_LABELS_MAX;
    h->key_len = PARCA_LABELS_KEY_LEN;
    h->val_len = PARCA_LABELS_VAL_LEN;
    __atomic_store_n(&h->magic, (uint64_t)PARCA_LABELS_MAGIC,
                     __ATOMIC_RELEASE);
  }
  parca_labels__hdr_ = h;
  return h;
}

static PARCA_LABELS_UNUSED parca_label_slot_t *parca_labels__slot(void) {
  if (parca_labels__slot_) return parca_labels__slot_;
  parca_label_header_t *h = parca_labels__map_();
  if (!h) return NULL;
  parca_label_slot_t *slots = (parca_label_slot_t *)(h + 1);
  uint32_t tid = (uint32_t)syscall(SYS_gettid);
  uint32_t idx = tid % PARCA_LABELS_NSLOTS;
  for (uint32_t probe = 0; probe < PARCA_LABELS_NSLOTS; ++probe) {
    parca_label_slot_t *s = &slots[(idx + probe) % PARCA_LABELS_NSLOTS];
    uint32_t cur = __atomic_load_n(&s->tid, __ATOMIC_ACQUIRE);
    if (cur == tid) { /* tid recycled into our process: take it over */
      parca_labels__slot_ = s;
      return s;
    }
    if (cur == 0) {
      uint32_t expect = 0;
      if (__atomic_compare_exchange_n(&s->tid, &expect, tid, 0,
                                      __ATOMIC_ACQ_REL,
                                      __ATOMIC_ACQUIRE)) {
        parca_labels__slot_ = s;
        return s;
      }
    }
  }
  return NULL; /* table full: labels silently off for this thread */
}

/* Set (or with value=="" delete) one label on the calling thread. */
static PARCA_LABELS_UNUSED void parca_label_set(const char *key, const char *value) {
  if (!key || !key[0]) return;
  parca_label_slot_t *s = parca_labels__slot();
  if (!s) return;
  __atomic_store_n(&s->seq, s->seq | 1u, __ATOMIC_RELEASE); /* odd */
  __atomic_thread_fence(__ATOMIC_ACQ_REL);
  int found = -1, freei = -1;
  for (uint32_t i = 0; i < PARCA_LABELS_MAX; ++i) {
    if (s->labels[i].key[0] == 0) {
      if (freei < 0) freei = (int)i;
    } else if (strncmp(s->labels[i].key, key,
                       PARCA_LABELS_KEY_LEN - 1) == 0) {
      found = (int)i;
      break;
    }
  }
  if (value && value[0]) {
    int at = found >= 0 ? found : freei;
    if (at >= 0) {
      parca_label_t *l = &s->labels[at];
      strncpy(l->key, key, PARCA_LABELS_KEY_LEN - 1);
      l->key[PARCA_LABELS_KEY_LEN - 1] = 0;
      strncpy(l->val, value, PARCA_LABELS_VAL_LEN - 1);
      l->val[PARCA_LABELS_VAL_LEN - 1] = 0;
      if (found < 0) s->count++;
    }
  } else if (found >= 0) {
    s->labels[found].key[0] = 0;
    s->labels[found].val[0] = 0;
    if (s->count) s->count--;
  }
  __atomic_thread_fence(__ATOMIC_ACQ_REL);
  __atomic_store_n(&s->seq, (s->seq + 1u) & ~1u, __ATOMIC_RELEASE);
}

/* Remove every label on the calling thread. */
static PARCA_LABELS_UNUSED void parca_labels_clear(void) {
  parca_label_slot_t *s = parca_labels__slot();
  if (!s) return;
  __atomic_store_n(&s->seq, s->seq | 1u, __ATOMIC_RELEASE);
  __atomic_thread_fence(__ATOMIC_ACQ_REL);
  memset(s->labels, 0, sizeof(s->labels));
  s->count = 0;
  __atomic_thread_fence(__ATOMIC_ACQ_REL);
  __atomic_store_n(&s->seq, (s->seq + 1u) & ~1u, __ATOMIC_RELEASE);
}

#ifdef __cplusplus
}
#endif

#endif /* PARCA_CUSTOM_LABELS_H */
