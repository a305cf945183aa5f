"""Native custom labels: include/parca_custom_labels.h writer +
parca_agent_amd/nativelabels.py reader + CPU-service join.

Reference capability: native custom labels attached per-sample
(parca_reporter.go:362-374; metrics/all.go:1442-1477). Transport here
is a per-process shm table keyed by tid (see header docstring)."""

import os
import subprocess
import time

import pytest

from parca_agent_amd.nativelabels import (HEADER, MAGIC, SLOT_HEAD,
                                          NativeLabelReader)

perf = pytest.mark.skipif(
    os.geteuid() != 0, reason="needs root for perf_event_open")

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

NSLOTS, MAXL, KLEN, VLEN = 512, 8, 32, 64
SLOT_SIZE = SLOT_HEAD.size + MAXL * (KLEN + VLEN)


def _mk_table(slots):
    """Build table bytes exactly as the C header lays them out."""
    buf = bytearray(HEADER.size + NSLOTS * SLOT_SIZE)
    HEADER.pack_into(buf, 0, MAGIC, 1, NSLOTS, MAXL, KLEN, VLEN, 0)
    for idx, (tid, seq, labels) in slots.items():
        off = HEADER.size + idx * SLOT_SIZE
        SLOT_HEAD.pack_into(buf, off, tid, seq, len(labels), 0)
        for i, (k, v) in enumerate(labels):
            p = off + SLOT_HEAD.size + i * (KLEN + VLEN)
            buf[p:p + len(k)] = k.encode()
            buf[p + KLEN:p + KLEN + len(v)] = v.encode()
    return bytes(buf)


def _reader_for(tmp_path, pid, table_bytes):
    (tmp_path / f"parca_labels_{pid}").write_bytes(table_bytes)
    return NativeLabelReader(directory=str(tmp_path))


def test_reader_basic(tmp_path):
    tid = 4242
    r = _reader_for(tmp_path, 100, _mk_table({
        tid % NSLOTS: (tid, 2, [("service", "billing"),
                                ("endpoint", "/checkout")])}))
    assert r.labels_for(100, tid) == (("service", "billing"),
                                      ("endpoint", "/checkout"))
    assert r.labels_for(100, tid + 1) == ()  # other tid: free slot
    assert r.samples_labeled == 1


def test_reader_probing_and_holes(tmp_path):
    # Two tids colliding on the same home slot; second lives one over.
    t1, t2 = 7, 7 + NSLOTS
    r = _reader_for(tmp_path, 5, _mk_table({
        7: (t1, 0, [("a", "1")]),
        8: (t2, 4, [("b", "2")]),
    }))
    assert r.labels_for(5, t1) == (("a", "1"),)
    assert r.labels_for(5, t2) == (("b", "2"),)

    # Deleted label leaves a hole: count=1 but label parked at index 2.
    buf = bytearray(_mk_table({33: (33, 0, [])}))
    off = HEADER.size + 33 * SLOT_SIZE
    SLOT_HEAD.pack_into(buf, off, 33, 0, 1, 0)
    p = off + SLOT_HEAD.size + 2 * (KLEN + VLEN)
    buf[p:p + 4] = b"late"
    buf[p + KLEN:p + KLEN + 2] = b"ok"
    (tmp_path / "parca_labels_6").write_bytes(bytes(buf))
    r2 = NativeLabelReader(directory=str(tmp_path))
    assert r2.labels_for(6, 33) == (("late", "ok"),)


def test_reader_seqlock_and_negative_cache(tmp_path):
    # Odd seq = writer mid-update: labels must be dropped, not torn.
    tid = 9
    r = _reader_for(tmp_path, 77, _mk_table({
        9: (tid, 3, [("half", "written")])}))
    assert r.labels_for(77, tid) == ()

    # Missing file is negative-cached.
    r2 = NativeLabelReader(directory=str(tmp_path))
    assert r2.labels_for(424242, 1) == ()
    assert 424242 in r2._negative


def test_reader_rejects_garbage(tmp_path):
    (tmp_path / "parca_labels_55").write_bytes(b"\x00" * 4096)
    r = NativeLabelReader(directory=str(tmp_path))
    assert r.labels_for(55, 1) == ()


LABEL_PROG = r"""
#include "parca_custom_labels.h"
#include <pthread.h>
#include <stdio.h>
#include <time.h>

static volatile unsigned long sink;

static void burn_until(double deadline) {
  struct timespec ts;
  do {
    for (int i = 0; i < 2000000; i++) sink += i;
    clock_gettime(CLOCK_MONOTONIC, &ts);
  } while (ts.tv_sec + ts.tv_nsec * 1e-9 < deadline);
}

static double now(void) {
  struct timespec ts;
  clock_gettime(CLOCK_MONOTONIC, &ts);
  return ts.tv_sec + ts.tv_nsec * 1e-9;
}

static double g_deadline;

static void *worker(void *arg) {
  (void)arg;
  parca_label_set("service", "billing");
  parca_label_set("endpoint", "/checkout");
  printf("worker_tid=%ld\n", syscall(SYS_gettid));
  fflush(stdout);
  burn_until(g_deadline);
  return NULL;
}

int main(int argc, char **argv) {
  double secs = argc > 1 ? atof(argv[1]) : 2.0;
  g_deadline = now() + secs;
  parca_label_set("service", "billing");
  parca_label_set("service", "frontend");  /* overwrite */
  printf("main_tid=%ld\n", syscall(SYS_gettid));
  fflush(stdout);
  pthread_t th;
  pthread_create(&th, NULL, worker, NULL);
  burn_until(g_deadline);
  pthread_join(th, NULL);
  return 0;
}
"""


@pytest.fixture(scope="module")
def label_binary(tmp_path_factory):
    d = tmp_path_factory.mktemp("lblbin")
    src = d / "lbl.c"
    src.write_text(LABEL_PROG)
    binary = d / "lbl"
    subprocess.run(
        ["gcc", "-O2", "-Wall", "-Werror", "-fno-omit-frame-pointer",
         "-I", os.path.join(REPO, "include"), str(src), "-o", str(binary),
         "-lpthread"],
        check=True)
    return str(binary)


def _start_prog(label_binary, tmp_path, secs):
    env = dict(os.environ, PARCA_LABELS_DIR=str(tmp_path))
    proc = subprocess.Popen([label_binary, str(secs)], env=env,
                            stdout=subprocess.PIPE, text=True)
    tids = {}
    for _ in range(2):
        line = proc.stdout.readline().strip()
        name, _, tid = line.partition("=")
        tids[name] = int(tid)
    return proc, tids


def test_c_writer_python_reader(label_binary, tmp_path):
    """ABI check: the C header's table parses with the Python reader."""
    proc, tids = _start_prog(label_binary, tmp_path, 1.0)
    try:
        deadline = time.monotonic() + 5
        r = NativeLabelReader(directory=str(tmp_path))
        main_l = worker_l = None
        while time.monotonic() < deadline and not (main_l and worker_l):
            main_l = r.labels_for(proc.pid, tids["main_tid"]) or main_l
            worker_l = r.labels_for(proc.pid,
                                    tids["worker_tid"]) or worker_l
            time.sleep(0.05)
        assert main_l == (("service", "frontend"),)  # overwrite won
        assert dict(worker_l or ()) == {"service": "billing",
                                        "endpoint": "/checkout"}
    finally:
        proc.wait(timeout=15)


@perf
def test_labels_attached_to_samples(label_binary, tmp_path):
    """End to end: samples of the labeled program carry its per-thread
    labels after the reporter's label merge."""
    from parca_agent_amd.cpu import CPUSamplerService
    from parca_agent_amd.reporter import Reporter

    class Dest:
        def __init__(self):
            self.samples = []

        def write_batch(self, batch):
            self.samples.extend(batch)

        def close(self):
            pass

    proc, tids = _start_prog(label_binary, tmp_path, 3.0)
    try:
        dest = Dest()
        rep = Reporter([dest], cpu_sampling_frequency=97)
        svc = CPUSamplerService(rep, freq=97, poll_interval=0.05)
        svc.native_labels.directory = str(tmp_path)
        svc.start()
        proc.wait(timeout=30)
        time.sleep(0.3)
        svc.stop()
        rep.flush()
    finally:
        if proc.poll() is None:
            proc.kill()

    by_tid = {}
    for s in dest.samples:
        t = s.labels.get("thread_id")
        if t is not None:
            by_tid.setdefault(int(t), []).append(dict(s.labels))
    main_lbls = by_tid.get(tids["main_tid"], [])
    worker_lbls = by_tid.get(tids["worker_tid"], [])
    assert main_lbls and worker_lbls, f"no samples per thread: {by_tid.keys()}"
    assert any(d.get("service") == "frontend" for d in main_lbls)
    assert any(d.get("service") == "billing" and
               d.get("endpoint") == "/checkout" for d in worker_lbls)


def test_python_writer_roundtrip(tmp_path, monkeypatch):
    """labels_client.py writes the same layout the agent reads — any
    mmap-capable language can publish labels, not just C."""
    import importlib
    import threading

    monkeypatch.setenv("PARCA_LABELS_DIR", str(tmp_path))
    import parca_agent_amd.labels_client as lc
    importlib.reload(lc)  # drop any table cached under another dir

    lc.label_set("service", "checkout")
    lc.label_set("tier", "gold")
    lc.label_set("tier", "platinum")       # overwrite
    lc.label_set("service", "")            # delete

    got = {}

    def worker():
        lc.label_set("service", "worker-svc")
        got["tid"] = threading.get_native_id()

    t = threading.Thread(target=worker)
    t.start()
    t.join()

    r = NativeLabelReader(directory=str(tmp_path))
    pid = os.getpid()
    me = threading.get_native_id()
    assert dict(r.labels_for(pid, me)) == {"tier": "platinum"}
    assert dict(r.labels_for(pid, got["tid"])) == {
        "service": "worker-svc"}

    lc.labels_clear()
    # reader caches the table; fresh reader sees the cleared slot
    r2 = NativeLabelReader(directory=str(tmp_path))
    assert r2.labels_for(pid, me) == ()
    os.unlink(tmp_path / f"parca_labels_{pid}")


@pytest.mark.skipif(os.geteuid() != 0, reason="chown needs root")
def test_spoofed_table_rejected(tmp_path):
    """A label table not owned by the process it claims to describe
    (nor by root) is ignored — other users can't inject labels into a
    victim's samples."""
    pid = os.getpid()  # a live root-owned process
    path = tmp_path / f"parca_labels_{pid}"
    path.write_bytes(_mk_table({1: (1, 0, [("evil", "spoof")])}))
    os.chown(path, 12345, 12345)
    r = NativeLabelReader(directory=str(tmp_path))
    assert r.labels_for(pid, 1) == ()
    assert pid in r._negative
