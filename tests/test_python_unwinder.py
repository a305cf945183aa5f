"""CPython interpreter unwinder tests.

Two layers, matching the reference's strategy for its per-language
unwinders:

1. Live end-to-end: a REAL python subprocess (a foreign process probed
   entirely through /proc + process_vm_readv — the unwinder no longer
   introspects the agent's own interpreter at all) is calibrated and
   unwound, including exact tid->tstate matching for worker threads.
2. Synthetic memory images laid out like CPython 3.8 (PyFrameObject
   chain), 3.11 (cframe at +8), 3.12 (cframe at +0) and 3.13 (direct
   current_frame) prove the invariant-driven calibrator handles every
   frame-chain era without version tables — the container only ships
   3.10, so the other eras are exercised against faithful struct mocks.
"""

import os
import struct
import subprocess
import sys
import textwrap
import time

import pytest

from parca_agent_amd.interp.python import (
    Anchors,
    Calibrator,
    PyProcess,
    PythonUnwinder,
    RemoteMem,
)

BUSY_PY = textwrap.dedent("""
    import sys, time, threading
    def py_hot_leaf():
        deadline = time.time() + 20
        while time.time() < deadline:
            sum(range(3000))
    def py_mid():
        py_hot_leaf()
    def py_entry():
        py_mid()
    if "--thread" in sys.argv:
        t = threading.Thread(target=py_entry); t.start(); t.join()
    else:
        py_entry()
""")


# -- synthetic struct images ----------------------------------------------


class Image:
    """A fake remote address space: one buffer at BASE, reads outside
    the allocated prefix fail like process_vm_readv would."""

    BASE = 0x200000

    def __init__(self, size=1 << 20):
        self.buf = bytearray(size)
        self.used = 0

    def alloc(self, size, align=512):
        self.used = (self.used + align - 1) // align * align
        addr = self.BASE + self.used
        self.used += size
        return addr

    def w64(self, addr, value):
        struct.pack_into("<Q", self.buf, addr - self.BASE, value)

    def w32(self, addr, value):
        struct.pack_into("<I", self.buf, addr - self.BASE, value)

    def wbytes(self, addr, data):
        self.buf[addr - self.BASE:addr - self.BASE + len(data)] = data

    def read(self, addr, size):
        off = addr - self.BASE
        if off < 0 or off + size > self.used:
            raise OSError("fault")
        return bytes(self.buf[off:off + size])


def build_image(era):
    """era: '3.8' | '3.11' | '3.12' | '3.13'. Returns (mem, anchors,
    tids, expected leaf-first names)."""
    img = Image()
    code_type = img.alloc(64)
    unicode_type = img.alloc(64)
    frame_type = img.alloc(64)
    ascii_off = 48 if era in ("3.8", "3.11") else 40

    def mk_str(s):
        addr = img.alloc(ascii_off + len(s) + 1)
        img.w64(addr + 8, unicode_type)
        img.w64(addr + 16, len(s))
        img.w32(addr + 32, (1 << 5) | (1 << 6) | (1 << 2))  # compact|ascii
        img.wbytes(addr + ascii_off, s.encode() + b"\x00")
        return addr

    fname = mk_str("/app/server.py")

    def mk_code(name, qualname=None):
        # unicode fields at era-plausible offsets: filename, name
        # (3.11+: +qualname). Other words left zero / non-pointer.
        addr = img.alloc(384)
        img.w64(addr + 8, code_type)
        img.w64(addr + 96, fname)          # co_filename
        img.w64(addr + 104, mk_str(name))  # co_name
        if era != "3.8" and qualname:
            img.w64(addr + 112, mk_str(qualname))
        return addr

    names = ["handler", "dispatch", "main"]
    codes = [mk_code(n, f"Svc.{n}") for n in names]

    if era == "3.8":
        # PyFrameObject: refcnt, type, ob_size, f_back@24, f_code@32
        frames = [img.alloc(256) for _ in names]
        for i, f in enumerate(frames):
            img.w64(f + 8, frame_type)
            img.w64(f + 16, 12)  # ob_size
            img.w64(f + 24, frames[i + 1] if i + 1 < len(frames) else 0)
            img.w64(f + 32, codes[i])
        top = frames[0]
    else:
        # _PyInterpreterFrame per era
        if era == "3.11":
            code_off, prev_off = 32, 48
        else:
            code_off, prev_off = 0, 8
        frames = [img.alloc(256) for _ in names]
        for i, f in enumerate(frames):
            img.w64(f + code_off, codes[i])
            img.w64(f + prev_off,
                    frames[i + 1] if i + 1 < len(frames) else 0)
        top = frames[0]

    tids = [4242, 4243]
    interp = img.alloc(4096)
    tstates = [img.alloc(1024) for _ in tids]
    for idx, (ts, tid) in enumerate(zip(tstates, tids)):
        nxt = tstates[idx + 1] if idx + 1 < len(tstates) else 0
        img.w64(ts + 0, tstates[idx - 1] if idx else 0)  # prev
        img.w64(ts + 8, nxt)
        img.w64(ts + 16, interp)
        if era == "3.8":
            # no native_thread_id: thread_id -> glibc struct pthread
            pt = img.alloc(2048)
            img.w32(pt + 720, tid)
            img.w64(ts + 176, pt)  # thread_id = pthread_self()
        else:
            img.w64(ts + 152, tid)  # native_thread_id

    # Only the FIRST tstate has a running frame chain (like a sampled
    # process where other threads idle in C code).
    ts0 = tstates[0]
    if era == "3.8":
        img.w64(ts0 + 88, top)  # tstate->frame
    elif era == "3.13":
        img.w64(ts0 + 72, top)  # tstate->current_frame (direct)
    else:
        cframe = img.alloc(64)
        cur_off = 8 if era == "3.11" else 0
        img.w64(cframe + cur_off, top)
        img.w64(ts0 + 64, cframe)  # tstate->cframe

    # interp: decoy words, then threads.head
    img.w64(interp + 8, 7)  # id
    img.w64(interp + 24, tstates[0])

    runtime = img.alloc(4096)
    img.w64(runtime + 16, 0xDEAD)          # non-pointer decoy
    img.w64(runtime + 40, code_type)       # pointer decoy (fails probe)
    img.w64(runtime + 56, interp)          # interpreters.head

    mem = RemoteMem(img.read)
    anchors = Anchors(runtime=runtime, code_type=code_type,
                      unicode_type=unicode_type,
                      frame_type=frame_type if era == "3.8" else 0)
    return mem, anchors, tids, names


@pytest.mark.parametrize("era", ["3.8", "3.11", "3.12", "3.13"])
def test_synthetic_era_calibration_and_walk(era):
    mem, anchors, tids, names = build_image(era)
    off = Calibrator(mem, anchors, tids).run()
    assert off is not None, f"calibration failed for {era} layout"
    assert off.complete()
    if era == "3.8":
        assert off.frame_kind == "pyframe"
        assert off.tstate_pthread >= 0 and off.pthread_tid == 720
    else:
        assert off.frame_kind == "iframe"
        assert off.tstate_native_tid == 152
    if era == "3.11":
        assert off.cframe_indirect == 8
    if era == "3.12":
        assert off.cframe_indirect == 0
    if era == "3.13":
        assert off.cframe_indirect == -1
    assert off.unicode_ascii_data == (48 if era in ("3.8", "3.11") else 40)

    u = PythonUnwinder()
    info = PyProcess(pid=1234, runtime_addr=anchors.runtime,
                     offsets=off, mem=mem)
    u._procs.put(1234, info)
    frames = u.stack_for(1234, tids[0])
    got = [f.function_name for f in frames]
    if era == "3.8":
        assert got == names
    else:
        assert got == [f"Svc.{n}" for n in names]  # qualname preferred
    assert frames[0].source_file == "/app/server.py"
    # the second thread has no frame chain
    assert u.stack_for(1234, tids[1]) == []
    # unknown tid resolves to nothing, never to a wrong thread
    assert u.stack_for(1234, 9999) == []


# -- live foreign-process tests -------------------------------------------


@pytest.fixture
def busy_child(tmp_path):
    script = tmp_path / "busy.py"
    script.write_text(BUSY_PY)
    proc = subprocess.Popen([sys.executable, str(script)])
    time.sleep(0.8)
    yield proc
    proc.kill()
    proc.wait()


def test_remote_calibration_on_live_process(busy_child):
    u = PythonUnwinder()
    assert u.available
    frames = u.stack_for(busy_child.pid, busy_child.pid, 0)
    names = [f.function_name for f in frames]
    assert names[:3] == ["py_hot_leaf", "py_mid", "py_entry"], names
    assert frames[0].source_file.endswith("busy.py")
    assert u.calibrations == 1
    assert u.stacks_resolved >= 1


def test_offsets_cached_per_build(busy_child, tmp_path):
    """A second process of the same build must reuse the calibrated
    offsets (fleet images share builds)."""
    u = PythonUnwinder()
    assert u.stack_for(busy_child.pid, busy_child.pid, 0)
    script = tmp_path / "busy2.py"
    script.write_text(BUSY_PY)
    proc2 = subprocess.Popen([sys.executable, str(script)])
    try:
        time.sleep(0.8)
        frames = u.stack_for(proc2.pid, proc2.pid, 0)
        assert [f.function_name for f in frames][:1] == ["py_hot_leaf"]
        assert u.calibrations == 1  # no second calibration
    finally:
        proc2.kill()
        proc2.wait()


def test_non_python_process_skipped(tmp_path):
    u = PythonUnwinder()
    proc = subprocess.Popen(["sleep", "5"])
    try:
        time.sleep(0.2)
        assert u.stack_for(proc.pid, proc.pid, 0) == []
    finally:
        proc.kill()


def _perf_available():
    try:
        from parca_agent_amd.native import sampler
        s = sampler().PerfSampler(freq=1, track_mmaps=False)
        s.start()
        s.stop()
        return True
    except Exception:
        return False


@pytest.mark.skipif(not _perf_available(), reason="perf unavailable")
def test_python_frames_in_cpu_profile(busy_child):
    from parca_agent_amd.cpu import CPUSamplerService
    from parca_agent_amd.model import FrameType
    from parca_agent_amd.reporter import Reporter

    class Dest:
        def __init__(self):
            self.samples = []

        def write_batch(self, batch):
            self.samples.extend(batch)

        def close(self):
            pass

    dest = Dest()
    rep = Reporter([dest], cpu_sampling_frequency=97)
    svc = CPUSamplerService(rep, freq=97, dwarf_stacks=True,
                            poll_interval=0.05,
                            python_unwinder=PythonUnwinder())
    svc.start()
    time.sleep(2.5)
    svc.stop()
    rep.flush()

    child_samples = [
        s for s in dest.samples
        if s.labels.get("thread_id") == str(busy_child.pid)
    ]
    assert child_samples, f"no samples of child (total {len(dest.samples)})"
    with_py = [
        s for s in child_samples
        if any(f.kind == FrameType.PYTHON for f in s.trace.frames)
    ]
    assert len(with_py) > len(child_samples) // 2, (
        f"python frames in {len(with_py)}/{len(child_samples)}")
    names = [f.function_name for f in with_py[0].trace.frames
             if f.kind == FrameType.PYTHON]
    assert "py_hot_leaf" in names
    assert svc.python_stacks > 0


def test_remote_stack_worker_thread(tmp_path):
    """Exact tid->tstate matching on a 3.10 build (no native_thread_id)
    via the target-calibrated glibc pthread tid offset."""
    script = tmp_path / "busy_t.py"
    script.write_text(BUSY_PY)
    proc = subprocess.Popen([sys.executable, str(script), "--thread"])
    try:
        time.sleep(0.8)
        u = PythonUnwinder()
        tids = [int(t) for t in os.listdir(f"/proc/{proc.pid}/task")]
        worker_tids = [t for t in tids if t != proc.pid]
        assert worker_tids
        resolved = {}
        for tid in worker_tids:
            frames = u.stack_for(proc.pid, tid, 0)
            if frames:
                resolved[tid] = [f.function_name for f in frames]
        assert any("py_hot_leaf" in names for names in resolved.values()), \
            resolved
    finally:
        proc.kill()
        proc.wait()


def test_remote_mem_partial_and_faults():
    """RemoteMem: read() is all-or-nothing, read_some() returns the
    page-bounded prefix when a scan runs off the mapping."""
    img = Image()
    base = img.alloc(8192, align=4096)
    img.wbytes(base, b"Z" * 8192)
    mem = RemoteMem(img.read)
    assert mem.read(base, 16) == b"Z" * 16
    assert mem.read(0, 8) is None          # implausible address
    end = Image.BASE + img.used
    assert mem.read(end - 8, 16) is None   # crosses the fault boundary
    got = mem.read_some(end - 100, 4096)
    assert 0 < len(got) <= 4096            # prefix up to the boundary
    assert got == b"\x00" * len(got) or got  # content from the image


def test_jitdump_partial_record_retry(tmp_path):
    import struct

    from parca_agent_amd.interp.perfmap import _JitDump

    path = tmp_path / "jit-5.dump"
    header = struct.pack("<IIIIIIQQ", 0x4A695444, 1, 40, 62, 0, 5, 0, 0)
    body = struct.pack("<IIQQQQ", 5, 5, 0x1000_0000, 0x1000_0000, 0x40,
                       0) + b"fn_a\x00"
    rec = struct.pack("<IIQ", 0, 16 + len(body), 1) + body
    with open(path, "wb") as fh:
        fh.write(header)
        fh.write(rec[: len(rec) // 2])  # torn write
    jd = _JitDump(str(path))
    jd.refresh()
    assert jd.entries == []  # partial record not consumed
    with open(path, "ab") as fh:
        fh.write(rec[len(rec) // 2:])
    jd.refresh()
    assert jd.lookup(0x1000_0010) == "fn_a"
