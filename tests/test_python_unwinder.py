"""CPython interpreter unwinder tests: offset self-calibration, remote
stack reads, and end-to-end python frames in CPU profiles (reference
capability: the fork's python interpreter unwinder)."""

import subprocess
import sys
import textwrap
import time

import pytest

from parca_agent_amd.interp.python import PythonUnwinder, calibrate

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


def test_calibration_complete():
    off = calibrate()
    assert off is not None, "calibration failed on own interpreter"
    assert off.complete()
    # sanity on a few stable facts
    assert off.unicode_data > off.unicode_length
    assert off.frame_code != off.frame_back


@pytest.fixture
def busy_child(tmp_path):
    script = tmp_path / "busy.py"
    script.write_text(BUSY_PY)
    proc = subprocess.Popen([sys.executable, str(script)])
    time.sleep(0.8)
    yield proc
    proc.kill()
    proc.wait()


def test_remote_stack_main_thread(busy_child):
    u = PythonUnwinder()
    assert u.available
    frames = u.stack_for(busy_child.pid, busy_child.pid, 0)
    names = [f.function_name for f in frames]
    assert names[:3] == ["py_hot_leaf", "py_mid", "py_entry"], names
    assert frames[0].source_file.endswith("busy.py")
    assert u.stacks_resolved >= 1


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
    """Exact tid->tstate matching on 3.10 via the calibrated glibc
    pthread tid offset: resolve the stack of a non-main thread."""
    import os

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
