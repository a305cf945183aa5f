"""Off-CPU profiling end-to-end: a sleeping process must produce
wallclock samples whose values reflect the blocked durations."""

import subprocess
import textwrap
import time

import pytest

from parca_agent_amd.reporter import Reporter


def _perf_available():
    try:
        from parca_agent_amd.native import sampler
        s = sampler().OffCpuSampler(sample_period=1000)
        s.start()
        s.stop()
        return True
    except Exception:
        return False


pytestmark = pytest.mark.skipif(not _perf_available(),
                                reason="perf context-switch events unavailable")

SLEEPER_C = textwrap.dedent("""
    #include <time.h>
    int main(void) {
        struct timespec ts = {0, 20 * 1000 * 1000}; /* 20 ms */
        for (int i = 0; i < 50; i++) nanosleep(&ts, 0);
        return 0;
    }
""")


class Dest:
    def __init__(self):
        self.samples = []

    def write_batch(self, batch):
        self.samples.extend(batch)

    def close(self):
        pass


def test_offcpu_sleeper(tmp_path):
    from parca_agent_amd.cpu.offcpu import OffCPUService

    src = tmp_path / "sleeper.c"
    src.write_text(SLEEPER_C)
    binary = tmp_path / "sleeper"
    subprocess.run(["gcc", "-O1", "-fno-omit-frame-pointer", str(src),
                    "-o", str(binary)], check=True)

    dest = Dest()
    rep = Reporter([dest])
    svc = OffCPUService(rep, threshold=1.0, poll_interval=0.1)
    svc.start()
    try:
        proc = subprocess.Popen([str(binary)])
        proc.wait(timeout=30)
        assert proc.returncode == 0
        time.sleep(0.5)
    finally:
        svc.stop()
    rep.flush()

    # Match the sleeper by tid (the syscall-boundary user stack may show
    # only libc frames when glibc wrappers lack frame pointers).
    wall = [s for s in dest.samples
            if s.sample_type.sample_type == "wallclock"
            and s.labels.get("thread_id") == str(proc.pid)]
    assert wall, f"no off-cpu samples of sleeper (total {len(dest.samples)})"
    # Each nanosleep blocks ~20 ms.
    long_blocks = [s for s in wall if s.value > 10_000_000]
    assert long_blocks, [s.value for s in wall][:10]
    # Kernel frames of the sleep path captured.
    sample = long_blocks[0]
    from parca_agent_amd.model import FrameType

    assert any(f.kind == FrameType.KERNEL for f in sample.trace.frames)
