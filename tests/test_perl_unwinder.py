"""Perl interpreter unwinder tests: build-exact offset extraction,
remote context-stack reads, and end-to-end perl frames in CPU profiles
(reference capability: the fork's perl unwinder)."""

import os
import shutil
import subprocess
import textwrap
import time

import pytest

from parca_agent_amd.interp.perl import OFFSETS_JSON, PerlUnwinder

perf = pytest.mark.skipif(
    not os.access("/proc/sys/kernel/perf_event_paranoid", os.R_OK),
    reason="no perf_event support")

requires_perl = pytest.mark.skipif(
    shutil.which("perl") is None or not os.path.exists(OFFSETS_JSON),
    reason="perl or extracted offsets unavailable "
           "(run build_native.py --only perl)")

BUSY_PL = textwrap.dedent("""
    sub pl_hot_leaf { my $x = 0; $x += $_ for 1 .. 5000; return $x }
    sub pl_mid { my $t = 0; $t += pl_hot_leaf() for 1 .. 20; return $t }
    sub pl_entry { pl_mid() }
    my $deadline = time() + 20;
    pl_entry() while time() < $deadline;
""")


@pytest.fixture
def busy_perl(tmp_path):
    script = tmp_path / "busy.pl"
    script.write_text(BUSY_PL)
    proc = subprocess.Popen(["perl", str(script)])
    time.sleep(0.5)
    yield proc
    proc.kill()
    proc.wait()


@requires_perl
def test_offsets_extracted():
    u = PerlUnwinder()
    assert u.available
    off = u.offsets
    # Stable structural facts of any 64-bit perl build.
    assert off["cx_size"] > 0 and off["cx_size"] % 8 == 0
    assert off["cxt_sub"] != off["cxt_eval"]
    assert off["hek_key"] > off["hek_len"]
    assert u.build_file_id


@requires_perl
def test_remote_stack(busy_perl):
    u = PerlUnwinder()
    names = []
    # The leaf sub runs ~100% of wall time; a few attempts tolerate
    # racy reads mid-callframe-push.
    for _ in range(20):
        frames = u.stack_for(busy_perl.pid, busy_perl.pid)
        names = [f.function_name for f in frames]
        if names[:3] == ["pl_hot_leaf", "pl_mid", "pl_entry"]:
            break
        time.sleep(0.02)
    assert names[:3] == ["pl_hot_leaf", "pl_mid", "pl_entry"], names
    assert names[-1] == "main::"
    leaf = u.stack_for(busy_perl.pid, busy_perl.pid)[0]
    assert leaf.source_file.endswith("busy.pl")
    assert leaf.source_line >= 1


@requires_perl
def test_non_perl_process_skipped():
    u = PerlUnwinder()
    assert u.stack_for(os.getpid(), os.getpid()) == []
    info = u._process(os.getpid())
    assert info is not None and not info.usable


@perf
@requires_perl
def test_perl_frames_in_profile(busy_perl):
    """End to end: perl frames appear in CPU-sampled traces."""
    from parca_agent_amd.cpu import CPUSamplerService
    from parca_agent_amd.model import FrameType
    from parca_agent_amd.reporter.reporter import Reporter

    class Dest:
        def __init__(self):
            self.samples = []

        def write_batch(self, batch):
            self.samples.extend(batch)

        def close(self):
            pass

    dest = Dest()
    rep = Reporter([dest], cpu_sampling_frequency=97)
    svc = CPUSamplerService(rep, freq=97, poll_interval=0.05)
    svc.perl_unwinder = PerlUnwinder()
    assert svc.perl_unwinder.available
    svc.start()
    time.sleep(2.0)
    svc.stop()
    rep.flush()

    perl_names = set()
    for s in dest.samples:
        for f in s.trace.frames:
            if f.kind == FrameType.PERL:
                perl_names.add(f.function_name)
    assert "pl_hot_leaf" in perl_names, (
        f"perl frames missing; saw {sorted(perl_names)[:10]}, "
        f"perl_stacks={svc.perl_stacks}")
    assert svc.perl_stacks > 0
