"""Race detection for the native SPSC ring (the Go-race-detector analog,
SURVEY.md §5.2): builds csrc/tests/ring_stress.cc with
-fsanitize=thread and runs producer/consumer concurrently; TSAN failures
or data corruption fail the test."""

import os
import subprocess

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture(scope="module")
def stress_binaries(tmp_path_factory):
    d = tmp_path_factory.mktemp("ringstress")
    out = {}
    for name, flags in [("tsan", ["-fsanitize=thread"]),
                        ("plain", ["-O2"])]:
        binary = d / f"ring_stress_{name}"
        subprocess.run(
            ["g++", "-std=c++17", "-g", "-pthread", f"-I{REPO}/csrc",
             *flags, f"{REPO}/csrc/tests/ring_stress.cc", "-o", str(binary)],
            check=True)
        out[name] = str(binary)
    return out


def test_ring_stress_plain(stress_binaries):
    r = subprocess.run([stress_binaries["plain"]], capture_output=True,
                       text=True, timeout=120)
    assert r.returncode == 0, r.stdout + r.stderr
    assert "OK" in r.stdout


def test_ring_stress_tsan(stress_binaries):
    env = dict(os.environ)
    env["TSAN_OPTIONS"] = "halt_on_error=1 exitcode=66"
    r = subprocess.run([stress_binaries["tsan"]], capture_output=True,
                       text=True, timeout=300, env=env)
    assert "WARNING: ThreadSanitizer" not in r.stderr, r.stderr[-3000:]
    assert r.returncode == 0, r.stdout + r.stderr[-2000:]
