"""perf-map JIT symbolization tests: map parsing plus a live node
--perf-basic-prof profile (node is present in this image)."""

import os
import shutil
import subprocess
import sys
import time

import pytest

from parca_agent_amd.interp.perfmap import PerfMapResolver, _PerfMap


def test_perf_map_parse_and_lookup(tmp_path):
    p = tmp_path / "perf-1234.map"
    p.write_text(
        "7f0000001000 40 LazyCompile:*hot fn.js:1\n"
        "7f0000002000 100 JS:^warm fn.js:9\n"
        "garbage line\n"
        "7f0000001000 60 LazyCompile:*hot-retier fn.js:1\n")
    pm = _PerfMap(path=str(p))
    pm.refresh()
    assert pm.lookup(0x7F0000001010) in ("LazyCompile:*hot fn.js:1",
                                         "LazyCompile:*hot-retier fn.js:1")
    assert pm.lookup(0x7F0000002050) == "JS:^warm fn.js:9"
    assert pm.lookup(0x7F0000009999) is None
    # Incremental growth: only the tail is reparsed.
    with open(p, "a") as fh:
        fh.write("7f0000003000 20 JS:later x.js:3\n")
    pm.refresh()
    assert pm.lookup(0x7F0000003004) == "JS:later x.js:3"


NODE_BUSY = """
function jsHotLoop() {
  let acc = 0;
  for (let i = 0; i < 5e7; i++) acc += i % 7;
  return acc;
}
let t0 = Date.now();
let out = 0;
while (Date.now() - t0 < 3000) out += jsHotLoop();
console.log(out > 0 ? "done" : "odd");
"""


def _perf_available():
    try:
        from parca_agent_amd.native import sampler
        s = sampler().PerfSampler(freq=1, track_mmaps=False)
        s.start()
        s.stop()
        return True
    except Exception:
        return False


@pytest.mark.skipif(shutil.which("node") is None, reason="no node")
@pytest.mark.skipif(not _perf_available(), reason="perf unavailable")
def test_node_jit_frames(tmp_path):
    from parca_agent_amd.cpu import CPUSamplerService
    from parca_agent_amd.model import FrameType
    from parca_agent_amd.reporter import Reporter

    script = tmp_path / "busy.js"
    script.write_text(NODE_BUSY)

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
    svc.start()
    proc = subprocess.Popen(
        ["node", "--perf-basic-prof", str(script)],
        stdout=subprocess.PIPE)
    proc.wait(timeout=60)
    time.sleep(0.3)
    svc.stop()
    rep.flush()
    # Clean the map file node leaves behind.
    try:
        os.unlink(f"/tmp/perf-{proc.pid}.map")
    except OSError:
        pass

    node_samples = [s for s in dest.samples
                    if s.labels.get("thread_id") == str(proc.pid)]
    jit = [f for s in node_samples for f in s.trace.frames
           if f.kind == FrameType.JIT]
    assert jit, (f"no JIT frames; node samples={len(node_samples)}, "
                 f"resolved={svc.perf_maps.symbols_resolved}")
    names = {f.function_name for f in jit}
    assert any("jsHotLoop" in n for n in names), sorted(names)[:10]


@pytest.mark.skipif(shutil.which("node") is None, reason="no node")
@pytest.mark.skipif(not _perf_available(), reason="perf unavailable")
def test_node_interpreted_bytecode_frames(tmp_path):
    """V8 BYTECODE stacks: with
    --interpreted-frames-native-stack V8 gives every interpreted
    function its own copy of the Ignition entry trampoline, so the
    native stack + perf map resolve interpreted (not just jitted)
    frames — sampled here live with TurboFan disabled so the hot
    function stays bytecode-only."""
    from parca_agent_amd.cpu import CPUSamplerService
    from parca_agent_amd.model import FrameType
    from parca_agent_amd.reporter import Reporter

    script = tmp_path / "interp.js"
    script.write_text(NODE_BUSY)

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
    svc.start()
    proc = subprocess.Popen(
        ["node", "--perf-basic-prof", "--interpreted-frames-native-stack",
         "--no-opt", str(script)],
        stdout=subprocess.PIPE)
    proc.wait(timeout=60)
    time.sleep(0.3)
    svc.stop()
    rep.flush()
    try:
        os.unlink(f"/tmp/perf-{proc.pid}.map")
    except OSError:
        pass

    node_samples = [s for s in dest.samples
                    if s.labels.get("thread_id") == str(proc.pid)]
    jit_names = {f.function_name for s in node_samples
                 for f in s.trace.frames if f.kind == FrameType.JIT}
    interpreted = [n for n in jit_names
                   if n.startswith("InterpretedFunction:")]
    assert any("jsHotLoop" in n for n in interpreted), \
        (len(node_samples), sorted(jit_names)[:10])


def test_jitdump_parse_and_lookup(tmp_path):
    """perf jitdump convention (Julia / LLVM-JIT runtimes): binary
    JIT_CODE_LOAD records resolve like perf-map lines, incl.
    incremental refresh of appended records."""
    import struct

    from parca_agent_amd.interp.perfmap import _JitDump

    path = tmp_path / "jit-77.dump"

    def code_load(addr, size, name, ts=1):
        body = struct.pack("<IIQQQQ", 77, 77, addr, addr, size, 0) + \
            name.encode() + b"\x00"
        rec = struct.pack("<IIQ", 0, 16 + len(body), ts) + body
        return rec

    header = struct.pack("<IIIIIIQQ", 0x4A695444, 1, 40, 62, 0, 77, 0, 0)
    with open(path, "wb") as fh:
        fh.write(header)
        fh.write(code_load(0x7F10_0000_1000, 0x80, "julia_hot_kernel"))
        fh.write(struct.pack("<IIQ", 3, 16, 2))  # unrelated record type
        fh.write(code_load(0x7F10_0000_2000, 0x40, "jl_apply_generic2"))

    jd = _JitDump(str(path))
    jd.refresh()
    assert jd.lookup(0x7F10_0000_1010) == "julia_hot_kernel"
    assert jd.lookup(0x7F10_0000_2004) == "jl_apply_generic2"
    assert jd.lookup(0x7F10_0000_3000) is None

    # appended records picked up incrementally
    with open(path, "ab") as fh:
        fh.write(code_load(0x7F10_0000_4000, 0x20, "late_fn"))
    jd.refresh()
    assert jd.lookup(0x7F10_0000_4008) == "late_fn"

    # a non-jitdump file is rejected quietly
    bad = tmp_path / "jit-78.dump"
    bad.write_bytes(b"not a jitdump file at all" * 4)
    jd2 = _JitDump(str(bad))
    jd2.refresh()
    assert jd2.entries == []


def test_resolver_finds_jitdump_file():
    """PerfMapResolver falls back to the jitdump convention when no
    perf map exists for the pid."""
    import struct as _struct

    from parca_agent_amd.interp.perfmap import PerfMapResolver

    pid = os.getpid()
    path = f"/tmp/jit-{pid}.dump"
    assert not os.path.exists(f"/tmp/perf-{pid}.map")
    body = _struct.pack("<IIQQQQ", pid, pid, 0x7F2000001000,
                        0x7F2000001000, 0x100, 0) + b"my_jitted_fn\x00"
    rec = _struct.pack("<IIQ", 0, 16 + len(body), 1) + body
    header = _struct.pack("<IIIIIIQQ", 0x4A695444, 1, 40, 62, 0, pid, 0, 0)
    with open(path, "wb") as fh:
        fh.write(header + rec)
    try:
        r = PerfMapResolver()
        assert r.lookup(pid, 0x7F2000001010) == "my_jitted_fn"
        assert r.lookup(pid, 0x7F2000009999) is None
    finally:
        os.unlink(path)
