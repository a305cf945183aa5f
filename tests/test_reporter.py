"""Reporter core tests — the reference pattern: bare reporter with fake
providers, no network (reference: reporter/parca_reporter_test.go:44-63)."""

import time

import pyarrow as pa
import pytest

from parca_agent_amd.model import (
    Frame,
    FrameType,
    MappingFile,
    Trace,
    TraceEventMeta,
    TraceOrigin,
)
from parca_agent_amd.relabel import RelabelConfig
from parca_agent_amd.reporter import (
    LocalStoreDestination,
    OfflineLogDestination,
    Reporter,
    build_arrow_record,
    read_offline_log,
    samples_to_pprof,
)
from parca_agent_amd.pprof import decode_profile

APP = MappingFile(file_id="a" * 32, path="/usr/bin/app", build_id="bid")


class FakeProvider:
    name = "fake"

    def __init__(self, labels):
        self._labels = labels
        self.calls = 0

    def add_metadata(self, pid, labels):
        self.calls += 1
        labels.update(self._labels)
        labels["pid"] = str(pid)
        return True


class CollectingDestination:
    def __init__(self):
        self.batches = []

    def write_batch(self, samples):
        self.batches.append(samples)

    def close(self):
        pass


def _trace(*addrs):
    return Trace(frames=tuple(
        Frame(kind=FrameType.NATIVE, address=a, mapping=APP) for a in addrs))


def _meta(pid=100, origin=TraceOrigin.SAMPLING, **kw):
    return TraceEventMeta(pid=pid, tid=pid, cpu=1, origin=origin,
                          timestamp_ns=12345, **kw)


def test_report_and_flush():
    dest = CollectingDestination()
    rep = Reporter([dest], metadata_providers=[FakeProvider({"node": "n1"})])
    rep.report_trace_event(_trace(1, 2), _meta())
    rep.report_trace_event(_trace(1, 2), _meta())
    n = rep.flush()
    assert n == 2
    assert len(dest.batches) == 1
    batch = dest.batches[0]
    assert batch[0].labels["node"] == "n1"
    assert batch[0].labels["pid"] == "100"
    assert batch[0].labels["cpu"] == "1"
    assert batch[0].sample_type.sample_type == "samples"
    assert batch[0].period == int(1e9 / 19)
    # Identical stack+labels rows aggregate within the flush window.
    assert len(batch) == 1
    assert batch[0].value == 2


def test_label_cache_amortizes_providers():
    prov = FakeProvider({"node": "n1"})
    rep = Reporter([CollectingDestination()], metadata_providers=[prov])
    for _ in range(10):
        rep.report_trace_event(_trace(1), _meta(pid=7))
    assert prov.calls == 1


def test_relabel_drop_cached():
    prov = FakeProvider({"comm": "secret"})
    cfgs = [RelabelConfig(source_labels=["comm"], regex="secret", action="drop")]
    rep = Reporter([CollectingDestination()], metadata_providers=[prov],
                   relabel_configs=cfgs)
    for _ in range(5):
        rep.report_trace_event(_trace(1), _meta(pid=8))
    assert rep.metrics.samples_dropped_relabel == 5
    assert prov.calls == 1  # drop decision cached
    assert rep.flush() == 0


def test_meta_labels_stripped():
    prov = FakeProvider({"__meta_internal": "x", "keep": "y"})
    dest = CollectingDestination()
    rep = Reporter([dest], metadata_providers=[prov])
    rep.report_trace_event(_trace(1), _meta())
    rep.flush()
    labels = dest.batches[0][0].labels
    assert "__meta_internal" not in labels
    assert labels["keep"] == "y"


def test_origin_sample_types():
    dest = CollectingDestination()
    rep = Reporter([dest])
    rep.set_gpu_config(100, 0, 2048.0)
    rep.report_trace_event(_trace(1), _meta(origin=TraceOrigin.SAMPLING))
    rep.report_trace_event(
        _trace(2), _meta(origin=TraceOrigin.GPU_KERNEL, gpu_id=0, value=5000))
    rep.report_trace_event(
        _trace(3), _meta(origin=TraceOrigin.GPU_PC, gpu_id=0, value=17))
    rep.flush()
    batch = dest.batches[0]
    by_type = {s.sample_type.sample_type: s for s in batch}
    assert by_type["samples"].sample_type.sample_unit == "count"
    assert by_type["gpu_kernel_time"].sample_type.sample_unit == "nanoseconds"
    assert by_type["gpu_pcsample"].period == 2048
    assert by_type["gpu_pcsample"].labels["gpu"] == "0"


def test_merge_gpu_profiles_mode():
    dest = CollectingDestination()
    rep = Reporter([dest], merge_gpu_profiles=True)
    rep.report_trace_event(
        _trace(1), _meta(origin=TraceOrigin.GPU_KERNEL, gpu_id=0))
    rep.report_trace_event(
        _trace(2), _meta(origin=TraceOrigin.GPU_PC, gpu_id=0))
    rep.flush()
    batch = dest.batches[0]
    assert all(s.sample_type.sample_type == "gpu_time" for s in batch)
    views = {s.labels["gpu_view"] for s in batch}
    assert views == {"kernel_time", "pc_sample"}


def test_thread_comm_patching():
    dest = CollectingDestination()
    prov = FakeProvider({"comm": "main-proc"})
    rep = Reporter([dest], metadata_providers=[prov])
    rep.report_trace_event(_trace(1), _meta(comm="worker-1"))
    rep.flush()
    labels = dest.batches[0][0].labels
    assert labels["comm"] == "main-proc"
    assert labels["thread_comm"] == "worker-1"


def test_arrow_record_from_batch():
    dest = CollectingDestination()
    rep = Reporter([dest], metadata_providers=[FakeProvider({"node": "n"})])
    for i in range(20):
        rep.report_trace_event(_trace(i % 3, 7), _meta(pid=i % 2 + 10))
    rep.flush()
    record = build_arrow_record(dest.batches[0])
    # 3 stacks x 2 pids aggregate to 6 rows summing to the 20 events.
    assert record.num_rows == 6
    assert record.column("sample_type").to_pylist() == ["samples"] * 6
    assert sum(record.column("value").to_pylist()) == 20


def test_local_store_destination(tmp_path):
    dest = LocalStoreDestination(str(tmp_path))
    rep = Reporter([dest])
    rep.report_trace_event(_trace(0x10, 0x20), _meta())
    rep.report_trace_event(_trace(0x30), _meta(origin=TraceOrigin.OFF_CPU,
                                               value=1000))
    rep.flush()
    files = sorted(p.name for p in tmp_path.iterdir())
    assert any(".samples." in f for f in files)
    assert any(".wallclock." in f for f in files)
    sample_file = next(p for p in tmp_path.iterdir() if ".samples." in p.name)
    prof = decode_profile(sample_file.read_bytes())
    assert prof.sample_types[0].type == "samples"
    assert len(prof.samples) == 1
    assert prof.samples[0]["labels"]["cpu"] == "1"


def test_offline_log_roundtrip(tmp_path):
    dest = OfflineLogDestination(str(tmp_path), rotation_interval=1e9)
    rep = Reporter([dest])
    rep.report_trace_event(_trace(1, 2), _meta())
    rep.flush()
    rep.report_trace_event(_trace(3), _meta())
    rep.flush()
    dest.close()
    files = [p for p in tmp_path.iterdir() if p.suffix == ".zst"]
    assert len(files) == 1
    payloads = read_offline_log(str(files[0]))
    assert len(payloads) == 2
    table = pa.ipc.open_stream(payloads[0]).read_all()
    assert table.num_rows == 1


def test_offline_log_torn_frame(tmp_path):
    dest = OfflineLogDestination(str(tmp_path), rotation_interval=1e9)
    rep = Reporter([dest])
    rep.report_trace_event(_trace(1), _meta())
    rep.flush()
    # Simulate a torn write: append garbage without patching the count.
    path = next(p for p in tmp_path.iterdir())
    with open(path, "ab") as fh:
        fh.write(b"\xff" * 32)
    payloads = read_offline_log(str(path))
    assert len(payloads) == 1  # the durable frame only


def test_flush_loop_thread():
    dest = CollectingDestination()
    rep = Reporter([dest], batch_write_interval=0.05)
    rep.start()
    rep.report_trace_event(_trace(1), _meta())
    deadline = time.time() + 3
    while not dest.batches and time.time() < deadline:
        time.sleep(0.02)
    rep.stop()
    assert dest.batches


def test_destination_error_does_not_lose_loop():
    class Exploding:
        def write_batch(self, samples):
            raise RuntimeError("boom")

        def close(self):
            pass

    ok = CollectingDestination()
    rep = Reporter([Exploding(), ok])
    rep.report_trace_event(_trace(1), _meta())
    rep.flush()
    assert rep.metrics.batch_errors == 1
    assert len(ok.batches) == 1


def test_label_sanitation():
    from parca_agent_amd.reporter.reporter import sanitize_label_value

    # multibyte truncation lands on a codepoint boundary
    s = "α" * 200  # 2 bytes each = 400 bytes
    out = sanitize_label_value(s, max_len=255)
    assert len(out.encode()) <= 255
    assert out == "α" * 127
    # 4-byte emoji boundary
    s = "x" * 253 + "😀"
    out = sanitize_label_value(s, max_len=255)
    assert out == "x" * 253
    # short values untouched
    assert sanitize_label_value("ok") == "ok"


def test_labels_sanitized_via_provider():
    class BadProvider:
        name = "bad"

        def add_metadata(self, pid, labels):
            labels["comm"] = "π" * 300
            return True

    dest = CollectingDestination()
    rep = Reporter([dest], metadata_providers=[BadProvider()])
    rep.report_trace_event(_trace(1), _meta())
    rep.flush()
    value = dest.batches[0][0].labels["comm"]
    assert len(value.encode()) <= 255


def test_flush_aggregation():
    dest = CollectingDestination()
    rep = Reporter([dest], aggregate_batches=True)
    t = _trace(1, 2)
    for i in range(10):
        rep.report_trace_event(t, _meta())
    rep.report_trace_event(_trace(9), _meta(origin=TraceOrigin.GPU_KERNEL,
                                            gpu_id=0, value=500))
    rep.report_trace_event(_trace(9), _meta(origin=TraceOrigin.GPU_KERNEL,
                                            gpu_id=0, value=700))
    n = rep.flush()
    assert n == 12  # pre-aggregation count returned
    batch = dest.batches[0]
    assert len(batch) == 2
    by_type = {s.sample_type.sample_type: s for s in batch}
    assert by_type["samples"].value == 10
    assert by_type["gpu_kernel_time"].value == 1200


def test_flush_no_aggregation_when_disabled():
    dest = CollectingDestination()
    rep = Reporter([dest], aggregate_batches=False)
    t = _trace(1)
    rep.report_trace_event(t, _meta())
    rep.report_trace_event(t, _meta())
    rep.flush()
    assert len(dest.batches[0]) == 2


def test_aggregation_distinct_labels_not_merged():
    dest = CollectingDestination()
    rep = Reporter([dest], aggregate_batches=True)
    t = _trace(1)
    m1 = _meta()
    m2 = _meta()
    m2.cpu = 5  # different cpu label
    rep.report_trace_event(t, m1)
    rep.report_trace_event(t, m2)
    rep.flush()
    assert len(dest.batches[0]) == 2


def test_local_store_symbolizes_native_frames(tmp_path):
    """Local-store pprof must carry symtab names for native frames —
    there is no server-side symbolizer in that mode."""
    import subprocess

    from parca_agent_amd.elf import ELFFile
    from parca_agent_amd.model import (
        Frame,
        FrameType,
        MappingFile,
        Trace,
        TraceEventMeta,
    )
    from parca_agent_amd.pprof.profile import decode_profile
    from parca_agent_amd.reporter.reporter import Reporter

    src = tmp_path / "s.c"
    src.write_text("void named_leaf(void){}\n"
                   "int main(void){named_leaf();return 0;}\n")
    exe = tmp_path / "s"
    subprocess.run(["gcc", "-O0", str(src), "-o", str(exe)], check=True)
    with ELFFile.open(str(exe)) as elf:
        sym = next(s for s in elf.symbols() if s.name == "named_leaf")

    dest = LocalStoreDestination(str(tmp_path / "store"), symbolize=True)
    rep = Reporter([dest], cpu_sampling_frequency=19)
    trace = Trace(frames=(Frame(
        kind=FrameType.NATIVE, address=sym.value + 1,
        mapping=MappingFile(path=str(exe), file_id="ab" * 16)),))
    rep.report_trace_event(trace, TraceEventMeta(
        timestamp_ns=1, pid=1, tid=1, value=1))
    rep.flush()  # first flush queues the index build (async)
    dest.symbolizer.wait_idle()
    rep.report_trace_event(trace, TraceEventMeta(
        timestamp_ns=2, pid=1, tid=1, value=1))
    rep.flush()  # second flush has the names
    files = sorted((tmp_path / "store").glob("*.samples.pb.gz"))[-1:]
    assert files
    prof = decode_profile(files[0].read_bytes())
    names = set()
    for s in prof.samples:
        names.update(prof.stack_names(s))
    assert "named_leaf" in names


def test_merged_gpu_pc_value_scaled_to_ns():
    """In merged mode PC-sample COUNTS must be scaled by ns_per_sample so
    they sum meaningfully with kernel-time nanoseconds (reference
    value *= nsPerSample, parca_reporter.go TraceOriginGpuPC)."""
    dest = CollectingDestination()
    rep = Reporter([dest], merge_gpu_profiles=True)
    rep.set_gpu_config(100, 0, 2048.0)
    rep.report_trace_event(
        _trace(1), _meta(origin=TraceOrigin.GPU_PC, gpu_id=0, value=17))
    # Unknown (pid,gpu) falls back to (pid,-1); no config at all => raw.
    rep.report_trace_event(
        _trace(2), _meta(origin=TraceOrigin.GPU_PC, gpu_id=3, value=5))
    rep.flush()
    by_addr = {s.trace.frames[0].address: s for s in dest.batches[0]}
    assert by_addr[1].value == 17 * 2048
    assert by_addr[2].value == 5  # no ns-per-sample known: unscaled

    # Unmerged mode keeps raw counts (unit is count, period carries ns).
    dest2 = CollectingDestination()
    rep2 = Reporter([dest2])
    rep2.set_gpu_config(100, 0, 2048.0)
    rep2.report_trace_event(
        _trace(1), _meta(origin=TraceOrigin.GPU_PC, gpu_id=0, value=17))
    rep2.flush()
    assert dest2.batches[0][0].value == 17


def test_offcpu_period_type_matches_reference():
    from parca_agent_amd.model import sample_type_for
    st = sample_type_for(TraceOrigin.OFF_CPU)
    assert (st.sample_type, st.sample_unit) == ("wallclock", "nanoseconds")
    assert (st.period_type, st.period_unit) == ("samples", "count")


def test_samples_to_pprof_distinct_periods_not_lost():
    """Two processes PC-sampling at different intervals share the
    gpu_pcsample type but differ in period: both groups must survive."""
    dest = CollectingDestination()
    rep = Reporter([dest])
    rep.set_gpu_config(100, 0, 2048.0)
    rep.set_gpu_config(200, 0, 4096.0)
    rep.report_trace_event(
        _trace(1), _meta(pid=100, origin=TraceOrigin.GPU_PC, gpu_id=0,
                         value=3))
    rep.report_trace_event(
        _trace(2), _meta(pid=200, origin=TraceOrigin.GPU_PC, gpu_id=0,
                         value=7))
    rep.flush()
    profiles = samples_to_pprof(dest.batches[0])
    assert len(profiles) == 2
    periods = set()
    total = 0
    for data in profiles.values():
        prof = decode_profile(data)
        periods.add(prof.period)
        total += sum(s["values"][0] for s in prof.samples)
    assert periods == {2048, 4096}
    assert total == 10


def test_metadata_disable_caching_requeries_providers():
    """--metadata-disable-caching (label_ttl_seconds=0): providers are
    consulted on every sample instead of LRU-cached."""
    import time as _t

    prov = FakeProvider({"node": "n1"})
    rep = Reporter([CollectingDestination()], metadata_providers=[prov],
                   label_ttl_seconds=0.0)
    rep.report_trace_event(_trace(1), _meta())
    _t.sleep(0.002)
    rep.report_trace_event(_trace(1), _meta())
    assert prov.calls == 2


def test_custom_labels_sanitized():
    """Per-sample custom labels get the reference's write-time pass
    (parca_reporter.go:362-374): non-UTF8 keys dropped, values clamped
    with multi-byte-safe truncation repair."""
    from parca_agent_amd.reporter.reporter import MAX_LABEL_VALUE_LEN

    dest = CollectingDestination()
    rep = Reporter([dest])
    bad_key = "k\udcff"          # lone surrogate: not encodable
    long_val = "é" * 300          # 600 UTF-8 bytes
    tr = Trace(
        frames=(Frame(kind=FrameType.NATIVE, address=1, mapping=APP),),
        custom_labels=(("probe", "myprobe"), (bad_key, "x"), ("", "y"),
                       ("long", long_val)))
    rep.report_trace_event(tr, _meta())
    rep.report_trace_event(tr, _meta())  # hits the memoized path
    rep.flush()
    samples = [s for b in dest.batches for s in b]
    assert len(samples) >= 1  # identical events may aggregate
    for s in samples:
        labels = dict(s.labels)
        assert labels["probe"] == "myprobe"
        assert bad_key not in labels and "" not in labels
        enc = labels["long"].encode("utf-8")
        assert len(enc) <= MAX_LABEL_VALUE_LEN
        enc.decode("utf-8")  # never split mid-codepoint


def test_probe_per_sample_relabel():
    """Probe-origin samples run the relabel pass per sample with
    per-sample fields visible (thread_id, probe); other origins keep
    the patch-and-ship path (parca_reporter.go:805-841)."""
    from parca_agent_amd.model import TraceOrigin
    from parca_agent_amd.relabel import RelabelConfig

    dest = CollectingDestination()
    rep = Reporter([dest], relabel_configs=[
        RelabelConfig(action="drop", source_labels=["probe"],
                      regex="noisy_.*"),
    ])
    frames = (Frame(kind=FrameType.NATIVE, address=1, mapping=APP),)
    noisy = Trace(frames=frames, custom_labels=(("probe", "noisy_poll"),))
    quiet = Trace(frames=frames, custom_labels=(("probe", "checkout"),))
    rep.report_trace_event(noisy, _meta(origin=TraceOrigin.PROBE))
    rep.report_trace_event(quiet, _meta(origin=TraceOrigin.PROBE))
    # Same label on a sampling-origin event: NOT dropped (gated pass).
    rep.report_trace_event(noisy, _meta(origin=TraceOrigin.SAMPLING))
    rep.flush()
    samples = [s for b in dest.batches for s in b]
    probes = [s for s in samples if s.labels.get("probe") == "checkout"]
    noisies = [s for s in samples if s.labels.get("probe") == "noisy_poll"]
    assert len(probes) == 1
    assert len(noisies) == 1  # only the sampling-origin copy survives
    assert rep.metrics.samples_dropped_relabel == 1
