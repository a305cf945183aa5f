"""Arrow v1 writer tests (reference: reporter/arrow.go schema — REE +
dictionary columns, two-record Write protocol)."""

import io

import pyarrow as pa

from parca_agent_amd.model import Frame, FrameType, MappingFile, Trace
from parca_agent_amd.reporter.arrow_v1 import (
    LocationsWriterV1,
    SampleWriterV1,
    decode_requested_ids,
)

APP = MappingFile(file_id="a" * 32, path="/usr/bin/app", build_id="bid")


def _trace(*addrs, named=False):
    return Trace(frames=tuple(
        Frame(kind=FrameType.NATIVE, address=a, mapping=APP,
              function_name=f"fn_{a:x}" if named else "")
        for a in addrs))


def _write(w, trace, labels=None, value=1, ts=100):
    w.append_sample(trace, labels or {}, value, ts,
                    sample_type="samples", sample_unit="count",
                    period_type="cpu", period_unit="nanoseconds",
                    period=52_631_578)


def test_v1_samples_record_schema():
    w = SampleWriterV1()
    _write(w, _trace(1, 2), labels={"node": "n1"})
    _write(w, _trace(1, 2), labels={"node": "n1"})
    batch = w.build_record()
    assert batch.schema.metadata[b"parca_write_schema_version"] == b"v1"
    # labels.* first, then the 11 fixed columns in reference order.
    assert batch.schema.names == [
        "labels.node", "stacktrace_id", "value", "producer", "sample_type",
        "sample_unit", "period_type", "period_unit", "temporality",
        "period", "duration", "timestamp"]
    assert batch.num_rows == 2
    # REE-dict types
    assert pa.types.is_run_end_encoded(batch.column("sample_type").type)
    assert pa.types.is_run_end_encoded(batch.column("labels.node").type)
    assert batch.column("value").to_pylist() == [1, 1]


def test_v1_stacktrace_ids_stable():
    w = SampleWriterV1()
    t = _trace(1, 2)
    _write(w, t)
    traces = w.traces()
    assert t.trace_hash() in traces


def test_locations_writer_roundtrip():
    t = _trace(0x10, 0x20, named=True)
    lw = LocationsWriterV1()
    lw.append_stacktrace(t.trace_hash(), t)
    t2 = _trace(0x99)  # unsymbolized
    lw.append_stacktrace(t2.trace_hash(), t2)
    batch = lw.build_record()
    assert batch.num_rows == 2
    locs = batch.column("locations")
    first = locs[0].as_py()
    assert len(first) == 2
    assert first[0]["address"] == 0x10
    assert first[0]["frame_type"] == "native"
    assert first[0]["mapping_build_id"] == "bid"
    assert first[0]["lines"][0]["function_name"] == "fn_10"
    second = locs[1].as_py()
    assert second[0]["lines"] is None  # unsymbolized -> null lines


def test_decode_requested_ids():
    ids = [b"a" * 16, b"b" * 16]
    batch = pa.record_batch([pa.array(ids, pa.binary())],
                            names=["stacktrace_id"])
    sink = io.BytesIO()
    with pa.ipc.new_stream(sink, batch.schema) as wr:
        wr.write_batch(batch)
    out = decode_requested_ids(sink.getvalue())
    assert out == set(ids)
