"""Arrow v2 writer tests — mirrors reference reporter/arrow_v2_test.go:
function/stacktrace dedup, ListView offsets, null-lines validity for
unsymbolized frames, mixed frame types."""

import io

import pyarrow as pa
import pytest

from parca_agent_amd.model import Frame, FrameType, MappingFile, Trace
from parca_agent_amd.reporter.arrow_v2 import (
    SCHEMA_VERSION_KEY,
    SCHEMA_VERSION_V2,
    SampleWriterV2,
    serialize_record,
)

LIBC = MappingFile(file_id="f" * 32, path="/lib/libc.so.6", build_id="bid1")
APP = MappingFile(file_id="a" * 32, path="/usr/bin/app", build_id="bid2")


def _trace(*addrs, mapping=APP, named=False):
    frames = tuple(
        Frame(kind=FrameType.NATIVE, address=a, mapping=mapping,
              function_name=f"fn_{a:x}" if named else "")
        for a in addrs
    )
    return Trace(frames=frames)


def _write_sample(w, trace, labels=None, value=1, ts=1000):
    w.append_sample(
        trace, labels or {}, value, ts,
        sample_type="samples", sample_unit="count",
        period_type="cpu", period_unit="nanoseconds", period=52_631_578,
    )


def test_schema_metadata_and_fields():
    w = SampleWriterV2()
    _write_sample(w, _trace(0x10, 0x20), labels={"node": "n1"})
    batch = w.build_record()
    assert batch.schema.metadata[SCHEMA_VERSION_KEY.encode()] == \
        SCHEMA_VERSION_V2.encode()
    assert batch.schema.names == [
        "labels", "stacktrace", "stacktrace_id", "value", "producer",
        "sample_type", "sample_unit", "period_type", "period_unit",
        "temporality", "period", "duration", "timestamp",
    ]
    assert batch.num_rows == 1


def test_stacktrace_listview_reuse():
    w = SampleWriterV2()
    t = _trace(0x10, 0x20, 0x30)
    _write_sample(w, t)
    _write_sample(w, t)  # same stack: must reuse offset/size
    _write_sample(w, _trace(0x40))
    batch = w.build_record()
    st = batch.column("stacktrace")
    # Identical stacks share their view: same offset & size.
    offsets = st.offsets.to_pylist()
    sizes = st.sizes.to_pylist()
    assert offsets[0] == offsets[1]
    assert sizes[0] == sizes[1] == 3
    assert sizes[2] == 1
    # Location dictionary has exactly 4 distinct locations.
    assert len(st.values.dictionary) == 4


def test_location_dedup_across_stacks():
    w = SampleWriterV2()
    _write_sample(w, _trace(0x10, 0x20))
    _write_sample(w, _trace(0x20, 0x30))  # 0x20 shared
    batch = w.build_record()
    st = batch.column("stacktrace")
    assert len(st.values.dictionary) == 3


def test_unsymbolized_frames_have_null_lines():
    w = SampleWriterV2()
    _write_sample(w, _trace(0x10, named=False))
    batch = w.build_record()
    loc = batch.column("stacktrace").values.dictionary
    lines = loc.field("lines")
    assert lines.is_valid().to_pylist() == [False]


def test_symbolized_frames_carry_function():
    w = SampleWriterV2()
    _write_sample(w, _trace(0x10, named=True))
    batch = w.build_record()
    loc = batch.column("stacktrace").values.dictionary
    lines = loc.field("lines")
    assert lines.is_valid().to_pylist() == [True]
    func = lines.values.field("function")
    names = func.dictionary.field("system_name").to_pylist()
    assert names == ["fn_10"]


def test_function_dedup():
    w = SampleWriterV2()
    m = APP
    f1 = Frame(kind=FrameType.NATIVE, address=0x10, mapping=m,
               function_name="shared_fn")
    f2 = Frame(kind=FrameType.NATIVE, address=0x99, mapping=m,
               function_name="shared_fn")
    _write_sample(w, Trace(frames=(f1,)))
    _write_sample(w, Trace(frames=(f2,)))
    batch = w.build_record()
    loc = batch.column("stacktrace").values.dictionary
    func_dict = loc.field("lines").values.field("function").dictionary
    assert len(func_dict) == 1  # one function entry, two locations


def test_mixed_frame_types():
    w = SampleWriterV2()
    frames = (
        Frame(kind=FrameType.KERNEL, address=0xffff000000001000,
              mapping=MappingFile(path="[kernel.kallsyms]")),
        Frame(kind=FrameType.NATIVE, address=0x10, mapping=APP),
        Frame(kind=FrameType.GPU_PC, address=0x300,
              mapping=MappingFile(file_id="c" * 32, path="codeobj-cc")),
    )
    _write_sample(w, Trace(frames=frames))
    batch = w.build_record()
    loc = batch.column("stacktrace").values.dictionary
    fts = loc.field("frame_type").to_pylist()
    assert fts == ["kernel", "native", "amdgpu_pc"]


def test_labels_struct_with_late_label():
    w = SampleWriterV2()
    _write_sample(w, _trace(0x1), labels={"node": "n1"})
    _write_sample(w, _trace(0x2), labels={"node": "n1", "pod": "p1"})
    batch = w.build_record()
    labels = batch.column("labels")
    node = labels.field("node").to_pylist()
    pod = labels.field("pod").to_pylist()
    assert node == ["n1", "n1"]
    assert pod == [None, "p1"]


def test_values_and_periods():
    w = SampleWriterV2()
    _write_sample(w, _trace(0x1), value=7, ts=123)
    batch = w.build_record()
    assert batch.column("value").to_pylist() == [7]
    assert batch.column("period").to_pylist() == [52_631_578]
    assert batch.column("sample_type").to_pylist() == ["samples"]
    assert batch.column("timestamp").cast(pa.int64()).to_pylist() == [123]


def test_stacktrace_id_is_trace_hash():
    w = SampleWriterV2()
    t = _trace(0x1, 0x2)
    _write_sample(w, t)
    batch = w.build_record()
    sid = batch.column("stacktrace_id").storage.to_pylist()
    assert sid == [t.trace_hash()]


def test_ipc_roundtrip_with_lz4():
    w = SampleWriterV2()
    for i in range(100):
        _write_sample(w, _trace(i, i + 1, named=True),
                      labels={"comm": f"c{i % 3}"}, ts=i)
    data = serialize_record(w.build_record())
    reader = pa.ipc.open_stream(io.BytesIO(data))
    table = reader.read_all()
    assert table.num_rows == 100
    assert table.schema.metadata[SCHEMA_VERSION_KEY.encode()] == b"v2"


def test_empty_labels_batch():
    w = SampleWriterV2()
    _write_sample(w, _trace(0x1))
    batch = w.build_record()
    assert batch.num_rows == 1
    assert pa.types.is_struct(batch.column("labels").type)
