"""Arrow v1 sample records: the two-record Write protocol.

Wire-compatible with the reference's v1 schema (reference:
reporter/arrow.go:260-392, 485-543): a *samples* record with
run-end-encoded/dictionary columns ending in the 11 fixed fields
(stacktrace_id, value, producer, sample_type, sample_unit, period_type,
period_unit, temporality, period, duration, timestamp; labels.* columns
first), and a *stacktraces* record (LocationsWriter) with the nested
List<Struct> locations/lines schema sent when the server requests
unknown stacktrace IDs (the two-phase Write round trip,
parca_reporter.go:1667-1803). Schema metadata
parca_write_schema_version=v1.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Set

import pyarrow as pa

from ..model import Trace
from .arrow_v2 import PRODUCER, _Interner, _REEColumn

SCHEMA_VERSION_KEY = "parca_write_schema_version"
SCHEMA_VERSION_V1 = "v1"


class _REEDictColumn(_REEColumn):
    """REE over dictionary-encoded *binary* values (v1 label/meta cols,
    arrow.go labelArrowType)."""

    def build_binary_dict(self) -> pa.Array:
        interner = _Interner()
        idxs = [None if v is None else interner.intern(v)
                for v in self.run_values]
        dict_arr = pa.DictionaryArray.from_arrays(
            pa.array(idxs, pa.uint32()),
            pa.array([v.encode() if isinstance(v, str) else v
                      for v in interner.values], pa.binary()),
        )
        return pa.RunEndEncodedArray.from_arrays(
            pa.array(self.run_ends, pa.int32()), dict_arr)


class SampleWriterV1:
    """Accumulates samples; emits the v1 samples record. Stacktraces are
    referenced by 16-byte hash only; their expansion lives in
    LocationsWriterV1."""

    def __init__(self) -> None:
        self._label_cols: Dict[str, List[Optional[str]]] = {}
        self._stacktrace_id = _REEDictColumn()
        self._values: List[int] = []
        self._producer = _REEDictColumn()
        self._sample_type = _REEDictColumn()
        self._sample_unit = _REEDictColumn()
        self._period_type = _REEDictColumn()
        self._period_unit = _REEDictColumn()
        self._temporality = _REEDictColumn()
        self._period = _REEColumn()
        self._duration = _REEColumn()
        self._timestamp = _REEColumn()
        self._nrows = 0
        self._traces: Dict[bytes, Trace] = {}

    def append_sample(self, trace: Trace, labels: Dict[str, str],
                      value: int, timestamp_ns: int, sample_type: str,
                      sample_unit: str, period_type: str, period_unit: str,
                      period: int, duration_ns: int = 0,
                      temporality: str = "delta") -> None:
        th = trace.trace_hash()
        self._traces.setdefault(th, trace)
        self._stacktrace_id.append(th)
        self._values.append(value)
        self._producer.append(PRODUCER)
        self._sample_type.append(sample_type)
        self._sample_unit.append(sample_unit)
        self._period_type.append(period_type)
        self._period_unit.append(period_unit)
        self._temporality.append(temporality)
        self._period.append(period)
        self._duration.append(duration_ns)
        self._timestamp.append(timestamp_ns)
        for name in self._label_cols:
            self._label_cols[name].append(labels.get(name))
        for name, val in labels.items():
            if name not in self._label_cols:
                self._label_cols[name] = [None] * self._nrows + [val]
        self._nrows += 1

    @property
    def n_rows(self) -> int:
        return self._nrows

    def traces(self) -> Dict[bytes, Trace]:
        return dict(self._traces)

    def build_record(self) -> pa.RecordBatch:
        arrays = []
        names = []
        for label in sorted(self._label_cols):
            col = _REEDictColumn()
            for v in self._label_cols[label]:
                col.append(v)
            arrays.append(col.build_binary_dict())
            names.append(f"labels.{label}")
        arrays += [
            self._stacktrace_id.build_binary_dict(),
            pa.array(self._values, pa.int64()),
            self._producer.build_binary_dict(),
            self._sample_type.build_binary_dict(),
            self._sample_unit.build_binary_dict(),
            self._period_type.build_binary_dict(),
            self._period_unit.build_binary_dict(),
            self._temporality.build_binary_dict(),
            self._period.build(pa.int64()),
            self._duration.build(pa.int64()),
            self._timestamp.build(pa.int64()),
        ]
        names += ["stacktrace_id", "value", "producer", "sample_type",
                  "sample_unit", "period_type", "period_unit", "temporality",
                  "period", "duration", "timestamp"]
        batch = pa.record_batch(arrays, names=names)
        return batch.replace_schema_metadata(
            {SCHEMA_VERSION_KEY: SCHEMA_VERSION_V1})


class LocationsWriterV1:
    """The stacktraces record: one row per requested stacktrace ID with a
    nested List<Struct> of locations (reference: LocationsWriter,
    arrow.go:209-258, 335-392)."""

    def __init__(self) -> None:
        self._ids: List[bytes] = []
        self._loc_offsets: List[int] = [0]
        self._addresses: List[int] = []
        self._frame_types: List[str] = []
        self._mapping_files: List[Optional[str]] = []
        self._mapping_build_ids: List[Optional[str]] = []
        self._line_offsets: List[int] = [0]
        self._func_names: List[str] = []
        self._file_names: List[str] = []
        self._lines: List[int] = []
        self._line_valid: List[bool] = []

    def append_stacktrace(self, trace_id: bytes, trace: Trace) -> None:
        self._ids.append(trace_id)
        for f in trace.frames:
            self._addresses.append(f.address)
            self._frame_types.append(f.kind.value)
            self._mapping_files.append(f.mapping.path if f.mapping else None)
            self._mapping_build_ids.append(
                f.mapping.id_label if f.mapping else None)
            if f.function_name:
                self._func_names.append(f.function_name)
                self._file_names.append(f.source_file)
                self._lines.append(f.source_line)
                self._line_offsets.append(len(self._func_names))
                self._line_valid.append(True)
            else:
                self._line_offsets.append(len(self._func_names))
                self._line_valid.append(False)
        self._loc_offsets.append(len(self._addresses))

    def build_record(self) -> pa.RecordBatch:
        line_struct = pa.StructArray.from_arrays(
            [pa.array(self._lines, pa.int64()),
             pa.array(self._func_names, pa.string()),
             pa.array(self._file_names, pa.string())],
            names=["line", "function_name", "function_filename"])
        lines = pa.ListArray.from_arrays(
            pa.array(self._line_offsets, pa.int32()), line_struct,
            mask=pa.array([not v for v in self._line_valid], pa.bool_())
            if self._line_valid else None)
        loc_struct = pa.StructArray.from_arrays(
            [pa.array(self._addresses, pa.uint64()),
             pa.array(self._frame_types, pa.string()),
             pa.array(self._mapping_files, pa.string()),
             pa.array(self._mapping_build_ids, pa.string()),
             lines],
            names=["address", "frame_type", "mapping_file",
                   "mapping_build_id", "lines"])
        locations = pa.ListArray.from_arrays(
            pa.array(self._loc_offsets, pa.int32()), loc_struct)
        batch = pa.record_batch(
            [pa.array(self._ids, pa.binary()), locations],
            names=["stacktrace_id", "locations"])
        return batch.replace_schema_metadata(
            {SCHEMA_VERSION_KEY: SCHEMA_VERSION_V1})


def decode_requested_ids(record_bytes: bytes) -> Set[bytes]:
    """The server's Write response: a record with a `stacktrace_id`
    column naming the IDs it wants expanded."""
    import io

    reader = pa.ipc.open_stream(io.BytesIO(record_bytes))
    table = reader.read_all()
    if "stacktrace_id" not in table.schema.names:
        return set()
    col = table.column("stacktrace_id")
    out: Set[bytes] = set()
    for chunk in col.chunks:
        if pa.types.is_run_end_encoded(chunk.type):
            chunk = chunk.values  # distinct values suffice for the ID set
        if pa.types.is_dictionary(chunk.type):
            chunk = chunk.dictionary_decode()
        out.update(v for v in chunk.to_pylist() if v is not None)
    return out
