"""Arrow v2 sample-record writer.

Produces record batches wire-compatible with the reference's v2 schema
(reference: reporter/arrow_v2.go:35-160, 576-609): 13 fields —
labels (struct of per-key REE<dict<u32,string>>), stacktrace
(ListView<dict<u32, LocationStruct>>), stacktrace_id (UUID ext), value
(int64), producer/sample_type/sample_unit/period_type/period_unit/
temporality (REE<string>), period (REE<int64>), duration (REE<uint64>),
timestamp (timestamp[ns, UTC]); schema metadata
parca_write_schema_version=v2.

Identical stacktraces reuse their ListView offset/size (dedup by
TraceHash, arrow_v2.go:220-330); locations and functions are
dictionary-deduplicated; frames without symbols carry null `lines` so the
server-side symbolizer picks them up (arrow_v2.go:399-431).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import pyarrow as pa

from ..model import Frame, Trace

SCHEMA_VERSION_KEY = "parca_write_schema_version"
SCHEMA_VERSION_V2 = "v2"

PRODUCER = "parca-agent-amd"


class _Interner:
    def __init__(self) -> None:
        self.index: Dict = {}
        self.values: List = []

    def intern(self, value) -> int:
        idx = self.index.get(value)
        if idx is None:
            idx = len(self.values)
            self.index[value] = idx
            self.values.append(value)
        return idx

    def __len__(self) -> int:
        return len(self.values)


class _REEColumn:
    """Run-end-encoded column accumulator."""

    def __init__(self) -> None:
        self.run_values: List = []
        self.run_ends: List[int] = []
        self._n = 0

    def append(self, value) -> None:
        self._n += 1
        if self.run_values and self.run_values[-1] == value:
            self.run_ends[-1] = self._n
        else:
            self.run_values.append(value)
            self.run_ends.append(self._n)

    def build(self, value_type: pa.DataType) -> pa.Array:
        return pa.RunEndEncodedArray.from_arrays(
            pa.array(self.run_ends, pa.int32()),
            pa.array(self.run_values, value_type),
        )

    def build_dict(self) -> pa.Array:
        """REE over dictionary-encoded strings (labels columns)."""
        interner = _Interner()
        idxs = [None if v is None else interner.intern(v) for v in self.run_values]
        dict_arr = pa.DictionaryArray.from_arrays(
            pa.array(idxs, pa.uint32()),
            pa.array(interner.values, pa.string()),
        )
        return pa.RunEndEncodedArray.from_arrays(
            pa.array(self.run_ends, pa.int32()), dict_arr)


class SampleWriterV2:
    """Accumulates samples and emits one Arrow record batch.

    Not thread-safe; the reporter swaps writers under its lock at flush
    (reference: parca_reporter.go:2056-2071).
    """

    def __init__(self) -> None:
        # Location dictionary (dedup across stacks).
        self._loc_interner: Dict[Tuple, int] = {}
        self._loc_rows: List[Tuple] = []
        # Function dictionary.
        self._func_interner = _Interner()
        # Stacktrace ListView reuse by trace hash -> (offset, size).
        self._stack_views: Dict[bytes, Tuple[int, int]] = {}
        self._stack_values: List[int] = []  # indices into location dict

        self._sample_offsets: List[int] = []
        self._sample_sizes: List[int] = []
        self._stacktrace_ids: List[bytes] = []
        self._values: List[int] = []
        self._timestamps: List[int] = []
        self._producer = _REEColumn()
        self._sample_type = _REEColumn()
        self._sample_unit = _REEColumn()
        self._period_type = _REEColumn()
        self._period_unit = _REEColumn()
        self._temporality = _REEColumn()
        self._period = _REEColumn()
        self._duration = _REEColumn()
        # label name -> list index; lazily materialized columns padded with
        # nulls for rows written before the label first appeared.
        self._label_cols: Dict[str, List[Optional[str]]] = {}
        self._nrows = 0

    # -- frames ------------------------------------------------------------

    def _intern_function(self, frame: Frame) -> int:
        return self._func_interner.intern(
            (frame.function_name, frame.source_file, frame.source_line))

    def _intern_location(self, frame: Frame) -> int:
        mapping_file = frame.mapping.path if frame.mapping else None
        mapping_build_id = frame.mapping.id_label if frame.mapping else None
        has_symbols = bool(frame.function_name)
        func_idx = self._intern_function(frame) if has_symbols else -1
        key = (frame.address, frame.kind.value, mapping_file,
               mapping_build_id, func_idx)
        idx = self._loc_interner.get(key)
        if idx is None:
            idx = len(self._loc_rows)
            self._loc_interner[key] = idx
            self._loc_rows.append(key)
        return idx

    def append_stacktrace(self, trace: Trace) -> Tuple[int, int]:
        """Intern a stacktrace; returns its (offset, size) view."""
        th = trace.trace_hash()
        view = self._stack_views.get(th)
        if view is None:
            offset = len(self._stack_values)
            for frame in trace.frames:
                self._stack_values.append(self._intern_location(frame))
            view = (offset, len(trace.frames))
            self._stack_views[th] = view
        return view

    # -- samples -----------------------------------------------------------

    def append_sample(
        self,
        trace: Trace,
        labels: Dict[str, str],
        value: int,
        timestamp_ns: int,
        sample_type: str,
        sample_unit: str,
        period_type: str,
        period_unit: str,
        period: int,
        duration_ns: int = 0,
        temporality: str = "delta",
    ) -> None:
        offset, size = self.append_stacktrace(trace)
        self._sample_offsets.append(offset)
        self._sample_sizes.append(size)
        self._stacktrace_ids.append(trace.trace_hash())
        self._values.append(value)
        self._timestamps.append(timestamp_ns)
        self._producer.append(PRODUCER)
        self._sample_type.append(sample_type)
        self._sample_unit.append(sample_unit)
        self._period_type.append(period_type)
        self._period_unit.append(period_unit)
        self._temporality.append(temporality)
        self._period.append(period)
        self._duration.append(duration_ns)

        for name in self._label_cols:
            self._label_cols[name].append(labels.get(name))
        for name, val in labels.items():
            if name not in self._label_cols:
                self._label_cols[name] = [None] * self._nrows + [val]
        self._nrows += 1

    @property
    def n_rows(self) -> int:
        return self._nrows

    # -- record assembly ---------------------------------------------------

    def _build_location_dict(self) -> pa.Array:
        addresses = []
        frame_types = []
        mapping_files = []
        mapping_build_ids = []
        ft_int = _Interner()
        mf_int = _Interner()
        mb_int = _Interner()
        fn_file_int = _Interner()

        # lines: ListView over line-structs with function dictionary;
        # unsymbolized locations get a NULL lines entry (not empty) to mark
        # them for server-side symbolization (arrow_v2.go:399-431).
        line_offsets: List[int] = []
        line_sizes: List[int] = []
        line_valid: List[bool] = []
        line_func_idx: List[int] = []
        line_lines: List[int] = []

        for (address, kind, mfile, mbuild, func_idx) in self._loc_rows:
            addresses.append(address)
            frame_types.append(ft_int.intern(kind))
            mapping_files.append(None if mfile is None else mf_int.intern(mfile))
            mapping_build_ids.append(
                None if mbuild is None else mb_int.intern(mbuild))
            if func_idx < 0:
                line_offsets.append(len(line_func_idx))
                line_sizes.append(0)
                line_valid.append(False)
            else:
                line_offsets.append(len(line_func_idx))
                line_sizes.append(1)
                line_valid.append(True)
                line_func_idx.append(func_idx)
                name, sfile, line = self._func_interner.values[func_idx]
                line_lines.append(line)

        func_names = []
        func_files = []
        func_start_lines = []
        for (name, sfile, _line) in self._func_interner.values:
            func_names.append(name)
            func_files.append(fn_file_int.intern(sfile))
            func_start_lines.append(0)

        # Field nullability mirrors the reference exactly
        # (arrow_v2.go:43-48): start_line is non-nullable.
        func_fields = [
            pa.field("system_name", pa.string_view(), nullable=True),
            pa.field("filename", pa.dictionary(pa.uint32(), pa.string()),
                     nullable=True),
            pa.field("start_line", pa.uint64(), nullable=False),
        ]
        func_struct = pa.StructArray.from_arrays(
            [
                pa.array(func_names, pa.string_view()),
                pa.DictionaryArray.from_arrays(
                    pa.array(func_files, pa.uint32()),
                    pa.array(fn_file_int.values or [""], pa.string())),
                pa.array(func_start_lines, pa.uint64()),
            ],
            fields=func_fields,
        ) if self._func_interner.values else pa.StructArray.from_arrays(
            [pa.array([], pa.string_view()),
             pa.DictionaryArray.from_arrays(pa.array([], pa.uint32()),
                                            pa.array([""], pa.string())),
             pa.array([], pa.uint64())],
            fields=func_fields)

        func_dict = pa.DictionaryArray.from_arrays(
            pa.array(line_func_idx, pa.uint32()), func_struct)
        # line/column/function are non-nullable (arrow_v2.go:56-60).
        line_struct = pa.StructArray.from_arrays(
            [pa.array(line_lines, pa.uint64()),
             pa.array([0] * len(line_lines), pa.uint64()),
             func_dict],
            fields=[
                pa.field("line", pa.uint64(), nullable=False),
                pa.field("column", pa.uint64(), nullable=False),
                pa.field("function", func_dict.type, nullable=False),
            ],
        )
        lines_lv = pa.ListViewArray.from_arrays(
            pa.array(line_offsets, pa.int32()),
            pa.array(line_sizes, pa.int32()),
            line_struct,
            mask=pa.array([not v for v in line_valid], pa.bool_())
            if line_valid else None,
        )

        # address is non-nullable; the rest nullable (arrow_v2.go:81-88).
        str_dict_t = pa.dictionary(pa.uint32(), pa.string())
        loc_struct = pa.StructArray.from_arrays(
            [
                pa.array(addresses, pa.uint64()),
                pa.DictionaryArray.from_arrays(
                    pa.array(frame_types, pa.uint32()),
                    pa.array(ft_int.values or [""], pa.string())),
                pa.DictionaryArray.from_arrays(
                    pa.array(mapping_files, pa.uint32()),
                    pa.array(mf_int.values or [""], pa.string())),
                pa.DictionaryArray.from_arrays(
                    pa.array(mapping_build_ids, pa.uint32()),
                    pa.array(mb_int.values or [""], pa.string())),
                lines_lv,
            ],
            fields=[
                pa.field("address", pa.uint64(), nullable=False),
                pa.field("frame_type", str_dict_t, nullable=True),
                pa.field("mapping_file", str_dict_t, nullable=True),
                pa.field("mapping_build_id", str_dict_t, nullable=True),
                pa.field("lines", lines_lv.type, nullable=True),
            ],
        )
        return pa.DictionaryArray.from_arrays(
            pa.array(self._stack_values, pa.uint32()), loc_struct)

    def build_record(self) -> pa.RecordBatch:
        loc_dict = self._build_location_dict()
        stacktrace = pa.ListViewArray.from_arrays(
            pa.array(self._sample_offsets, pa.int32()),
            pa.array(self._sample_sizes, pa.int32()),
            loc_dict,
        )
        stacktrace_id = pa.ExtensionArray.from_storage(
            pa.uuid(), pa.array(self._stacktrace_ids, pa.binary(16)))

        label_names = sorted(self._label_cols)
        label_arrays = []
        for name in label_names:
            col = _REEColumn()
            for v in self._label_cols[name]:
                col.append(v)
            label_arrays.append(col.build_dict())
        if label_names:
            # v2 label struct children carry the BARE label name —
            # "labels.<name>" is v1's flat-column convention only
            # (arrow_v2.go:567-573 labelField uses labelName as-is).
            labels = pa.StructArray.from_arrays(
                label_arrays,
                fields=[pa.field(n, label_arrays[i].type, nullable=True)
                        for i, n in enumerate(label_names)])
        else:
            # Empty struct column with the right length.
            labels = pa.array([{}] * self._nrows, pa.struct([]))

        arrays = [
            labels,
            stacktrace,
            stacktrace_id,
            pa.array(self._values, pa.int64()),
            self._producer.build(pa.string()),
            self._sample_type.build(pa.string()),
            self._sample_unit.build(pa.string()),
            self._period_type.build(pa.string()),
            self._period_unit.build(pa.string()),
            self._temporality.build(pa.string()),
            self._period.build(pa.int64()),
            self._duration.build(pa.uint64()),
            pa.array(self._timestamps, pa.timestamp("ns", tz="UTC")),
        ]
        names = ["labels", "stacktrace", "stacktrace_id", "value", "producer",
                 "sample_type", "sample_unit", "period_type", "period_unit",
                 "temporality", "period", "duration", "timestamp"]
        batch = pa.record_batch(arrays, names=names)
        return batch.replace_schema_metadata({SCHEMA_VERSION_KEY: SCHEMA_VERSION_V2})


def serialize_record(batch: pa.RecordBatch, compress: str = "lz4") -> bytes:
    """Arrow IPC stream bytes for one batch (LZ4-framed like the reference's
    WriteArrow payload, parca_reporter.go:2150-2190)."""
    import io

    sink = io.BytesIO()
    options = pa.ipc.IpcWriteOptions(compression=compress)
    with pa.ipc.new_stream(sink, batch.schema, options=options) as writer:
        writer.write_batch(batch)
    return sink.getvalue()
