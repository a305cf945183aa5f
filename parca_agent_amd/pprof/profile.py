"""pprof profile builder and encoder (perftools.profiles.Profile).

The pprof protobuf is the lingua franca of the Parca ecosystem: the
reference agent emits it for OOM profiles (reference: oom/oomprof.go:16-125)
and the Parca server stores everything as pprof-compatible profiles. Our
local-store mode and the node-level merged GPU profile both emit this
format directly.

Schema follows github.com/google/pprof/proto/profile.proto (stable,
public). Encoded with the in-repo minimal proto writer so no codegen is
required.
"""

from __future__ import annotations

import gzip
import time
from dataclasses import dataclass, field
from typing import Dict, Iterable, List, Optional, Sequence, Tuple

from .proto import Writer, decode_packed_varints, iter_fields, to_int64


@dataclass(frozen=True)
class ValueType:
    type: str
    unit: str


@dataclass(frozen=True)
class MappingKey:
    memory_start: int
    memory_limit: int
    file_offset: int
    filename: str
    build_id: str


@dataclass(frozen=True)
class FrameKey:
    """One resolved frame: address within a mapping plus optional symbols."""

    address: int
    mapping: Optional[MappingKey] = None
    function_name: str = ""
    source_file: str = ""
    line: int = 0


class ProfileBuilder:
    """Accumulates samples and emits a serialized pprof Profile.

    Deduplicates strings, mappings, functions and locations, mirroring the
    behaviour of pprof writers. Not thread-safe; callers hold their own lock
    (the reporter serializes access).
    """

    def __init__(
        self,
        sample_types: Sequence[ValueType],
        period_type: Optional[ValueType] = None,
        period: int = 0,
        time_nanos: Optional[int] = None,
        duration_nanos: int = 0,
        default_sample_type: str = "",
        doc_url: str = "",
    ) -> None:
        self.sample_types = list(sample_types)
        self.period_type = period_type
        self.period = period
        self.time_nanos = time_nanos if time_nanos is not None else time.time_ns()
        self.duration_nanos = duration_nanos
        self.default_sample_type = default_sample_type
        self.doc_url = doc_url
        self.comments: List[str] = []

        self._strings: Dict[str, int] = {"": 0}
        self._string_list: List[str] = [""]
        self._mappings: Dict[MappingKey, int] = {}
        self._mapping_list: List[MappingKey] = []
        self._functions: Dict[Tuple[str, str], int] = {}
        self._function_list: List[Tuple[str, str]] = []
        self._locations: Dict[FrameKey, int] = {}
        self._location_list: List[FrameKey] = []
        # (location_ids tuple, labels tuple) -> values list (accumulated)
        self._samples: Dict[Tuple[Tuple[int, ...], Tuple], List[int]] = {}

    # -- interning ---------------------------------------------------------

    def _string(self, s: str) -> int:
        idx = self._strings.get(s)
        if idx is None:
            idx = len(self._string_list)
            self._strings[s] = idx
            self._string_list.append(s)
        return idx

    def mapping_id(self, key: MappingKey) -> int:
        idx = self._mappings.get(key)
        if idx is None:
            idx = len(self._mapping_list) + 1
            self._mappings[key] = idx
            self._mapping_list.append(key)
        return idx

    def location_id(self, frame: FrameKey) -> int:
        idx = self._locations.get(frame)
        if idx is None:
            idx = len(self._location_list) + 1
            self._locations[frame] = idx
            self._location_list.append(frame)
            if frame.mapping is not None:
                self.mapping_id(frame.mapping)
        return idx

    def _function_id(self, name: str, filename: str) -> int:
        key = (name, filename)
        idx = self._functions.get(key)
        if idx is None:
            idx = len(self._function_list) + 1
            self._functions[key] = idx
            self._function_list.append(key)
        return idx

    # -- sample ingestion --------------------------------------------------

    def add_sample(
        self,
        frames: Iterable[FrameKey],
        values: Sequence[int],
        labels: Sequence[Tuple[str, str]] = (),
        num_labels: Sequence[Tuple[str, int, str]] = (),
    ) -> None:
        """Add one stack sample. ``frames`` are leaf-first, pprof order."""
        if len(values) != len(self.sample_types):
            raise ValueError(
                f"sample has {len(values)} values, profile has "
                f"{len(self.sample_types)} sample types"
            )
        loc_ids = tuple(self.location_id(f) for f in frames)
        self.add_sample_by_ids(loc_ids, values, labels, num_labels)

    def add_sample_by_ids(
        self,
        loc_ids: Tuple[int, ...],
        values: Sequence[int],
        labels: Sequence[Tuple[str, str]] = (),
        num_labels: Sequence[Tuple[str, int, str]] = (),
    ) -> None:
        """add_sample for pre-resolved location ids (hot flush path:
        traces are shared objects, so their id tuples are memoizable)."""
        label_key = (tuple(sorted(labels)), tuple(sorted(num_labels)))
        existing = self._samples.get((loc_ids, label_key))
        if existing is None:
            self._samples[(loc_ids, label_key)] = list(values)
        else:
            for i, v in enumerate(values):
                existing[i] += v

    @property
    def n_samples(self) -> int:
        return len(self._samples)

    # -- encoding ----------------------------------------------------------

    def _encode_value_type(self, vt: ValueType) -> Writer:
        w = Writer()
        w.varint(1, self._string(vt.type))
        w.varint(2, self._string(vt.unit))
        return w

    def serialize(self) -> bytes:
        w = Writer()
        for vt in self.sample_types:
            w.message(1, self._encode_value_type(vt))

        # Label sets repeat across samples (per-cpu/per-tid spread over a
        # few thousand distinct dicts): encode each distinct set once.
        label_blob_cache: Dict[tuple, bytes] = {}
        for (loc_ids, (labels, num_labels)), values in self._samples.items():
            sw = Writer()
            sw.packed_varints(1, list(loc_ids))
            sw.packed_varints(2, values)
            blob = label_blob_cache.get((labels, num_labels))
            if blob is None:
                bw = Writer()
                for key, val in labels:
                    lw = Writer()
                    lw.varint(1, self._string(key))
                    lw.varint(2, self._string(val))
                    bw.message(3, lw)
                for key, num, unit in num_labels:
                    lw = Writer()
                    lw.varint(1, self._string(key))
                    lw.varint(3, num)
                    if unit:
                        lw.varint(4, self._string(unit))
                    bw.message(3, lw)
                blob = bw.getvalue()
                label_blob_cache[(labels, num_labels)] = blob
            sw._parts.append(blob)
            w.message(2, sw)

        # has_functions only for mappings where at least one location was
        # actually symbolized — claiming it unconditionally suppresses
        # pprof consumers' local-symbolization fallback for mappings we
        # never resolved.
        mappings_with_functions = set()
        for fk in self._location_list:
            if fk.mapping is not None and (fk.function_name or fk.source_file):
                mappings_with_functions.add(self.mapping_id(fk.mapping))
        for i, mk in enumerate(self._mapping_list):
            mw = Writer()
            mw.varint(1, i + 1)
            mw.varint(2, mk.memory_start)
            mw.varint(3, mk.memory_limit)
            mw.varint(4, mk.file_offset)
            mw.varint(5, self._string(mk.filename))
            mw.varint(6, self._string(mk.build_id))
            if (i + 1) in mappings_with_functions:
                mw.bool(7, True)
            w.message(3, mw)

        # Locations reference functions; build function list as a side effect
        # first so ids are stable, then emit both.
        loc_writers = []
        for i, fk in enumerate(self._location_list):
            lw = Writer()
            lw.varint(1, i + 1)
            if fk.mapping is not None:
                lw.varint(2, self.mapping_id(fk.mapping))
            lw.varint(3, fk.address)
            if fk.function_name or fk.source_file:
                fid = self._function_id(fk.function_name, fk.source_file)
                linew = Writer()
                linew.varint(1, fid)
                linew.varint(2, fk.line)
                lw.message(4, linew)
            loc_writers.append(lw)
        for lw in loc_writers:
            w.message(4, lw)

        for i, (name, filename) in enumerate(self._function_list):
            fw = Writer()
            fw.varint(1, i + 1)
            fw.varint(2, self._string(name))
            fw.varint(3, self._string(name))  # system_name = name (no mangling info)
            fw.varint(4, self._string(filename))
            w.message(5, fw)

        w.varint(9, self.time_nanos)
        w.varint(10, self.duration_nanos)
        if self.period_type is not None:
            w.message(11, self._encode_value_type(self.period_type))
        w.varint(12, self.period)
        for c in self.comments:
            w.varint(13, self._string(c))
        if self.default_sample_type:
            w.varint(14, self._string(self.default_sample_type))
        if self.doc_url:
            w.varint(15, self._string(self.doc_url))

        # String table last: every _string() interning above must be complete
        # before the table is written (protobuf field order is free). The
        # empty first entry must still be emitted explicitly — zero-length
        # repeated-string entries need their tag.
        from .proto import encode_varint

        for s in self._string_list:
            w.tag(6, 2)
            encoded = s.encode("utf-8")
            w._parts.append(encode_varint(len(encoded)))
            w._parts.append(encoded)
        return w.getvalue()

    def serialize_gzip(self) -> bytes:
        # level 6: ~3x faster than the gzip default (9) for a few
        # percent of size — this runs on every flush interval.
        return gzip.compress(self.serialize(), compresslevel=6, mtime=0)


# -- decoding (test support + offline tooling) ----------------------------


@dataclass
class DecodedProfile:
    sample_types: List[ValueType] = field(default_factory=list)
    samples: List[dict] = field(default_factory=list)
    mappings: Dict[int, dict] = field(default_factory=dict)
    locations: Dict[int, dict] = field(default_factory=dict)
    functions: Dict[int, dict] = field(default_factory=dict)
    strings: List[str] = field(default_factory=list)
    period_type: Optional[ValueType] = None
    period: int = 0
    time_nanos: int = 0
    duration_nanos: int = 0

    def stack_names(self, sample: dict) -> List[str]:
        """Leaf-first function names for one decoded sample."""
        names = []
        for loc_id in sample["location_ids"]:
            loc = self.locations[loc_id]
            if loc["lines"]:
                fid = loc["lines"][0]["function_id"]
                names.append(self.strings[self.functions[fid]["name"]])
            else:
                names.append(hex(loc["address"]))
        return names


def decode_profile(data: bytes) -> DecodedProfile:
    """Decode a (possibly gzipped) pprof Profile for verification."""
    if data[:2] == b"\x1f\x8b":
        data = gzip.decompress(data)
    p = DecodedProfile()
    raw_sample_types: List[bytes] = []
    raw_period_type: Optional[bytes] = None

    for fieldno, _wt, value in iter_fields(data):
        if fieldno == 1:
            raw_sample_types.append(value)
        elif fieldno == 2:
            sample = {"location_ids": [], "values": [], "labels": {}, "num_labels": {}}
            for sf, swt, sv in iter_fields(value):
                if sf == 1:
                    if swt == 2:
                        sample["location_ids"] = decode_packed_varints(sv)
                    else:
                        sample["location_ids"].append(sv)
                elif sf == 2:
                    if swt == 2:
                        sample["values"] = [to_int64(v) for v in decode_packed_varints(sv)]
                    else:
                        sample["values"].append(to_int64(sv))
                elif sf == 3:
                    lbl = {"key": 0, "str": 0, "num": 0, "num_unit": 0}
                    for lf, _lwt, lv in iter_fields(sv):
                        if lf == 1:
                            lbl["key"] = lv
                        elif lf == 2:
                            lbl["str"] = lv
                        elif lf == 3:
                            lbl["num"] = to_int64(lv)
                        elif lf == 4:
                            lbl["num_unit"] = lv
                    sample.setdefault("raw_labels", []).append(lbl)
            p.samples.append(sample)
        elif fieldno == 3:
            m = {"id": 0, "memory_start": 0, "memory_limit": 0, "file_offset": 0,
                 "filename": 0, "build_id": 0, "has_functions": False}
            for mf, _mwt, mv in iter_fields(value):
                if mf == 1:
                    m["id"] = mv
                elif mf == 2:
                    m["memory_start"] = mv
                elif mf == 3:
                    m["memory_limit"] = mv
                elif mf == 4:
                    m["file_offset"] = mv
                elif mf == 5:
                    m["filename"] = mv
                elif mf == 6:
                    m["build_id"] = mv
                elif mf == 7:
                    m["has_functions"] = bool(mv)
            p.mappings[m["id"]] = m
        elif fieldno == 4:
            loc = {"id": 0, "mapping_id": 0, "address": 0, "lines": []}
            for lf, _lwt, lv in iter_fields(value):
                if lf == 1:
                    loc["id"] = lv
                elif lf == 2:
                    loc["mapping_id"] = lv
                elif lf == 3:
                    loc["address"] = lv
                elif lf == 4:
                    line = {"function_id": 0, "line": 0}
                    for llf, _llwt, llv in iter_fields(lv):
                        if llf == 1:
                            line["function_id"] = llv
                        elif llf == 2:
                            line["line"] = to_int64(llv)
                    loc["lines"].append(line)
            p.locations[loc["id"]] = loc
        elif fieldno == 5:
            fn = {"id": 0, "name": 0, "system_name": 0, "filename": 0}
            for ff, _fwt, fv in iter_fields(value):
                if ff == 1:
                    fn["id"] = fv
                elif ff == 2:
                    fn["name"] = fv
                elif ff == 3:
                    fn["system_name"] = fv
                elif ff == 4:
                    fn["filename"] = fv
            p.functions[fn["id"]] = fn
        elif fieldno == 6:
            p.strings.append(value.decode("utf-8"))
        elif fieldno == 9:
            p.time_nanos = value
        elif fieldno == 10:
            p.duration_nanos = value
        elif fieldno == 11:
            raw_period_type = value
        elif fieldno == 12:
            p.period = value

    def parse_vt(raw: bytes) -> ValueType:
        t = u = 0
        for vf, _vwt, vv in iter_fields(raw):
            if vf == 1:
                t = vv
            elif vf == 2:
                u = vv
        return ValueType(p.strings[t], p.strings[u])

    p.sample_types = [parse_vt(r) for r in raw_sample_types]
    if raw_period_type is not None:
        p.period_type = parse_vt(raw_period_type)

    # Resolve label string indices now the table is complete.
    for sample in p.samples:
        for lbl in sample.get("raw_labels", []):
            key = p.strings[lbl["key"]]
            if lbl["str"]:
                sample["labels"][key] = p.strings[lbl["str"]]
            else:
                sample["num_labels"][key] = lbl["num"]
    return p
