"""pprof encoder round-trip tests.

Mirrors the reference's schema/encoding unit-test strategy
(SURVEY.md section 4): encode with our writer, decode with our independent
decoder, and verify against `protoc`-free expectations."""

from parca_agent_amd.pprof import (
    FrameKey,
    MappingKey,
    ProfileBuilder,
    ValueType,
    decode_profile,
)
from parca_agent_amd.pprof.proto import (
    Writer,
    decode_varint,
    encode_varint,
    iter_fields,
    to_int64,
)


def test_varint_roundtrip():
    for v in [0, 1, 127, 128, 300, 2**32, 2**63 - 1]:
        buf = encode_varint(v)
        out, pos = decode_varint(buf, 0)
        assert out == v
        assert pos == len(buf)


def test_negative_int64_varint():
    buf = encode_varint(-5)
    assert len(buf) == 10  # negative int64 takes full 10 bytes
    out, _ = decode_varint(buf, 0)
    assert to_int64(out) == -5


def test_writer_skips_zero_varint():
    w = Writer()
    w.varint(1, 0)
    assert w.getvalue() == b""
    w.varint_keep_zero(1, 0)
    assert w.getvalue() == b"\x08\x00"


def test_iter_fields():
    w = Writer()
    w.varint(1, 42)
    w.string(2, "hello")
    w.fixed64(3, 7)
    fields = list(iter_fields(w.getvalue()))
    assert fields == [(1, 0, 42), (2, 2, b"hello"), (3, 1, 7)]


def _mapping():
    return MappingKey(
        memory_start=0x400000,
        memory_limit=0x500000,
        file_offset=0,
        filename="/usr/bin/python3",
        build_id="abcd1234",
    )


def test_profile_roundtrip():
    b = ProfileBuilder(
        sample_types=[ValueType("samples", "count")],
        period_type=ValueType("cpu", "nanoseconds"),
        period=52_631_578,  # 1e9 / 19 Hz
        time_nanos=123456789,
    )
    m = _mapping()
    frames = [
        FrameKey(address=0x401000, mapping=m, function_name="leaf",
                 source_file="leaf.c", line=10),
        FrameKey(address=0x402000, mapping=m, function_name="main",
                 source_file="main.c", line=99),
    ]
    b.add_sample(frames, [3], labels=[("comm", "python3")])
    b.add_sample(frames, [2], labels=[("comm", "python3")])  # should merge
    b.add_sample(frames[1:], [1])

    p = decode_profile(b.serialize())
    assert p.sample_types == [ValueType("samples", "count")]
    assert p.period_type == ValueType("cpu", "nanoseconds")
    assert p.period == 52_631_578
    assert p.time_nanos == 123456789
    assert len(p.samples) == 2

    merged = next(s for s in p.samples if len(s["location_ids"]) == 2)
    assert merged["values"] == [5]
    assert merged["labels"] == {"comm": "python3"}
    assert p.stack_names(merged) == ["leaf", "main"]

    single = next(s for s in p.samples if len(s["location_ids"]) == 1)
    assert single["values"] == [1]
    assert p.stack_names(single) == ["main"]

    # Mapping carried through with filename + build id.
    mp = list(p.mappings.values())[0]
    assert p.strings[mp["filename"]] == "/usr/bin/python3"
    assert p.strings[mp["build_id"]] == "abcd1234"


def test_profile_gzip():
    b = ProfileBuilder(sample_types=[ValueType("samples", "count")])
    b.add_sample([FrameKey(address=1)], [1])
    p = decode_profile(b.serialize_gzip())
    assert len(p.samples) == 1


def test_unsymbolized_frame_has_no_line():
    b = ProfileBuilder(sample_types=[ValueType("samples", "count")])
    b.add_sample([FrameKey(address=0xdead, mapping=_mapping())], [1])
    p = decode_profile(b.serialize())
    loc = list(p.locations.values())[0]
    assert loc["address"] == 0xdead
    assert loc["lines"] == []


def test_num_labels():
    b = ProfileBuilder(sample_types=[ValueType("samples", "count")])
    b.add_sample([FrameKey(address=1)], [1],
                 num_labels=[("thread_id", 1234, "")])
    p = decode_profile(b.serialize())
    assert p.samples[0]["num_labels"] == {"thread_id": 1234}


def test_multi_value_types():
    b = ProfileBuilder(
        sample_types=[ValueType("alloc_objects", "count"),
                      ValueType("alloc_space", "bytes")]
    )
    b.add_sample([FrameKey(address=1)], [2, 4096])
    p = decode_profile(b.serialize())
    assert p.samples[0]["values"] == [2, 4096]


def test_has_functions_only_for_symbolized_mappings():
    """Mappings with no symbolized location must not claim has_functions
    (it would suppress consumers' local-symbolization fallback)."""
    from parca_agent_amd.pprof import (FrameKey, MappingKey, ProfileBuilder,
                                       ValueType, decode_profile)
    sym_map = MappingKey(memory_start=0x1000, memory_limit=0x2000,
                         file_offset=0, filename="/bin/sym", build_id="s")
    raw_map = MappingKey(memory_start=0x3000, memory_limit=0x4000,
                         file_offset=0, filename="/bin/raw", build_id="r")
    b = ProfileBuilder(sample_types=[ValueType("samples", "count")])
    b.add_sample([
        FrameKey(address=0x1100, mapping=sym_map, function_name="f"),
        FrameKey(address=0x3100, mapping=raw_map),
    ], [1])
    prof = decode_profile(b.serialize())
    by_name = {prof.strings[m["filename"]]: m for m in prof.mappings.values()}
    assert by_name["/bin/sym"]["has_functions"] is True
    assert by_name["/bin/raw"]["has_functions"] is False
