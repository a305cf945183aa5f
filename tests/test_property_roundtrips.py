"""Property-based round-trips (hypothesis) for the wire layers: GPU ring
event codecs and the pprof builder/decoder. These layers carry every
byte the agent emits, so structured fuzzing beats hand-picked cases."""

import hypothesis.strategies as st
from hypothesis import given, settings

from parca_agent_amd.gpu import events as ev
from parca_agent_amd.model import SampleType
from parca_agent_amd.pprof import FrameKey, MappingKey, ProfileBuilder, ValueType
from parca_agent_amd.pprof.profile import decode_profile

u64 = st.integers(min_value=0, max_value=(1 << 64) - 1)
u32 = st.integers(min_value=0, max_value=(1 << 32) - 1)
dim3 = st.tuples(u32, u32, u32)
text = st.text(max_size=64)


@settings(max_examples=200, deadline=None)
@given(corr=u64, disp=u64, kid=u64, start=u64, end=u64, tid=u64,
       gpu=u32, pid=u32, grid=dim3, wg=dim3, priv=u32, group=u32)
def test_kernel_dispatch_roundtrip(corr, disp, kid, start, end, tid, gpu,
                                   pid, grid, wg, priv, group):
    e = ev.KernelDispatch(
        correlation_id=corr, dispatch_id=disp, kernel_id=kid,
        start_ns=start, end_ns=end, tid=tid, gpu_index=gpu, pid=pid,
        grid=grid, workgroup=wg, private_segment_size=priv,
        group_segment_size=group)
    d = ev.decode_kernel_dispatch(ev.encode_kernel_dispatch(e))
    assert d == e


@settings(max_examples=100, deadline=None)
@given(corr=u64, tid=u64, pid=u32,
       ips=st.lists(u64, max_size=64).map(tuple))
def test_launch_stack_roundtrip(corr, tid, pid, ips):
    e = ev.LaunchStack(correlation_id=corr, tid=tid, pid=pid, ips=ips)
    d = ev.decode_launch_stack(ev.encode_launch_stack(e))
    assert d == e


@settings(max_examples=100, deadline=None)
@given(kid=u64, co=u64, kobj=u64, name=text)
def test_kernel_symbol_roundtrip(kid, co, kobj, name):
    e = ev.KernelSymbol(kernel_id=kid, code_object_id=co,
                        kernel_object=kobj, name=name)
    d = ev.decode_kernel_symbol(ev.encode_kernel_symbol(e))
    assert d.kernel_id == kid and d.code_object_id == co
    assert d.name == name


frame = st.builds(
    FrameKey,
    address=u64,
    mapping=st.one_of(
        st.none(),
        st.builds(MappingKey, memory_start=u64, memory_limit=u64,
                  file_offset=u64, filename=st.text(min_size=1, max_size=32),
                  build_id=st.text(max_size=16))),
    function_name=text,
    source_file=text,
    line=st.integers(min_value=0, max_value=1 << 30),
)


@settings(max_examples=50, deadline=None)
@given(stacks=st.lists(
    st.tuples(st.lists(frame, min_size=1, max_size=12),
              st.integers(min_value=1, max_value=1 << 40)),
    min_size=1, max_size=20))
def test_pprof_totals_preserved(stacks):
    b = ProfileBuilder(sample_types=[ValueType("samples", "count")],
                       period_type=ValueType("cpu", "nanoseconds"),
                       period=52_631_578)
    total = 0
    for frames, value in stacks:
        b.add_sample(frames, [value])
        total += value
    p = decode_profile(b.serialize_gzip())
    assert sum(s["values"][0] for s in p.samples) == total
    # Every referenced location/function/mapping id must resolve.
    for s in p.samples:
        for lid in s["location_ids"]:
            loc = p.locations[lid]
            if loc.get("mapping_id"):
                assert loc["mapping_id"] in p.mappings
            for ln in loc.get("lines", []):
                assert ln["function_id"] in p.functions
    # String table invariant: index 0 is "".
    assert p.strings[0] == ""


@settings(max_examples=30, deadline=None)
@given(rows=st.lists(
    st.tuples(
        st.lists(u64.map(lambda a: a % (1 << 48)), min_size=1,
                 max_size=10),
        st.integers(min_value=1, max_value=1 << 40),
        st.dictionaries(st.sampled_from(["node", "comm", "pod"]),
                        st.text(max_size=16), max_size=3)),
    min_size=1, max_size=30))
def test_arrow_v2_fuzz(rows):
    """Random traces through the v2 writer must survive a pyarrow read
    with row count, value sums, and per-row stack depth intact."""
    import pyarrow as pa

    from parca_agent_amd.model import Frame, FrameType, MappingFile, Trace
    from parca_agent_amd.reporter.arrow_v2 import (
        SampleWriterV2,
        serialize_record,
    )

    w = SampleWriterV2()
    mf = MappingFile(path="/bin/x", file_id="cd" * 16)
    for addrs, value, labels in rows:
        t = Trace(frames=tuple(
            Frame(kind=FrameType.NATIVE, address=a, mapping=mf)
            for a in addrs))
        w.append_sample(t, labels, value, 1000,
                        sample_type="samples", sample_unit="count",
                        period_type="cpu", period_unit="nanoseconds",
                        period=52_631_578)
    batch = w.build_record()
    data = serialize_record(batch)
    with pa.ipc.open_stream(pa.BufferReader(data)) as rd:
        got = rd.read_all()
    assert got.num_rows == len(rows)
    assert sum(got.column("value").to_pylist()) == \
        sum(v for _, v, _ in rows)
    stacks = got.column("stacktrace").to_pylist()
    for (addrs, _, _), stack in zip(rows, stacks):
        assert len(stack) == len(addrs)
        assert [f["address"] for f in stack] == list(addrs)
