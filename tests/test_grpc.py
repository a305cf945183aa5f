"""gRPC client + uploader tests against an in-process fake Parca
(reference strategy: no real backend in unit tests)."""

import subprocess
import textwrap
import time

import grpc
import pyarrow as pa
import pytest

from parca_agent_amd.model import Frame, FrameType, MappingFile, Trace
from parca_agent_amd.reporter import Reporter, protos
from parca_agent_amd.reporter.grpc_client import (
    ParcaClient,
    RemoteStoreDestination,
    RetryingCaller,
)
from parca_agent_amd.reporter.uploader import DebuginfoUploader, UploadItem

from fake_parca import start_fake_parca


@pytest.fixture
def parca():
    fake, server, addr = start_fake_parca()
    yield fake, addr
    server.stop(grace=None)


def _client(addr):
    channel = grpc.insecure_channel(addr)
    return ParcaClient(channel, retry=RetryingCaller(
        max_tries=3, base_delay=0.01, per_try_timeout=10))


def test_write_arrow_roundtrip(parca):
    fake, addr = parca
    client = _client(addr)
    client.write_arrow(b"arrow-ipc-bytes")
    assert fake.arrow_payloads == [b"arrow-ipc-bytes"]


def test_write_arrow_retry_on_unavailable(parca):
    fake, addr = parca
    fake.fail_next_write = 2
    client = _client(addr)
    client.write_arrow(b"payload")
    assert fake.arrow_payloads == [b"payload"]


def test_remote_store_destination_sends_decodable_record(parca):
    fake, addr = parca
    dest = RemoteStoreDestination.__new__(RemoteStoreDestination)
    dest.flags = None
    dest.write_format = "arrow_v2"
    dest.client = _client(addr)
    dest.bytes_sent = 0
    dest.batches_sent = 0
    dest.errors = 0

    rep = Reporter([dest])
    trace = Trace(frames=(
        Frame(kind=FrameType.NATIVE, address=0x10,
              mapping=MappingFile(file_id="f" * 32, path="/bin/x")),))
    from parca_agent_amd.model import TraceEventMeta

    rep.report_trace_event(trace, TraceEventMeta(pid=1, timestamp_ns=5))
    rep.flush()

    assert dest.batches_sent == 1
    payload = fake.arrow_payloads[0]
    table = pa.ipc.open_stream(payload).read_all()
    assert table.num_rows == 1
    assert table.schema.metadata[b"parca_write_schema_version"] == b"v2"


def test_write_raw(parca):
    fake, addr = parca
    client = _client(addr)
    client.write_raw([protos.Label("job", "oomprof")], b"pprof-bytes")
    [series] = fake.raw_requests
    [(labels, profiles)] = series
    assert labels[0].name == "job"
    assert profiles == [b"pprof-bytes"]


def test_report_panic(parca):
    fake, addr = parca
    client = _client(addr)
    client.report_panic("panic: boom", {"version": "0.1.0"})
    [(stderr, md)] = fake.panics
    assert "boom" in stderr
    assert md["version"] == "0.1.0"


def test_debuginfo_upload_protocol(parca, tmp_path):
    fake, addr = parca
    client = _client(addr)
    src = tmp_path / "lib.c"
    src.write_text("int answer(void){return 42;}\n"
                   "int question(void){return answer()-1;}\n")
    binary = tmp_path / "lib.so"
    subprocess.run(["gcc", "-g", "-shared", "-fPIC", str(src), "-o",
                    str(binary)], check=True)

    up = DebuginfoUploader(client, max_parallel=2, strip=True)
    up.start()
    assert up.enqueue(UploadItem(build_id="bid-1", hash="h1",
                                 path=str(binary)))
    # duplicate suppressed
    assert not up.enqueue(UploadItem(build_id="bid-1", hash="h1",
                                     path=str(binary)))
    deadline = time.time() + 10
    while up.uploaded < 1 and time.time() < deadline:
        time.sleep(0.05)
    up.stop()
    assert up.uploaded == 1
    assert fake.finished == ["bid-1"]
    data = fake.uploads["upload-1"]
    # Stripped ELF: valid, keeps .debug_* and symtab, drops .text bytes.
    from parca_agent_amd.elf import ELFFile

    elf = ELFFile.from_bytes(data)
    names = {s.name for s in elf.sections}
    assert any(n.startswith(".debug_") for n in names)
    assert ".symtab" in names
    text = elf.section(".text")
    assert text is not None and text.sh_type == 8  # SHT_NOBITS
    # Symbols still resolvable from the stripped file.
    syms = {s.name for s in elf.symbols()}
    assert "answer" in syms
    # And it is smaller than the original.
    assert len(data) < binary.stat().st_size


def test_upload_already_known_skipped(parca, tmp_path):
    fake, addr = parca
    fake.known_build_ids.add("known")
    client = _client(addr)
    up = DebuginfoUploader(client, max_parallel=1)
    up.start()
    up.enqueue(UploadItem(build_id="known", hash="h", data=b"x"))
    deadline = time.time() + 5
    while up.skipped < 1 and time.time() < deadline:
        time.sleep(0.05)
    up.stop()
    assert up.uploaded == 0
    assert fake.initiated == {}


def test_in_memory_code_object_upload(parca):
    fake, addr = parca
    client = _client(addr)
    up = DebuginfoUploader(client, max_parallel=1)
    up.start()
    payload = b"\x7fELFfake-code-object" * 1000
    up.enqueue(UploadItem(build_id="codeobj-abc", hash="h2", data=payload,
                          type=protos.DEBUGINFO_TYPE_UNSPECIFIED))
    deadline = time.time() + 5
    while up.uploaded < 1 and time.time() < deadline:
        time.sleep(0.05)
    up.stop()
    assert fake.uploads["upload-1"] == payload


def _dest_with_format(addr, fmt):
    dest = RemoteStoreDestination.__new__(RemoteStoreDestination)
    dest.flags = None
    dest.write_format = fmt
    dest.client = _client(addr)
    dest.bytes_sent = 0
    dest.batches_sent = 0
    dest.errors = 0
    return dest


def _sample_batch():
    from parca_agent_amd.model import SampleType, TraceEventMeta
    from parca_agent_amd.reporter.reporter import PendingSample

    trace = Trace(frames=(
        Frame(kind=FrameType.NATIVE, address=0x10,
              mapping=MappingFile(file_id="f" * 32, path="/bin/x"),
              function_name="fn"),))
    return [PendingSample(
        trace=trace, labels={"node": "n1"}, value=1, timestamp_ns=5,
        sample_type=SampleType("samples", "count", "cpu", "nanoseconds"),
        period=52_631_578)]


def test_write_format_arrow_v1_two_phase(parca):
    fake, addr = parca
    fake.want_stacktraces = True
    dest = _dest_with_format(addr, "arrow_v1")
    dest.write_batch(_sample_batch())
    # Two records on the stream: samples, then requested locations.
    assert len(fake.v1_records) == 2
    t1 = pa.ipc.open_stream(fake.v1_records[0]).read_all()
    assert t1.schema.metadata[b"parca_write_schema_version"] == b"v1"
    assert "stacktrace_id" in t1.schema.names
    t2 = pa.ipc.open_stream(fake.v1_records[1]).read_all()
    assert "locations" in t2.schema.names
    locs = t2.column("locations")[0].as_py()
    assert locs[0]["address"] == 0x10


def test_write_format_arrow_v1_no_resolution(parca):
    fake, addr = parca
    fake.want_stacktraces = False
    dest = _dest_with_format(addr, "arrow_v1")
    dest.write_batch(_sample_batch())
    assert len(fake.v1_records) == 1


def test_write_format_pprof(parca):
    fake, addr = parca
    dest = _dest_with_format(addr, "pprof")
    dest.write_batch(_sample_batch())
    [series] = fake.raw_requests
    [(labels, profiles)] = series
    names = {l.name: l.value for l in labels}
    assert names["__name__"] == "samples"
    from parca_agent_amd.pprof import decode_profile

    prof = decode_profile(profiles[0])
    assert prof.sample_types[0].type == "samples"


def test_external_debuginfo_directory_lookup(tmp_path):
    """--debuginfo-directories: split DWARF found via the GNU build-id
    convention wins over re-stripping the binary (reference
    flags.go:375)."""
    from parca_agent_amd.reporter.uploader import DebuginfoUploader

    bid = "deadbeef" * 5
    droot = tmp_path / "debugroot"
    dfile = droot / ".build-id" / bid[:2] / (bid[2:] + ".debug")
    dfile.parent.mkdir(parents=True)
    dfile.write_bytes(b"SPLIT-DWARF-PAYLOAD")

    up = DebuginfoUploader(client=None, max_parallel=1,
                           debug_directories=[str(droot)])
    assert up._find_external_debug(bid) == str(dfile)
    assert up._find_external_debug("f" * 40) is None
    assert up._find_external_debug("not-hex!") is None


def test_custom_grpc_headers_reach_the_server():
    """--remote-store-grpc-headers key=value must arrive as call
    metadata on every RPC (reference flags/grpc.go:179-183)."""
    from parca_agent_amd.flags import Flags
    from parca_agent_amd.reporter.grpc_client import (ParcaClient,
                                                      build_channel)
    from tests.fake_parca import start_fake_parca

    fake, server, addr = start_fake_parca()
    try:
        f = Flags()
        f.remote_store.address = addr if ":" in str(addr) \
            else f"127.0.0.1:{addr}"
        f.remote_store.insecure = True
        f.remote_store.grpc_headers = ["x-scope-orgid=tenant-7",
                                       "x-team=infra"]
        client = ParcaClient(build_channel(f))
        client.write_arrow(b"\x00")
        md = fake.seen_metadata[-1]
        assert md.get("x-scope-orgid") == "tenant-7"
        assert md.get("x-team") == "infra"
    finally:
        server.stop(0)
