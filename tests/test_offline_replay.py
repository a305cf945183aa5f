"""Offline-mode record + replay end-to-end (reference:
uploader/log_uploader.go): record batches to .padata, rotate to zstd,
replay to a (fake) server, verify deletion and payload fidelity."""

import pyarrow as pa

from parca_agent_amd.flags import Flags
from parca_agent_amd.model import Frame, FrameType, MappingFile, Trace, TraceEventMeta
from parca_agent_amd.reporter import OfflineLogDestination, Reporter
from parca_agent_amd.reporter.grpc_client import ParcaClient, RetryingCaller
from parca_agent_amd.uploader import offline_upload

import grpc

from fake_parca import start_fake_parca


def test_offline_record_then_replay(tmp_path, monkeypatch):
    store = tmp_path / "offline"
    dest = OfflineLogDestination(str(store), rotation_interval=1e9)
    rep = Reporter([dest])
    trace = Trace(frames=(Frame(kind=FrameType.NATIVE, address=0x10,
                                mapping=MappingFile(file_id="f" * 32,
                                                    path="/bin/app")),))
    for i in range(3):
        rep.report_trace_event(trace, TraceEventMeta(pid=1, timestamp_ns=i))
        rep.flush()
    dest.close()  # rotates to .zst

    fake, server, addr = start_fake_parca()
    try:
        flags = Flags()
        flags.offline_mode.storage_path = str(store)
        flags.offline_mode.upload = True
        flags.remote_store.address = addr
        flags.remote_store.insecure = True
        flags.validate()

        rc = offline_upload(flags)
        assert rc == 0
        assert len(fake.arrow_payloads) == 3
        table = pa.ipc.open_stream(fake.arrow_payloads[0]).read_all()
        assert table.num_rows == 1
        # Files removed after successful upload.
        assert not any(store.iterdir())
    finally:
        server.stop(grace=None)
