"""In-process fake Parca server for tests (reference test strategy:
fake event sources + schema verification without a real backend,
SURVEY.md §4)."""

from __future__ import annotations

import threading
from concurrent import futures
from typing import Dict, List

import grpc

from parca_agent_amd.reporter import protos

_identity = lambda b: b  # noqa: E731


class FakeParca:
    def __init__(self):
        self.arrow_payloads: List[bytes] = []
        self.raw_requests: List = []
        self.v1_records: List[bytes] = []
        self.want_stacktraces = False
        self.initiated: Dict[str, str] = {}  # build_id -> upload_id
        self.uploads: Dict[str, bytes] = {}  # upload_id -> data
        self.finished: List[str] = []
        self.known_build_ids: set = set()
        self.panics: List = []
        self.fail_next_write = 0
        self.seen_metadata = []  # invocation metadata per WriteArrow
        self._lock = threading.Lock()
        self._seq = 0

    # -- handlers ----------------------------------------------------------

    def write_arrow(self, request: bytes, context):
        with self._lock:
            self.seen_metadata.append(dict(context.invocation_metadata()))
            if self.fail_next_write > 0:
                self.fail_next_write -= 1
                context.abort(grpc.StatusCode.UNAVAILABLE, "try again")
            self.arrow_payloads.append(
                protos.decode_write_arrow_request(request))
        return b""

    def write_raw(self, request: bytes, context):
        with self._lock:
            self.raw_requests.append(
                protos.decode_write_raw_request(request))
        return b""

    def write_stream(self, request_iter, context):
        """Bidi v1 Write: record the samples record; optionally request
        stacktrace expansion (set want_stacktraces before the call)."""
        first = True
        for req in request_iter:
            record = protos.decode_write_request(req)
            with self._lock:
                self.v1_records.append(record)
            if not (first and self.want_stacktraces):
                first = False
                yield b""  # ack: no stacktraces wanted
                continue
            if True:
                first = False
                import io

                import pyarrow as pa

                table = pa.ipc.open_stream(io.BytesIO(record)).read_all()
                col = table.column("stacktrace_id")
                ids = set()
                for chunk in col.chunks:
                    if pa.types.is_run_end_encoded(chunk.type):
                        chunk = chunk.values
                    if pa.types.is_dictionary(chunk.type):
                        chunk = chunk.dictionary_decode()
                    ids.update(v for v in chunk.to_pylist()
                               if v is not None)
                batch = pa.record_batch(
                    [pa.array(sorted(ids), pa.binary())],
                    names=["stacktrace_id"])
                sink = io.BytesIO()
                with pa.ipc.new_stream(sink, batch.schema) as w:
                    w.write_batch(batch)
                from parca_agent_amd.pprof.proto import Writer

                w2 = Writer()
                w2.bytes(1, sink.getvalue())
                yield w2.getvalue()
        return

    def should_initiate_upload(self, request: bytes, context):
        build_id, _h, _force, _t = \
            protos.decode_should_initiate_upload_request(request)
        should = build_id not in self.known_build_ids
        return protos.encode_should_initiate_upload_response(
            should, "" if should else "already exists")

    def initiate_upload(self, request: bytes, context):
        build_id, _h, size, _f, type_ = \
            protos.decode_initiate_upload_request(request)
        with self._lock:
            self._seq += 1
            upload_id = f"upload-{self._seq}"
            self.initiated[build_id] = upload_id
        return protos.encode_initiate_upload_response(
            protos.UploadInstructions(
                build_id=build_id, upload_id=upload_id,
                upload_strategy=protos.UPLOAD_STRATEGY_GRPC, type=type_))

    def upload(self, request_iter, context):
        upload_id = None
        chunks = []
        for req in request_iter:
            kind, value = protos.decode_upload_request(req)
            if kind == "info":
                upload_id = value
            elif kind == "chunk":
                chunks.append(value)
        data = b"".join(chunks)
        with self._lock:
            self.uploads[upload_id or ""] = data
        return protos.encode_upload_response("", len(data))

    def mark_upload_finished(self, request: bytes, context):
        build_id, upload_id, _t = \
            protos.decode_mark_upload_finished_request(request)
        with self._lock:
            self.finished.append(build_id)
            self.known_build_ids.add(build_id)
        return b""

    def report_panic(self, request: bytes, context):
        with self._lock:
            self.panics.append(protos.decode_report_panic_request(request))
        return b""


def start_fake_parca(port: int = 0):
    """Returns (FakeParca, grpc.Server, address)."""
    fake = FakeParca()
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=8))

    def unary(fn):
        return grpc.unary_unary_rpc_method_handler(
            fn, request_deserializer=_identity, response_serializer=_identity)

    ps = grpc.method_handlers_generic_handler(
        protos.PROFILE_STORE_SERVICE,
        {
            "WriteArrow": unary(fake.write_arrow),
            "WriteRaw": unary(fake.write_raw),
            "Write": grpc.stream_stream_rpc_method_handler(
                fake.write_stream, request_deserializer=_identity,
                response_serializer=_identity),
        })
    di = grpc.method_handlers_generic_handler(
        protos.DEBUGINFO_SERVICE,
        {
            "ShouldInitiateUpload": unary(fake.should_initiate_upload),
            "InitiateUpload": unary(fake.initiate_upload),
            "Upload": grpc.stream_unary_rpc_method_handler(
                fake.upload, request_deserializer=_identity,
                response_serializer=_identity),
            "MarkUploadFinished": unary(fake.mark_upload_finished),
        })
    tel = grpc.method_handlers_generic_handler(
        protos.TELEMETRY_SERVICE,
        {"ReportPanic": unary(fake.report_panic)})
    server.add_generic_rpc_handlers((ps, di, tel))
    bound = server.add_insecure_port(f"127.0.0.1:{port}")
    server.start()
    return fake, server, f"127.0.0.1:{bound}"
