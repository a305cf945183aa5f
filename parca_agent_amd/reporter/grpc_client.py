"""gRPC connection factory + remote-store destination.

Mirrors the reference's connection behaviour (reference: flags/grpc.go):
TLS or insecure channels, bearer-token per-call credentials, exponential
retry with jitter (×10, flags/grpc.go:143-170), 32 MiB message caps
(flags.go:360-361). Methods are built with grpc generic stubs over the
hand-rolled codecs in protos.py — no protoc needed.
"""

from __future__ import annotations

import logging
import random
import time
from typing import Callable, List, Optional

import grpc

from . import protos
from .reporter import PendingSample, build_arrow_record

log = logging.getLogger("parca_agent_amd.grpc")

_identity = lambda b: b  # noqa: E731


def build_channel(flags) -> grpc.Channel:
    rs = flags.remote_store
    options = [
        ("grpc.max_receive_message_length", rs.grpc_max_call_recv_msg_size),
        ("grpc.max_send_message_length", rs.grpc_max_call_send_msg_size),
        ("grpc.keepalive_time_ms", 30000),
    ]
    if rs.insecure:
        channel = grpc.insecure_channel(rs.address, options=options)
    else:
        # mTLS client auth when configured (reference remote-store
        # client-cert/client-key flags).
        key = cert = None
        if getattr(rs, "client_cert", "") and getattr(rs, "client_key", ""):
            with open(rs.client_key, "rb") as fh:
                key = fh.read()
            with open(rs.client_cert, "rb") as fh:
                cert = fh.read()
        creds = grpc.ssl_channel_credentials(private_key=key,
                                             certificate_chain=cert)
        if rs.bearer_token:
            call_creds = grpc.access_token_call_credentials(rs.bearer_token)
            creds = grpc.composite_channel_credentials(creds, call_creds)
        channel = grpc.secure_channel(rs.address, creds, options=options)
    headers = [tuple(h.split("=", 1)) for h in
               getattr(rs, "grpc_headers", []) if "=" in h]
    if headers:
        channel = grpc.intercept_channel(
            channel, _HeaderInterceptor(headers))
    if getattr(rs, "rpc_logging_enable", False):
        channel = grpc.intercept_channel(channel, _LoggingInterceptor())
    return channel


class _LoggingInterceptor(grpc.UnaryUnaryClientInterceptor):
    """--remote-store-rpc-logging-enable: one debug line per unary RPC
    with method and outcome (reference grpc.go logging interceptor)."""

    def intercept_unary_unary(self, cont, details, request):
        import time as _t

        t0 = _t.monotonic()
        try:
            result = cont(details, request)
            log.debug("rpc %s ok (%.1f ms)", details.method,
                      1e3 * (_t.monotonic() - t0))
            return result
        except grpc.RpcError as e:
            log.debug("rpc %s failed: %s", details.method, e.code())
            raise


class _HeaderInterceptor(grpc.UnaryUnaryClientInterceptor,
                         grpc.StreamUnaryClientInterceptor,
                         grpc.UnaryStreamClientInterceptor,
                         grpc.StreamStreamClientInterceptor):
    """Attach custom metadata to every call (reference
    --remote-store-grpc-headers key=value)."""

    def __init__(self, headers):
        self._headers = headers

    def _with(self, details):
        md = list(details.metadata or []) + self._headers
        return details._replace(metadata=md) if hasattr(
            details, "_replace") else details

    def intercept_unary_unary(self, cont, details, request):
        return cont(self._with(details), request)

    def intercept_stream_unary(self, cont, details, request_it):
        return cont(self._with(details), request_it)

    def intercept_unary_stream(self, cont, details, request):
        return cont(self._with(details), request)

    def intercept_stream_stream(self, cont, details, request_it):
        return cont(self._with(details), request_it)


class RetryingCaller:
    """Exponential-backoff retry wrapper (flags/grpc.go:143-170: 10 tries,
    jittered exponential, capped per-try timeout)."""

    RETRYABLE = {
        grpc.StatusCode.UNAVAILABLE,
        grpc.StatusCode.DEADLINE_EXCEEDED,
        grpc.StatusCode.RESOURCE_EXHAUSTED,
        grpc.StatusCode.ABORTED,
    }

    def __init__(self, max_tries: int = 10, base_delay: float = 0.2,
                 max_delay: float = 30.0, per_try_timeout: float = 120.0):
        self.max_tries = max_tries
        self.base_delay = base_delay
        self.max_delay = max_delay
        self.per_try_timeout = per_try_timeout

    def call(self, fn: Callable, *args, **kwargs):
        delay = self.base_delay
        last = None
        for attempt in range(self.max_tries):
            try:
                return fn(*args, timeout=self.per_try_timeout, **kwargs)
            except grpc.RpcError as e:
                if e.code() not in self.RETRYABLE:
                    raise
                last = e
                sleep = min(delay, self.max_delay) * random.uniform(0.5, 1.5)
                log.debug("rpc retry %d after %s: %.2fs",
                          attempt + 1, e.code(), sleep)
                time.sleep(sleep)
                delay *= 2
        assert last is not None
        raise last


class ParcaClient:
    """ProfileStore / Debuginfo / Telemetry over one shared channel
    (reference: one grpc.ClientConn for everything, main.go:210-218)."""

    def __init__(self, channel: grpc.Channel,
                 retry: Optional[RetryingCaller] = None,
                 extra_metadata: Optional[List] = None) -> None:
        self.channel = channel
        self.retry = retry or RetryingCaller()
        self.metadata = tuple(extra_metadata or ())
        p = protos.PROFILE_STORE_SERVICE
        self._write_arrow = channel.unary_unary(
            f"/{p}/WriteArrow",
            request_serializer=_identity, response_deserializer=_identity)
        self._write = channel.stream_stream(
            f"/{p}/Write",
            request_serializer=_identity, response_deserializer=_identity)
        self._write_raw = channel.unary_unary(
            f"/{p}/WriteRaw",
            request_serializer=_identity, response_deserializer=_identity)
        d = protos.DEBUGINFO_SERVICE
        self._should_initiate = channel.unary_unary(
            f"/{d}/ShouldInitiateUpload",
            request_serializer=_identity, response_deserializer=_identity)
        self._initiate = channel.unary_unary(
            f"/{d}/InitiateUpload",
            request_serializer=_identity, response_deserializer=_identity)
        self._upload = channel.stream_unary(
            f"/{d}/Upload",
            request_serializer=_identity, response_deserializer=_identity)
        self._mark_finished = channel.unary_unary(
            f"/{d}/MarkUploadFinished",
            request_serializer=_identity, response_deserializer=_identity)
        t = protos.TELEMETRY_SERVICE
        self._report_panic = channel.unary_unary(
            f"/{t}/ReportPanic",
            request_serializer=_identity, response_deserializer=_identity)

    # -- profilestore ------------------------------------------------------

    def write_arrow(self, ipc_buffer: bytes) -> None:
        req = protos.encode_write_arrow_request(ipc_buffer)
        self.retry.call(self._write_arrow, req, metadata=self.metadata)

    def write_raw(self, labels, raw_profile: bytes) -> None:
        req = protos.encode_write_raw_request([labels], [[raw_profile]])
        self.retry.call(self._write_raw, req, metadata=self.metadata)

    # -- debuginfo ---------------------------------------------------------

    def should_initiate_upload(self, build_id: str, hash_: str,
                               type_: int = 0):
        req = protos.encode_should_initiate_upload_request(
            build_id, hash_, type_=type_)
        resp = self.retry.call(self._should_initiate, req,
                               metadata=self.metadata)
        return protos.decode_should_initiate_upload_response(resp)

    def initiate_upload(self, build_id: str, hash_: str, size: int,
                        type_: int = 0) -> protos.UploadInstructions:
        req = protos.encode_initiate_upload_request(build_id, hash_, size,
                                                    type_=type_)
        resp = self.retry.call(self._initiate, req, metadata=self.metadata)
        return protos.decode_initiate_upload_response(resp)

    def upload(self, upload_id: str, data: bytes,
               chunk_size: int = 8 * 1024 * 1024, type_: int = 0):
        """8 MB chunked client-streaming upload (reference:
        grpc_upload_client.go:33-39)."""

        def requests():
            yield protos.encode_upload_request_info(upload_id, type_)
            for off in range(0, len(data), chunk_size):
                yield protos.encode_upload_request_chunk(
                    data[off : off + chunk_size])

        resp = self.retry.call(self._upload, requests(),
                               metadata=self.metadata)
        return protos.decode_upload_response(resp)

    def mark_upload_finished(self, build_id: str, upload_id: str,
                             type_: int = 0) -> None:
        req = protos.encode_mark_upload_finished_request(
            build_id, upload_id, type_)
        self.retry.call(self._mark_finished, req, metadata=self.metadata)

    # -- telemetry ---------------------------------------------------------

    def report_panic(self, stderr: str, metadata: dict) -> None:
        req = protos.encode_report_panic_request(stderr, metadata)
        self.retry.call(self._report_panic, req, metadata=self.metadata)


class RemoteStoreDestination:
    """Reporter destination shipping batches to the ProfileStore in the
    configured wire format (--remote-store-write-format):

    - arrow_v2 (default): self-contained WriteArrow
      (reference: reportDataToBackendV2, parca_reporter.go:2150-2190)
    - arrow_v1: the two-phase Write stream — samples record out, the
      server may answer with the stacktrace IDs it wants expanded, we
      send the locations record (parca_reporter.go:1667-1803)
    - pprof: WriteRaw with per-sample-type pprof payloads (the legacy
      path oomprof uses, oom/oomprof.go:100-116)
    """

    def __init__(self, flags, client: Optional[ParcaClient] = None) -> None:
        self.flags = flags
        self.write_format = getattr(
            flags.remote_store, "write_format", "arrow_v2") \
            if flags is not None else "arrow_v2"
        if client is None:
            client = ParcaClient(build_channel(flags))
        self.client = client
        self.bytes_sent = 0
        self.batches_sent = 0
        self.errors = 0

    def write_batch(self, samples: List[PendingSample]) -> None:
        try:
            if self.write_format == "arrow_v1":
                self._write_v1(samples)
            elif self.write_format == "pprof":
                self._write_pprof(samples)
            else:
                self._write_v2(samples)
        except grpc.RpcError:
            self.errors += 1
            raise
        self.batches_sent += 1

    def _write_v2(self, samples: List[PendingSample]) -> None:
        from .arrow_v2 import serialize_record

        record = build_arrow_record(samples)
        if record.num_rows == 0:
            return
        payload = serialize_record(record)
        self.client.write_arrow(payload)
        self.bytes_sent += len(payload)

    def _write_v1(self, samples: List[PendingSample]) -> None:
        import io

        import pyarrow as pa

        from . import protos
        from .arrow_v1 import (
            LocationsWriterV1,
            SampleWriterV1,
            decode_requested_ids,
        )

        writer = SampleWriterV1()
        for s in samples:
            writer.append_sample(
                s.trace, s.labels, s.value, s.timestamp_ns,
                sample_type=s.sample_type.sample_type,
                sample_unit=s.sample_type.sample_unit,
                period_type=s.sample_type.period_type,
                period_unit=s.sample_type.period_unit,
                period=s.period, duration_ns=s.duration_ns)
        if writer.n_rows == 0:
            return
        batch = writer.build_record()
        sink = io.BytesIO()
        with pa.ipc.new_stream(sink, batch.schema) as w:
            w.write_batch(batch)
        payload = sink.getvalue()

        # Bidi two-phase: a queue-backed request iterator lets us answer
        # the server's stacktrace-ID request on the SAME stream.
        import queue as queue_mod

        requests: "queue_mod.Queue" = queue_mod.Queue()
        requests.put(protos.encode_write_request(payload))

        def request_iter():
            while True:
                item = requests.get()
                if item is None:
                    return
                yield item

        call = self.client._write(request_iter(), timeout=60,
                                  metadata=self.client.metadata)
        self.bytes_sent += len(payload)
        try:
            # The server acknowledges each record; a non-empty response
            # carries the stacktrace IDs it wants expanded
            # (parca_reporter.go:1698-1800).
            resp = next(iter(call), None)
            wanted = decode_requested_ids(
                protos.decode_write_response(resp)) if resp else set()
            if wanted:
                traces = writer.traces()
                lw = LocationsWriterV1()
                for tid_ in wanted:
                    if tid_ in traces:
                        lw.append_stacktrace(tid_, traces[tid_])
                loc_batch = lw.build_record()
                sink2 = io.BytesIO()
                with pa.ipc.new_stream(sink2, loc_batch.schema) as w:
                    w.write_batch(loc_batch)
                requests.put(protos.encode_write_request(sink2.getvalue()))
                next(iter(call), None)  # ack of the locations record
        finally:
            requests.put(None)

    def _write_pprof(self, samples: List[PendingSample]) -> None:
        from . import protos
        from .destinations import samples_to_pprof

        node = ""
        if samples:
            node = samples[0].labels.get("node", "")
        for stype, data in samples_to_pprof(samples).items():
            labels = [protos.Label("__name__", stype)]
            if node:
                labels.append(protos.Label("node", node))
            self.client.write_raw(labels, data)
            self.bytes_sent += len(data)

    def close(self) -> None:
        self.client.channel.close()
