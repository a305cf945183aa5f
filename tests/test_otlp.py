"""OTLP log/metric egress tests with an in-process collector fake
(reference strategy: in-memory fake OTel logger, logrus_hook_test.go)."""

import logging
import time
from concurrent import futures

import grpc
import pytest

from parca_agent_amd.otlp import (
    GaugePoint,
    LOGS_SERVICE,
    METRICS_SERVICE,
    LogRecord,
    MetricsExporter,
    OTLPLogHandler,
    encode_logs_request,
    encode_metrics_request,
)
from parca_agent_amd.pprof.proto import iter_fields

_identity = lambda b: b  # noqa: E731


class FakeCollector:
    def __init__(self):
        self.log_payloads = []
        self.metric_payloads = []
        self.trace_payloads = []

    def export_logs(self, request, context):
        self.log_payloads.append(request)
        return b""

    def export_metrics(self, request, context):
        self.metric_payloads.append(request)
        return b""

    def export_traces(self, request, context):
        self.trace_payloads.append(request)
        return b""


@pytest.fixture
def collector():
    fake = FakeCollector()
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=4))

    def unary(fn):
        return grpc.unary_unary_rpc_method_handler(
            fn, request_deserializer=_identity,
            response_serializer=_identity)

    from parca_agent_amd.otlp import TRACE_SERVICE

    server.add_generic_rpc_handlers((
        grpc.method_handlers_generic_handler(
            LOGS_SERVICE, {"Export": unary(fake.export_logs)}),
        grpc.method_handlers_generic_handler(
            METRICS_SERVICE, {"Export": unary(fake.export_metrics)}),
        grpc.method_handlers_generic_handler(
            TRACE_SERVICE, {"Export": unary(fake.export_traces)}),
    ))
    port = server.add_insecure_port("127.0.0.1:0")
    server.start()
    yield fake, f"127.0.0.1:{port}"
    server.stop(grace=None)


def _walk_strings(buf, out):
    try:
        fields = list(iter_fields(buf))
    except Exception:
        return
    for _f, wt, v in fields:
        if wt == 2:
            try:
                out.append(v.decode("utf-8"))
            except UnicodeDecodeError:
                pass
            _walk_strings(v, out)


def test_encode_logs_request_contains_fields():
    payload = encode_logs_request(
        [LogRecord(time_ns=123, severity=9, severity_text="INFO",
                   body="hello world", attributes={"logger": "test"})],
        {"service.name": "parca-agent-amd"})
    strings = []
    _walk_strings(payload, strings)
    assert "hello world" in strings
    assert "service.name" in strings
    assert "INFO" in strings


def test_encode_metrics_request():
    payload = encode_metrics_request(
        [GaugePoint(name="gpu_ring_backlog", value=5.0, time_ns=1,
                    attributes={"gpu": "0"}),
         GaugePoint(name="samples_total", value=100.0, time_ns=1,
                    is_sum=True)],
        {"service.name": "x"})
    strings = []
    _walk_strings(payload, strings)
    assert "gpu_ring_backlog" in strings
    assert "samples_total" in strings


def test_log_handler_exports(collector):
    fake, addr = collector
    channel = grpc.insecure_channel(addr)
    handler = OTLPLogHandler(channel, {"service.name": "test"},
                             flush_interval=0.05)
    logger = logging.getLogger("otlp-test")
    logger.setLevel(logging.INFO)
    logger.addHandler(handler)
    logger.info("profiled %d things", 42)
    logger.info("skipped", extra={"otlp_skip": True})
    deadline = time.time() + 5
    while handler.exported < 1 and time.time() < deadline:
        time.sleep(0.05)
    logger.removeHandler(handler)
    handler.close()
    assert handler.exported == 1  # otlp_skip record suppressed
    strings = []
    _walk_strings(fake.log_payloads[0], strings)
    assert "profiled 42 things" in strings
    assert not any("skipped" == s for s in strings)


def test_metrics_exporter_producers(collector):
    fake, addr = collector
    channel = grpc.insecure_channel(addr)
    exp = MetricsExporter(channel, {"service.name": "test"}, interval=999)
    exp.register(lambda: [GaugePoint(name="m1", value=1.0,
                                     time_ns=time.time_ns())])
    exp.export_once()
    assert exp.exported == 1
    assert fake.metric_payloads


def test_span_exporter(collector):
    from parca_agent_amd.otlp import Span, SpanExporter, TRACE_SERVICE
    fake, addr = collector
    channel = grpc.insecure_channel(addr)
    exp = SpanExporter(channel, {"service.name": "test"},
                       flush_interval=999)
    exp.add(Span(name="my_probe", start_ns=100, end_ns=5100,
                 attributes={"pid": "42"}))
    exp.flush_batch()
    exp.close()
    assert exp.exported == 1 and exp.errors == 0
    strings = []
    _walk_strings(fake.trace_payloads[0], strings)
    assert "my_probe" in strings
    assert "node.callback_scope" in strings
