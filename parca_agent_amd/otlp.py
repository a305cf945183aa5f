"""OTLP log/metric egress over gRPC — hand-rolled protos.

Reference components: the logrus->OTLP hook with its `otlp_skip` loop
guard (reference: reporter/logrus_hook.go:31-91), the LoggerProvider
batching (log_streamer.go:40-78), and the producer-based OTLP metrics
exporter scaffold (metricexport/exporter.go:44-158). Encodings follow
the public opentelemetry-proto shapes
(opentelemetry.proto.collector.{logs,metrics}.v1 Export*ServiceRequest).
"""

from __future__ import annotations

import logging
import threading
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

from .pprof.proto import Writer

LOGS_SERVICE = "opentelemetry.proto.collector.logs.v1.LogsService"
METRICS_SERVICE = "opentelemetry.proto.collector.metrics.v1.MetricsService"
TRACE_SERVICE = "opentelemetry.proto.collector.trace.v1.TraceService"

_SEVERITY = {
    logging.DEBUG: (5, "DEBUG"),
    logging.INFO: (9, "INFO"),
    logging.WARNING: (13, "WARN"),
    logging.ERROR: (17, "ERROR"),
    logging.CRITICAL: (21, "FATAL"),
}

# Escape hatch: records carrying this attribute are not exported,
# breaking the log->export->log feedback loop (logrus_hook.go's
# otlp_skip).
OTLP_SKIP = "otlp_skip"


def _encode_any_string(value: str) -> Writer:
    w = Writer()
    w.string(1, value)  # AnyValue.string_value
    return w


def _encode_keyvalue(key: str, value: str) -> Writer:
    w = Writer()
    w.string(1, key)
    w.message(2, _encode_any_string(value))
    return w


def encode_resource(attributes: Dict[str, str]) -> Writer:
    w = Writer()
    for k, v in sorted(attributes.items()):
        w.message(1, _encode_keyvalue(k, v))
    return w


@dataclass
class LogRecord:
    time_ns: int
    severity: int
    severity_text: str
    body: str
    attributes: Dict[str, str] = field(default_factory=dict)


def encode_logs_request(records: List[LogRecord],
                        resource_attrs: Dict[str, str],
                        scope_name: str = "parca-agent-amd") -> bytes:
    """ExportLogsServiceRequest{resource_logs=1{resource=1,
    scope_logs=2{scope=1{name=1}, log_records=2{...}}}}."""
    scope_logs = Writer()
    scope = Writer()
    scope.string(1, scope_name)
    scope_logs.message(1, scope)
    for r in records:
        lr = Writer()
        lr.fixed64(1, r.time_ns)          # time_unix_nano
        lr.varint(2, r.severity)          # severity_number
        lr.string(3, r.severity_text)
        lr.message(5, _encode_any_string(r.body))  # body
        for k, v in r.attributes.items():
            lr.message(6, _encode_keyvalue(k, v))
        scope_logs.message(2, lr)
    resource_logs = Writer()
    resource_logs.message(1, encode_resource(resource_attrs))
    resource_logs.message(2, scope_logs)
    out = Writer()
    out.message(1, resource_logs)
    return out.getvalue()


@dataclass
class GaugePoint:
    name: str
    value: float
    time_ns: int
    attributes: Dict[str, str] = field(default_factory=dict)
    description: str = ""
    unit: str = ""
    is_sum: bool = False  # cumulative sum instead of gauge


def encode_metrics_request(points: List[GaugePoint],
                           resource_attrs: Dict[str, str],
                           scope_name: str = "parca-agent-amd") -> bytes:
    """ExportMetricsServiceRequest{resource_metrics=1{resource=1,
    scope_metrics=2{scope=1, metrics=2{name=1, description=2, unit=3,
    gauge=5{data_points=1}|sum=7{data_points=1, temporality=2,
    monotonic=3}}}}}."""
    scope_metrics = Writer()
    scope = Writer()
    scope.string(1, scope_name)
    scope_metrics.message(1, scope)
    for p in points:
        metric = Writer()
        metric.string(1, p.name)
        metric.string(2, p.description)
        metric.string(3, p.unit)
        dp = Writer()
        for k, v in p.attributes.items():
            dp.message(7, _encode_keyvalue(k, v))
        dp.fixed64(3, p.time_ns)
        dp.double(4, p.value)  # as_double
        data = Writer()
        data.message(1, dp)
        if p.is_sum:
            data.varint(2, 2)  # CUMULATIVE
            data.bool(3, True)  # monotonic
            metric.message(7, data)
        else:
            metric.message(5, data)
        scope_metrics.message(2, metric)
    resource_metrics = Writer()
    resource_metrics.message(1, encode_resource(resource_attrs))
    resource_metrics.message(2, scope_metrics)
    out = Writer()
    out.message(1, resource_metrics)
    return out.getvalue()


@dataclass
class Span:
    name: str
    start_ns: int
    end_ns: int
    attributes: Dict[str, str] = field(default_factory=dict)
    trace_id: bytes = b""
    span_id: bytes = b""


def encode_trace_request(spans: List[Span],
                         resource_attrs: Dict[str, str],
                         scope_name: str = "node.callback_scope") -> bytes:
    """ExportTraceServiceRequest{resource_spans=1{resource=1,
    scope_spans=2{scope=1, spans=2{trace_id=1, span_id=2, name=5,
    kind=6, start=7, end=8, attributes=9}}}}. Scope name mirrors the
    reference's probe span scope (probes/service.go:180-199)."""
    import os as _os

    scope_spans = Writer()
    scope = Writer()
    scope.string(1, scope_name)
    scope_spans.message(1, scope)
    for s in spans:
        sw = Writer()
        sw.bytes(1, s.trace_id or _os.urandom(16))
        sw.bytes(2, s.span_id or _os.urandom(8))
        sw.string(5, s.name)
        sw.varint(6, 1)  # SPAN_KIND_INTERNAL
        sw.fixed64(7, s.start_ns)
        sw.fixed64(8, s.end_ns)
        for k, v in s.attributes.items():
            sw.message(9, _encode_keyvalue(k, v))
        scope_spans.message(2, sw)
    resource_spans = Writer()
    resource_spans.message(1, encode_resource(resource_attrs))
    resource_spans.message(2, scope_spans)
    out = Writer()
    out.message(1, resource_spans)
    return out.getvalue()


class SpanExporter:
    """Batched OTLP span export for probe fires (the reference's twin
    TracerProvider, trace_exporter.go:36-75)."""

    def __init__(self, channel, resource_attrs: Dict[str, str],
                 flush_interval: float = 0.25, max_batch: int = 512) -> None:
        self._export = channel.unary_unary(
            f"/{TRACE_SERVICE}/Export",
            request_serializer=lambda b: b,
            response_deserializer=lambda b: b)
        self.resource_attrs = resource_attrs
        self.flush_interval = flush_interval
        self.max_batch = max_batch
        self._mu = threading.Lock()
        self._buffer: List[Span] = []
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name="otlp-spans")
        self._thread.start()
        self.exported = 0
        self.errors = 0

    def add(self, span: Span) -> None:
        with self._mu:
            self._buffer.append(span)
            if len(self._buffer) > self.max_batch * 4:
                del self._buffer[: self.max_batch]

    def _run(self) -> None:
        while not self._stop.wait(self.flush_interval):
            self.flush_batch()

    def flush_batch(self) -> None:
        with self._mu:
            batch, self._buffer = self._buffer[: self.max_batch], \
                self._buffer[self.max_batch:]
        if not batch:
            return
        try:
            payload = encode_trace_request(batch, self.resource_attrs)
            self._export(payload, timeout=10)
            self.exported += len(batch)
        except Exception:
            self.errors += 1

    def close(self) -> None:
        self._stop.set()
        self._thread.join(timeout=2)
        self.flush_batch()


class OTLPLogHandler(logging.Handler):
    """logging.Handler shipping agent logs as OTLP log records over the
    shared gRPC channel, batched (250 ms / 512 records like the
    reference's BatchSpanProcessor settings, log_streamer.go:40-78)."""

    def __init__(self, channel, resource_attrs: Dict[str, str],
                 flush_interval: float = 0.25, max_batch: int = 512) -> None:
        super().__init__()
        self._export = channel.unary_unary(
            f"/{LOGS_SERVICE}/Export",
            request_serializer=lambda b: b,
            response_deserializer=lambda b: b)
        self.resource_attrs = resource_attrs
        self.flush_interval = flush_interval
        self.max_batch = max_batch
        self._mu = threading.Lock()
        self._buffer: List[LogRecord] = []
        self._stop = threading.Event()
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name="otlp-logs")
        self._thread.start()
        self.exported = 0
        self.errors = 0

    def emit(self, record: logging.LogRecord) -> None:
        if getattr(record, OTLP_SKIP, False):
            return
        sev, sev_text = _SEVERITY.get(
            record.levelno - record.levelno % 10,
            (9, record.levelname))
        lr = LogRecord(
            time_ns=int(record.created * 1e9),
            severity=sev, severity_text=sev_text,
            body=record.getMessage(),
            attributes={"logger": record.name})
        with self._mu:
            self._buffer.append(lr)
            if len(self._buffer) > self.max_batch * 4:
                del self._buffer[: self.max_batch]  # shed, never block

    def _run(self) -> None:
        while not self._stop.wait(self.flush_interval):
            self.flush_batch()

    def flush_batch(self) -> None:
        with self._mu:
            batch, self._buffer = self._buffer[: self.max_batch], \
                self._buffer[self.max_batch:]
        if not batch:
            return
        try:
            payload = encode_logs_request(batch, self.resource_attrs)
            self._export(payload, timeout=10)
            self.exported += len(batch)
        except Exception:
            self.errors += 1  # fire-and-forget; never log (loop risk)

    def close(self) -> None:
        self._stop.set()
        self._thread.join(timeout=2)
        self.flush_batch()
        super().close()


class MetricsExporter:
    """Producer-registry OTLP metrics egress (metricexport/exporter.go
    analog): producers are callables returning GaugePoint lists,
    collected every `interval`."""

    def __init__(self, channel, resource_attrs: Dict[str, str],
                 interval: float = 30.0) -> None:
        self._export = channel.unary_unary(
            f"/{METRICS_SERVICE}/Export",
            request_serializer=lambda b: b,
            response_deserializer=lambda b: b)
        self.resource_attrs = resource_attrs
        self.interval = interval
        self._producers: List[Callable[[], List[GaugePoint]]] = []
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.exported = 0
        self.errors = 0

    def register(self, producer: Callable[[], List[GaugePoint]]) -> None:
        self._producers.append(producer)

    def start(self) -> None:
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name="otlp-metrics")
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=2)
        self.export_once()

    def _run(self) -> None:
        while not self._stop.wait(self.interval):
            self.export_once()

    def export_once(self) -> None:
        points: List[GaugePoint] = []
        for producer in self._producers:
            try:
                points.extend(producer())
            except Exception:
                continue
        if not points:
            return
        try:
            payload = encode_metrics_request(points, self.resource_attrs)
            self._export(payload, timeout=10)
            self.exported += len(points)
        except Exception:
            self.errors += 1
