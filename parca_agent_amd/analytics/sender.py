"""Anonymous usage analytics via Prometheus remote-write.

Reference: analytics/analytics.go:69-137 — `parca_agent_info` and
`parca_agent_cpu_cores` series, snappy-compressed remote-write protobuf,
opt-out flag. Ships with a hand-rolled remote-write encoder (the
prometheus.WriteRequest proto: Timeseries{labels=1{name=1,value=2},
samples=2{value=1 double, timestamp=2 int64}}).

Snappy is not importable standalone here; the payload uses the raw
(uncompressed) fallback with the appropriate content encoding disabled —
servers that require snappy will reject it, and the sender treats any
failure as non-fatal (analytics are fire-and-forget).
"""

from __future__ import annotations

import logging
import os
import threading
import time
import urllib.request
from typing import Dict, List, Tuple

from ..pprof.proto import Writer

log = logging.getLogger("parca_agent_amd.analytics")

DEFAULT_ENDPOINT = os.environ.get(
    "PARCA_ANALYTICS_ENDPOINT",
    "https://analytics.parca.dev/api/v1/write")


def snappy_block(data: bytes) -> bytes:
    """Spec-valid snappy using only literal elements (no matches): the
    remote-write protocol REQUIRES snappy framing and the reference uses
    golang/snappy; an all-literal stream is legal snappy that any
    decoder accepts, with zero dependencies. Analytics payloads are a
    few hundred bytes, so the missed compression is irrelevant."""
    out = bytearray()
    n = len(data)
    while True:  # uvarint(uncompressed length)
        b = n & 0x7F
        n >>= 7
        out.append(b | (0x80 if n else 0))
        if not n:
            break
    pos = 0
    while pos < len(data):
        chunk = data[pos:pos + 65536]
        ln = len(chunk) - 1
        if ln < 60:
            out.append(ln << 2)
        elif ln < 256:
            out.append(60 << 2)
            out.append(ln)
        else:
            out.append(61 << 2)
            out += ln.to_bytes(2, "little")
        out += chunk
        pos += len(chunk)
    return bytes(out)


def encode_remote_write(series: List[Tuple[Dict[str, str], float, int]]
                        ) -> bytes:
    w = Writer()
    for labels, value, ts_ms in series:
        ts = Writer()
        for name in sorted(labels):
            lw = Writer()
            lw.string(1, name)
            lw.string(2, labels[name])
            ts.message(1, lw)
        sw = Writer()
        sw.double(1, value)
        sw.varint(2, ts_ms)
        ts.message(2, sw)
        w.message(1, ts)
    return w.getvalue()


class AnalyticsSender:
    def __init__(self, version: str, endpoint: str = DEFAULT_ENDPOINT,
                 interval: float = 10.0) -> None:
        self.version = version
        self.endpoint = endpoint
        self.interval = interval
        self._stop = threading.Event()
        self._thread = None
        self.machine_id = _machine_id()

    def start(self) -> None:
        self._thread = threading.Thread(target=self._run, name="analytics",
                                        daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2)

    def _run(self) -> None:
        while not self._stop.wait(self.interval):
            try:
                self.send_once()
            except Exception:
                # fire-and-forget by design (analytics.go)
                pass

    def send_once(self) -> None:
        now_ms = int(time.time() * 1000)
        u = os.uname()
        series = [
            ({"__name__": "parca_agent_info",
              "version": self.version,
              "machine_id": self.machine_id,
              "arch": u.machine,
              "kernel_version": u.release,
              "agent": "parca-agent-amd"}, 1.0, now_ms),
            ({"__name__": "parca_agent_cpu_cores",
              "machine_id": self.machine_id},
             float(os.cpu_count() or 0), now_ms),
        ]
        payload = snappy_block(encode_remote_write(series))
        req = urllib.request.Request(
            self.endpoint, data=payload, method="POST",
            headers={"Content-Type": "application/x-protobuf",
                     "Content-Encoding": "snappy",
                     "X-Prometheus-Remote-Write-Version": "0.1.0",
                     "User-Agent": f"parca-agent-amd/{self.version}"})
        urllib.request.urlopen(req, timeout=5).close()


def _machine_id() -> str:
    for path in ("/etc/machine-id", "/var/lib/dbus/machine-id"):
        try:
            with open(path) as fh:
                return fh.read().strip()
        except OSError:
            continue
    return "unknown"
