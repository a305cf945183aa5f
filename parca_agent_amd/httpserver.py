"""Agent HTTP endpoint: /metrics, /healthz, /debug/stats, /debug/pprof/*.

Reference: main.go:326-340 serves /metrics + /debug/pprof + fgprof. The
/debug/pprof/profile endpoint here self-profiles the agent with its own
perf sampler (target_pid = self) and returns pprof bytes — the agent
profiles itself with the same machinery it uses for the fleet.
"""

from __future__ import annotations

import json
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Optional
from urllib.parse import parse_qs, urlparse

from prometheus_client import generate_latest
from prometheus_client.exposition import CONTENT_TYPE_LATEST


class AgentHTTPServer:
    def __init__(self, address: str, agent, registry) -> None:
        host, _, port = address.rpartition(":")
        self.agent = agent
        self.registry = registry
        outer = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, fmt, *args):  # quiet
                pass

            def _send(self, code: int, body: bytes,
                      content_type: str = "text/plain") -> None:
                self.send_response(code)
                self.send_header("Content-Type", content_type)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def do_GET(self):
                parsed = urlparse(self.path)
                if parsed.path == "/metrics":
                    self._send(200, generate_latest(outer.registry),
                               CONTENT_TYPE_LATEST)
                elif parsed.path == "/healthz":
                    self._send(200, b"ok")
                elif parsed.path == "/":
                    self._send(200, b"parca-agent-amd\n")
                elif parsed.path == "/debug/stats":
                    stats = outer.agent.stats()
                    body = json.dumps({
                        "cpu_samples": stats.cpu_samples,
                        "cpu_samples_lost": stats.cpu_samples_lost,
                        "gpu_events": stats.gpu_events,
                        "gpu_ring_dropped": stats.gpu_ring_dropped,
                        "pc_samples": stats.pc_samples,
                        "kernels_reported": stats.kernels_reported,
                        "dropped_sample_pct": stats.dropped_sample_pct,
                    }).encode()
                    self._send(200, body, "application/json")
                elif parsed.path == "/debug/pprof/profile":
                    qs = parse_qs(parsed.query)
                    seconds = min(int(qs.get("seconds", ["10"])[0]), 60)
                    try:
                        body = outer.self_profile(seconds)
                        self._send(200, body, "application/octet-stream")
                    except Exception as e:  # pragma: no cover
                        self._send(500, str(e).encode())
                elif parsed.path == "/debug/pprof/fgprof":
                    qs = parse_qs(parsed.query)
                    seconds = min(int(qs.get("seconds", ["5"])[0]), 60)
                    try:
                        body = outer.wallclock_profile(seconds)
                        self._send(200, body, "application/octet-stream")
                    except Exception as e:  # pragma: no cover
                        self._send(500, str(e).encode())
                else:
                    self._send(404, b"not found")

        self._httpd = ThreadingHTTPServer((host or "127.0.0.1", int(port)),
                                          Handler)
        self._thread: Optional[threading.Thread] = None

    @property
    def port(self) -> int:
        return self._httpd.server_address[1]

    def self_profile(self, seconds: int) -> bytes:
        """Profile the agent itself at 99 Hz for `seconds`."""
        import os

        from .cpu import CPUSamplerService
        from .reporter import Reporter, samples_to_pprof

        class Dest:
            def __init__(self):
                self.samples = []

            def write_batch(self, batch):
                self.samples.extend(batch)

            def close(self):
                pass

        dest = Dest()
        rep = Reporter([dest], cpu_sampling_frequency=99)
        svc = CPUSamplerService(rep, freq=99, poll_interval=0.05,
                                target_pid=os.getpid())
        svc.start()
        time.sleep(seconds)
        svc.stop()
        rep.flush()
        profiles = samples_to_pprof(dest.samples)
        return profiles.get("samples", b"")

    def wallclock_profile(self, seconds: int, hz: int = 97) -> bytes:
        """fgprof analog (reference: main.go:332): wallclock profile of
        every agent thread — on- AND off-CPU — by sampling
        sys._current_frames(); shows where threads wait, which on-CPU
        profiles cannot."""
        import sys

        from .pprof import FrameKey, ProfileBuilder, ValueType

        counts: dict = {}
        thread_names = {}
        interval = 1.0 / hz
        deadline = time.monotonic() + seconds
        while time.monotonic() < deadline:
            for t in threading.enumerate():
                if t.ident is not None:
                    thread_names[t.ident] = t.name
            for tid, frame in sys._current_frames().items():
                stack = []
                f = frame
                while f is not None and len(stack) < 64:
                    code = f.f_code
                    stack.append((code.co_name, code.co_filename,
                                  f.f_lineno))
                    f = f.f_back
                key = (thread_names.get(tid, str(tid)), tuple(stack))
                counts[key] = counts.get(key, 0) + 1
            time.sleep(interval)
        b = ProfileBuilder(
            sample_types=[ValueType("wallclock", "nanoseconds")],
            period_type=ValueType("wallclock", "nanoseconds"),
            period=int(1e9 / hz))
        for (tname, stack), n in counts.items():
            frames = [FrameKey(address=0, mapping=None, function_name=nm,
                               source_file=fn, line=ln)
                      for nm, fn, ln in stack]
            b.add_sample(frames, [n * int(1e9 / hz)],
                         labels=[("thread_name", tname)])
        return b.serialize_gzip()

    def start(self) -> None:
        self._thread = threading.Thread(
            target=self._httpd.serve_forever, name="http", daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._httpd.shutdown()
        if self._thread:
            self._thread.join(timeout=5)
