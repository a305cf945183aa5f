"""GPU hardware metrics producer: utilization / VRAM / power per device.

The reference's metricexport scaffold was written for a fork-side
`gpumetrics` producer (reference: metricexport/exporter.go:14-24). This
is that producer, MI355X-native: `rocm-smi --json` polled on demand,
surfaced both as OTLP gauge points (otlp.MetricsExporter) and through
the Prometheus collector. Injectable reader for CPU tests.
"""

from __future__ import annotations

import json
import logging
import subprocess
import time
from dataclasses import dataclass
from typing import Callable, Dict, List, Optional

log = logging.getLogger("parca_agent_amd.gpu.hwmetrics")

ROCM_SMI = "/opt/rocm/bin/rocm-smi"


@dataclass
class GpuHwSample:
    gpu_index: int
    utilization_pct: float = -1.0
    vram_used_bytes: float = -1.0
    vram_total_bytes: float = -1.0
    power_watts: float = -1.0
    temperature_c: float = -1.0


def _default_reader() -> Dict:
    out = subprocess.run(
        [ROCM_SMI, "--showuse", "--showmemuse", "--showpower",
         "--showtemp", "--showmeminfo", "vram", "--json"],
        capture_output=True, text=True, timeout=10)
    return json.loads(out.stdout or "{}")


def _to_float(v, scale: float = 1.0) -> float:
    try:
        return float(str(v).rstrip("%cW ")) * scale
    except (ValueError, TypeError):
        return -1.0


def parse_rocm_smi(doc: Dict) -> List[GpuHwSample]:
    samples = []
    for key, fields in sorted(doc.items()):
        if not key.startswith("card"):
            continue
        try:
            idx = int(key[4:])
        except ValueError:
            continue
        s = GpuHwSample(gpu_index=idx)
        for name, value in fields.items():
            low = name.lower()
            if "gpu use" in low:
                s.utilization_pct = _to_float(value)
            elif "vram total memory" in low:
                s.vram_total_bytes = _to_float(value)
            elif "vram total used" in low:
                s.vram_used_bytes = _to_float(value)
            elif "power" in low and "cap" not in low:
                s.power_watts = _to_float(value)
            elif "temperature" in low and "junction" in low:
                s.temperature_c = _to_float(value)
            elif "temperature" in low and s.temperature_c < 0:
                s.temperature_c = _to_float(value)
        samples.append(s)
    return samples


class GpuHwMetrics:
    def __init__(self, reader: Optional[Callable[[], Dict]] = None,
                 min_interval: float = 5.0) -> None:
        self._reader = reader or _default_reader
        self._min_interval = min_interval
        self._last_read = 0.0
        self._cached: List[GpuHwSample] = []
        self.errors = 0

    def read(self) -> List[GpuHwSample]:
        now = time.monotonic()
        if now - self._last_read < self._min_interval:
            return self._cached
        self._last_read = now
        try:
            self._cached = parse_rocm_smi(self._reader())
        except Exception:
            self.errors += 1
            self._cached = []
        return self._cached

    # -- OTLP producer (otlp.MetricsExporter.register) ---------------------

    def produce(self):
        from ..otlp import GaugePoint

        now_ns = time.time_ns()
        points = []
        for s in self.read():
            attrs = {"gpu": str(s.gpu_index)}
            if s.utilization_pct >= 0:
                points.append(GaugePoint(
                    "gpu.utilization", s.utilization_pct, now_ns,
                    attributes=attrs, unit="percent"))
            if s.vram_used_bytes >= 0:
                points.append(GaugePoint(
                    "gpu.vram.used", s.vram_used_bytes, now_ns,
                    attributes=attrs, unit="bytes"))
            if s.power_watts >= 0:
                points.append(GaugePoint(
                    "gpu.power", s.power_watts, now_ns,
                    attributes=attrs, unit="watts"))
            if s.temperature_c >= 0:
                points.append(GaugePoint(
                    "gpu.temperature", s.temperature_c, now_ns,
                    attributes=attrs, unit="celsius"))
        return points
