"""GPU hardware metrics producer tests (fake rocm-smi JSON)."""

from parca_agent_amd.gpu.hwmetrics import GpuHwMetrics, parse_rocm_smi

FAKE = {
    "card0": {
        "GPU use (%)": "87",
        "GPU Memory use (%)": "40",
        "VRAM Total Memory (B)": "309237645312",
        "VRAM Total Used Memory (B)": "123456789",
        "Average Graphics Package Power (W)": "612.0",
        "Temperature (Sensor junction) (C)": "78.0",
    },
    "card1": {
        "GPU use (%)": "3",
        "VRAM Total Memory (B)": "309237645312",
        "VRAM Total Used Memory (B)": "1024",
        "Current Socket Graphics Package Power (W)": "120.5",
        "Temperature (Sensor edge) (C)": "44.0",
    },
    "system": {"Driver version": "6.xx"},
}


def test_parse_rocm_smi():
    samples = parse_rocm_smi(FAKE)
    assert len(samples) == 2
    s0 = samples[0]
    assert s0.gpu_index == 0
    assert s0.utilization_pct == 87.0
    assert s0.vram_total_bytes == 309237645312.0
    assert s0.vram_used_bytes == 123456789.0
    assert s0.power_watts == 612.0
    assert s0.temperature_c == 78.0
    assert samples[1].power_watts == 120.5


def test_producer_points():
    m = GpuHwMetrics(reader=lambda: FAKE)
    points = m.produce()
    names = {(p.name, p.attributes["gpu"]) for p in points}
    assert ("gpu.utilization", "0") in names
    assert ("gpu.vram.used", "1") in names
    util0 = next(p for p in points
                 if p.name == "gpu.utilization" and
                 p.attributes["gpu"] == "0")
    assert util0.value == 87.0


def test_reader_error_counted():
    def boom():
        raise RuntimeError("no gpu")

    m = GpuHwMetrics(reader=boom, min_interval=0)
    assert m.read() == []
    assert m.errors == 1
