"""Prometheus metrics bridge.

The reference bridges fork-side metric IDs into Prometheus counters via a
generated table and serves /metrics with build/process collectors
(reference: reporter/parca_reporter.go:985-1024, metrics/all.go,
main.go:164-171). Here a custom collector snapshots the live subsystem
metric structs on scrape — no double bookkeeping on the hot path.
"""

from __future__ import annotations

from prometheus_client import CollectorRegistry
from prometheus_client.core import CounterMetricFamily, GaugeMetricFamily
from prometheus_client import ProcessCollector, PlatformCollector

from .version import __version__


class AgentCollector:
    def __init__(self, agent) -> None:
        self.agent = agent

    def collect(self):
        a = self.agent
        info = GaugeMetricFamily(
            "parca_agent_info", "Agent build info", labels=["version"])
        info.add_metric([__version__], 1)
        yield info

        rm = a.reporter.metrics
        c = CounterMetricFamily(
            "parca_agent_samples_ingested_total",
            "Samples ingested by the reporter", labels=["origin"])
        for origin, n in rm.samples_by_origin.items():
            c.add_metric([origin], n)
        yield c
        yield CounterMetricFamily(
            "parca_agent_samples_dropped_relabel_total",
            "Samples dropped by relabeling",
            value=rm.samples_dropped_relabel)
        yield CounterMetricFamily(
            "parca_agent_batches_written_total",
            "Flush batches written", value=rm.batches_written)
        yield CounterMetricFamily(
            "parca_agent_batch_errors_total",
            "Flush batch destination errors", value=rm.batch_errors)

        if a.cpu_service is not None:
            yield CounterMetricFamily(
                "parca_agent_cpu_samples_total",
                "perf samples decoded", value=a.cpu_service.total_samples)
            # bpf.errors.ringbuf_output analog for the perf rings.
            yield CounterMetricFamily(
                "parca_agent_cpu_samples_lost_total",
                "perf ring records lost", value=a.cpu_service.lost_samples)
            yield CounterMetricFamily(
                "parca_agent_cpu_samples_no_mapping_total",
                "sampled IPs without a mapping",
                value=a.cpu_service.samples_no_mapping)
            yield CounterMetricFamily(
                "parca_agent_dwarf_unwound_total",
                "stacks recovered by the .eh_frame unwinder",
                value=a.cpu_service.stacks_unwound_dwarf)
            uw = getattr(a.cpu_service, "unwinder", None)
            if uw is not None:
                # Dropped-frame accounting (reference discipline,
                # metrics/all.go:1328-1441): stacks whose CFA walk ran
                # off the end of the copied stack dump lost their roots.
                yield CounterMetricFamily(
                    "parca_agent_dwarf_stacks_truncated_total",
                    "stacks truncated at the copied-stack-dump boundary",
                    value=uw._native.stacks_truncated)
                yield GaugeMetricFamily(
                    "parca_agent_dwarf_table_bytes",
                    "packed stack-delta table memory",
                    value=uw._native.total_bytes)
            interp = CounterMetricFamily(
                "parca_agent_interpreter_stacks_total",
                "samples with interpreter frames", labels=["runtime"])
            interp.add_metric(["python"], a.cpu_service.python_stacks)
            interp.add_metric(["ruby"], a.cpu_service.ruby_stacks)
            interp.add_metric(["jvm"], a.cpu_service.jvm_stacks)
            interp.add_metric(["php"], a.cpu_service.php_stacks)
            interp.add_metric(["perl"], a.cpu_service.perl_stacks)
            interp.add_metric(
                ["native_labels"],
                a.cpu_service.native_labels.samples_labeled)
            interp.add_metric(
                ["native_labels_read_errors"],
                a.cpu_service.native_labels.read_errors)
            yield interp

        if a.gpu_service is not None:
            if getattr(a.gpu_service, "hw_metrics", None) is not None:
                util = GaugeMetricFamily(
                    "parca_agent_gpu_utilization_percent",
                    "GPU busy percent (rocm-smi)", labels=["gpu"])
                vram = GaugeMetricFamily(
                    "parca_agent_gpu_vram_used_bytes",
                    "GPU VRAM used (rocm-smi)", labels=["gpu"])
                for s in a.gpu_service.hw_metrics.read():
                    if s.utilization_pct >= 0:
                        util.add_metric([str(s.gpu_index)],
                                        s.utilization_pct)
                    if s.vram_used_bytes >= 0:
                        vram.add_metric([str(s.gpu_index)],
                                        s.vram_used_bytes)
                yield util
                yield vram
            gm = a.gpu_service.metrics
            yield GaugeMetricFamily(
                "parca_agent_gpu_rings_open", "attached GPU event rings",
                value=gm.rings_open)
            # rocm.errors.ringbuf_full (cuda.errors.ringbuf_full analog,
            # metrics/all.go:1334-1339).
            yield CounterMetricFamily(
                "parca_agent_gpu_ring_dropped_total",
                "GPU ring records dropped producer-side",
                value=gm.ring_dropped)
            yield CounterMetricFamily(
                "parca_agent_gpu_pc_samples_total",
                "GPU PC samples bucketed", value=gm.pc_samples)
            yield CounterMetricFamily(
                "parca_agent_gpu_kernels_reported_total",
                "GPU kernel timings reported", value=gm.kernels_reported)
            yield CounterMetricFamily(
                "parca_agent_gpu_tool_errors_total",
                "tool-side error events", value=gm.tool_errors)
            # rocm.traces_awaiting_times analog gauges.
            pending = GaugeMetricFamily(
                "parca_agent_gpu_fixer_pending",
                "fixer correlation entries pending", labels=["side"])
            times_n = stacks_n = 0
            for state in list(a.gpu_service._rings.values()):
                t, s = state.fixer.pending_counts()
                times_n += t
                stacks_n += s
            pending.add_metric(["times_awaiting_traces"], times_n)
            pending.add_metric(["traces_awaiting_times"], stacks_n)
            yield pending
            # rocm.traces_cleared analog (cuda.traces_cleared,
            # metrics/all.go:1400-1411).
            cleared = sum(
                state.fixer.metrics.traces_cleared
                for state in list(a.gpu_service._rings.values()))
            yield CounterMetricFamily(
                "parca_agent_gpu_fixer_traces_cleared_total",
                "launch stacks evicted unmatched (age/cap)",
                value=cleared)

        probes = getattr(a, "probes_service", None)
        if probes is not None and getattr(probes, "_group", None) is not None:
            g = probes._group
            yield CounterMetricFamily(
                "parca_agent_probe_fires_lost_total",
                "probe fires lost (perf ring overrun / drain backlog)",
                value=g.lost)
            yield CounterMetricFamily(
                "parca_agent_probe_throttles_total",
                "probe pairs disabled for flood cooldown",
                value=g.throttles)
            yield CounterMetricFamily(
                "parca_agent_probe_fires_throttled_total",
                "probe fires suppressed by the flood limiter",
                value=g.throttled_fires)


def build_registry(agent) -> CollectorRegistry:
    registry = CollectorRegistry()
    ProcessCollector(registry=registry)
    PlatformCollector(registry=registry)
    registry.register(AgentCollector(agent))
    return registry
