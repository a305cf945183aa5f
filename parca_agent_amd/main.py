"""Agent entry point (mainWithExitCode analog, reference main.go:118).

Startup order mirrors the reference: flags -> validate -> offline-upload
early exit -> telemetry supervisor fork -> HTTP server -> metadata/
reporter construction -> uploader -> GPU + CPU services -> analytics ->
run until signal (reference call stack: SURVEY.md §3.1).
"""

from __future__ import annotations

import json
import logging
import os
import signal
import sys
import threading
import time
from typing import List, Optional

from . import flags as flagsmod
from .agent import Agent
from .config import load_relabel_configs
from .version import __version__

log = logging.getLogger("parca_agent_amd")

EXIT_SUCCESS = 0
EXIT_FAILURE = 1
EXIT_PARSE_ERROR = 2


def setup_logging(level: str, fmt: str) -> None:
    lvl = getattr(logging, level.upper(), logging.INFO)
    if fmt == "json":
        formatter = logging.Formatter(
            '{"ts":"%(asctime)s","level":"%(levelname)s",'
            '"logger":"%(name)s","msg":"%(message)s"}')
    else:
        formatter = logging.Formatter(
            "ts=%(asctime)s level=%(levelname)s logger=%(name)s "
            "msg=%(message)s")
    handler = logging.StreamHandler(sys.stderr)
    handler.setFormatter(formatter)
    root = logging.getLogger()
    root.handlers[:] = [handler]
    root.setLevel(lvl)


def run_telemetry_supervisor(f) -> int:
    """Parent process: re-exec the agent as a child, capture its stderr
    tail in a ring buffer, report abnormal exits via ReportPanic, and
    propagate the exit code (reference: main.go:230-315)."""
    import subprocess
    from collections import deque

    args = [sys.executable, "-m", "parca_agent_amd",
            "--telemetry-disable-panic-reporting", "true"] + sys.argv[1:]
    buf_limit = f.telemetry.stderr_buffer_size_kb * 1024
    ring: deque = deque()
    ring_size = 0

    # Survive the child's OOM so the panic report goes out
    # (main.go:242-249).
    try:
        with open("/proc/self/oom_score_adj", "w") as fh:
            fh.write("-100")
    except OSError:
        pass

    proc = subprocess.Popen(args, stderr=subprocess.PIPE)
    stop = {"sig": None}

    def forward(signum, _frame):
        stop["sig"] = signum
        proc.send_signal(signum)

    signal.signal(signal.SIGTERM, forward)
    signal.signal(signal.SIGINT, forward)

    assert proc.stderr is not None
    for line in proc.stderr:
        sys.stderr.buffer.write(line)
        sys.stderr.buffer.flush()
        ring.append(line)
        ring_size += len(line)
        while ring_size > buf_limit and len(ring) > 1:
            ring_size -= len(ring.popleft())
    rc = proc.wait()

    if rc != 0 and stop["sig"] is None and f.remote_store.address:
        stderr_tail = b"".join(ring).decode("utf-8", "replace")
        try:
            from .reporter.grpc_client import ParcaClient, build_channel

            client = ParcaClient(build_channel(f))
            client.report_panic(stderr_tail, {
                "agent_version": __version__,
                "kernel_release": os.uname().release,
                "exit_code": str(rc),
                "node": f.node,
            })
            log.info("reported abnormal child exit rc=%d", rc)
        except Exception:
            log.warning("panic report failed", exc_info=True)
    return rc


def main(argv: Optional[List[str]] = None) -> int:
    try:
        f = flagsmod.parse(argv)
    except (ValueError, OSError) as e:
        print(f"flag error: {e}", file=sys.stderr)
        return EXIT_PARSE_ERROR

    if f.version:
        print(f"parca-agent-amd {__version__}")
        return EXIT_SUCCESS

    setup_logging(f.log.level, f.log.format)

    # Cap glibc malloc arenas: the default scales with cores (8x), and
    # on 256-CPU nodes the churny native allocations (perf stack dumps,
    # encode buffers) spread across dozens of 64 MB arena heaps —
    # hundreds of MB of retained RSS for no throughput benefit at our
    # allocation rates. M_ARENA_MAX = -8.
    try:
        import ctypes

        libc = ctypes.CDLL("libc.so.6")
        libc.mallopt(-8, 2)  # M_ARENA_MAX
        # Pin M_MMAP_THRESHOLD at 1 MB: glibc's dynamic threshold grows
        # to 32 MB after large frees, after which libtorch-scale table
        # builds (hundreds of MB of transient vectors) land on the sbrk
        # heap and are RETAINED; mmap'd blocks return to the kernel on
        # free.
        libc.mallopt(-3, 1 << 20)
    except Exception:
        pass

    if os.environ.get("PARCA_TRACEMALLOC"):
        import tracemalloc

        tracemalloc.start(1)

    # Offline replay mode: upload recorded logs and exit
    # (reference: main.go:156-162).
    if f.offline_mode.upload:
        from .uploader import offline_upload

        return offline_upload(f)

    # Telemetry supervisor: the PARENT branch re-execs us
    # (disabled inside the child and when reporting is off).
    if not f.telemetry.disable_panic_reporting and \
            os.environ.get("PARCA_SUPERVISED") != "1":
        os.environ["PARCA_SUPERVISED"] = "1"
        return run_telemetry_supervisor(f)

    # Fault injector for the supervisor path — raised in the CHILD so the
    # parent observes the crash (reference: flags.go:413).
    if f.hidden.force_panic:
        raise RuntimeError("forced panic (--force-panic)")

    relabel_configs = ()
    if f.config_path:
        try:
            relabel_configs = load_relabel_configs(f.config_path)
        except (OSError, ValueError) as e:
            print(f"config error: {e}", file=sys.stderr)
            return EXIT_PARSE_ERROR

    agent = Agent(f, relabel_configs=relabel_configs)

    # Debuginfo uploader over the shared connection.
    uploader = None
    if f.remote_store.address and not f.debuginfo.upload_disable:
        from .reporter.grpc_client import ParcaClient, build_channel
        from .reporter.uploader import DebuginfoUploader, UploadItem

        client = ParcaClient(build_channel(f))
        uploader = DebuginfoUploader(
            client,
            max_parallel=f.debuginfo.upload_max_parallel,
            queue_size=f.debuginfo.upload_queue_size,
            strip=f.debuginfo.strip,
            compress=f.debuginfo.compress,
            temp_dir=f.debuginfo.temp_dir,
            debug_directories=f.debuginfo.directories,
            retry_cache_ttl=f.debuginfo.upload_cache_duration)
        uploader.start()

        def on_code_object(info):
            uploader.enqueue(UploadItem(
                build_id=f"codeobj-{info.file_id[:16]}",
                hash=info.file_id, data=info.data))

        if agent.gpu_service is not None:
            agent.gpu_service.code_objects.on_executable = on_code_object

    # Executable-discovery fan-out: debuginfo upload + probes late-attach
    # (reference: ReportExecutable -> uploader.Upload + probes.
    # OnExecutable, parca_reporter.go:856-917).
    executable_callbacks = []
    if uploader is not None:
        executable_callbacks.append(lambda info: uploader.enqueue(
            UploadItem(build_id=info.build_id or info.file_id,
                       hash=info.file_id, path=info.path)))

    def on_executable(info):
        for cb in executable_callbacks:
            try:
                cb(info)
            except Exception:
                log.debug("executable callback failed", exc_info=True)

    if agent.cpu_service is not None:
        agent.cpu_service.on_executable = on_executable
    if agent.gpu_service is not None:
        agent.gpu_service.resolver.on_executable = on_executable

    # OTLP log forwarding over the shared connection (--otlp-logging,
    # reference: logrus hook main.go:448-455).
    otlp_handler = None
    metrics_exporter = None
    if f.otlp_logging and f.remote_store.address:
        try:
            from .otlp import MetricsExporter, OTLPLogHandler
            from .reporter.grpc_client import build_channel

            otlp_channel = build_channel(f)
            resource = {"service.name": "parca-agent-amd", "host.name": f.node}
            # --otel-tags k=v,k2=v2 lands on every exported resource
            # (reference flags.go:132).
            for pair in (f.otel_tags or "").split(","):
                if "=" in pair:
                    k, _, v = pair.partition("=")
                    if k.strip():
                        resource[k.strip()] = v.strip()
            otlp_handler = OTLPLogHandler(otlp_channel, resource)
            logging.getLogger().addHandler(otlp_handler)
            metrics_exporter = MetricsExporter(otlp_channel, resource)
            from .otlp import GaugePoint

            def agent_metrics():
                import time as _t

                s = agent.stats()
                now = _t.time_ns()
                return [
                    GaugePoint("parca_agent.cpu_samples", s.cpu_samples,
                               now, is_sum=True),
                    GaugePoint("parca_agent.gpu.ring_dropped",
                               s.gpu_ring_dropped, now, is_sum=True),
                    GaugePoint("parca_agent.gpu.kernels_reported",
                               s.kernels_reported, now, is_sum=True),
                ]

            metrics_exporter.register(agent_metrics)
            if agent.gpu_service is not None:
                from .gpu.hwmetrics import GpuHwMetrics

                metrics_exporter.register(GpuHwMetrics().produce)
            metrics_exporter.start()
        except Exception:
            log.warning("otlp logging unavailable", exc_info=True)

    from .metrics import build_registry
    from .httpserver import AgentHTTPServer

    registry = build_registry(agent)
    http_server = AgentHTTPServer(f.http_address, agent, registry)
    http_server.start()
    log.info("http server on %s", f.http_address)

    # Probes service (uprobe pairs from YAML), with OTLP span export when
    # a remote store is configured (reference: probe spans over the
    # shared conn, probes/service.go:180-199).
    probes_service = None
    span_exporter = None
    if f.probe_config_file:
        try:
            from .probes.service import ProbesService

            if f.remote_store.address:
                try:
                    from .otlp import SpanExporter
                    from .reporter.grpc_client import build_channel

                    span_resource = {"service.name": "parca-agent-amd",
                                     "host.name": f.node}
                    for pair in (f.otel_tags or "").split(","):
                        if "=" in pair:
                            k, _, v = pair.partition("=")
                            if k.strip():
                                span_resource[k.strip()] = v.strip()
                    span_exporter = SpanExporter(
                        build_channel(f), span_resource)
                except Exception:
                    log.debug("span exporter unavailable", exc_info=True)
            probes_service = ProbesService.from_config_file(
                f.probe_config_file, agent.reporter)
            probes_service.span_exporter = span_exporter
            probes_service.start()
            agent.probes_service = probes_service  # /metrics visibility
            # Late-attach regex probes when new executables appear.
            executable_callbacks.append(
                lambda info: probes_service.on_executable(info.path))
        except Exception:
            log.error("probes service failed to start", exc_info=True)

    # OOM watcher.
    oom_watcher = None
    if f.enable_oom_watch:
        try:
            from .oom.watcher import OOMWatcher

            lookup = None
            if agent.cpu_service is not None:
                lookup = agent.cpu_service.last_trace_by_pid.get
            oom_watcher = OOMWatcher(agent.reporter,
                                     report_allocs=f.enable_oom_prof_allocs,
                                     last_stack_lookup=lookup)
            oom_watcher.start()
        except Exception:
            log.error("oom watcher failed to start", exc_info=True)

    # Anonymous analytics (opt-out).
    analytics_sender = None
    if not f.analytics_opt_out:
        try:
            from .analytics.sender import AnalyticsSender

            analytics_sender = AnalyticsSender(__version__)
            analytics_sender.start()
        except Exception:
            log.debug("analytics unavailable", exc_info=True)

    agent.start()
    log.info("parca-agent-amd %s started (freq=%d Hz, node=%s)",
             __version__, f.profiling.cpu_sampling_frequency, f.node)
    # System-test marker, mirroring the reference's documented hook
    # "Attached sched monitor" (main.go:554-556).
    log.info("Agent services attached")

    stop_event = threading.Event()

    def handle_signal(signum, _frame):
        log.info("signal %d: shutting down", signum)
        stop_event.set()

    signal.signal(signal.SIGTERM, handle_signal)
    signal.signal(signal.SIGINT, handle_signal)

    stop_event.wait()

    agent.stop()
    if probes_service is not None:
        probes_service.stop()
    if oom_watcher is not None:
        oom_watcher.stop()
    if analytics_sender is not None:
        analytics_sender.stop()
    if uploader is not None:
        uploader.stop()
    if otlp_handler is not None:
        logging.getLogger().removeHandler(otlp_handler)
        otlp_handler.close()
    if span_exporter is not None:
        span_exporter.close()
    if metrics_exporter is not None:
        metrics_exporter.stop()
    http_server.stop()

    stats_file = os.environ.get("PARCA_STATS_FILE")
    if stats_file:
        s = agent.stats()
        mem_diag = {}
        if os.environ.get("PARCA_TRACEMALLOC"):
            import tracemalloc

            if tracemalloc.is_tracing():
                top = tracemalloc.take_snapshot().statistics("lineno")[:25]
                mem_diag["tracemalloc_top"] = [
                    f"{st.traceback} size_mb={st.size/1e6:.1f} "
                    f"count={st.count}" for st in top]
        with open(stats_file, "w") as fh:
            json.dump({
                "mem_diag": mem_diag,
                "cpu_samples": s.cpu_samples,
                "cpu_samples_lost": s.cpu_samples_lost,
                "gpu_events": s.gpu_events,
                "gpu_ring_dropped": s.gpu_ring_dropped,
                "pc_samples": s.pc_samples,
                "kernels_reported": s.kernels_reported,
                "python_stacks": s.python_stacks,
                "perl_stacks": s.perl_stacks,
                "native_label_samples": s.native_label_samples,
                "ruby_stacks": s.ruby_stacks,
                "jvm_stacks": s.jvm_stacks,
                "php_stacks": s.php_stacks,
                "dwarf_stacks": s.dwarf_stacks,
                "dropped_sample_pct": s.dropped_sample_pct,
            }, fh)
    return EXIT_SUCCESS


if __name__ == "__main__":
    sys.exit(main())
