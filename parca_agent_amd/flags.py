"""CLI flag and config system.

Mirrors the reference's Kong-based flag surface (reference:
flags/flags.go:123-179) with the same flag names and defaults so operators
can switch agents without rewriting deployments. Every CLI flag is also
settable from the YAML config file; CLI values win over YAML values over
defaults (reference: flags/flags.go:88-112). ROCm-specific flags replace
the CUDA ones (--rocm-* instead of --instrument-cuda-launch /
CUPTIEventScaleFactor, flags.go:62-66).
"""

from __future__ import annotations

import argparse
import dataclasses
import os
import re
import socket
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

import yaml

# Sampling at a prime frequency avoids aliasing with periodic user-space
# activity (100 Hz timers etc.) — same rationale and default as the
# reference (flags/flags.go:44-51).
DEFAULT_SAMPLE_FREQ = 19

# GPU event shm ring: 1 MiB default, doubled per scale-factor unit, capped
# at 256 MiB (reference: flags/flags.go:62-66 for the CUPTI ringbuf).
DEFAULT_GPU_RING_BYTES = 32 << 20
MAX_GPU_RING_SCALE = 4  # 32 MiB << 4 = 512 MiB cap


@dataclass
class LogFlags:
    level: str = "info"
    format: str = "logfmt"


@dataclass
class ProfilingFlags:
    duration: float = 5.0  # seconds per profile batch (flags.go:316)
    cpu_sampling_frequency: int = DEFAULT_SAMPLE_FREQ
    perf_event_buffer_poll_interval: float = 0.1
    # Per-CPU perf mmap ring, in pages (the reference's
    # trace-buffer-size-multiplier analog, flags.go Hidden group).
    perf_ring_pages: int = 64
    probabilistic_interval: float = 60.0
    probabilistic_threshold: int = 100  # of 100: always profile
    label_ttl: float = 600.0  # per-pid label refresh (flags.go:317)
    # Accepted for parity (flags.go:327); this build keeps error frames
    # out by construction (no fork error-frame channel).
    enable_error_frames: bool = False


@dataclass
class MetadataFlags:
    external_labels: Dict[str, str] = field(default_factory=dict)
    enable_process_cmdline: bool = False
    disable_on_prem: bool = False
    container_runtime_socket_path: str = ""  # CRI socket (flags.go:333)
    disable_caching: bool = False
    disable_cpu_label: bool = False          # flags.go:338-340
    disable_thread_id_label: bool = False
    disable_thread_comm_label: bool = False


@dataclass
class LocalStoreFlags:
    directory: str = ""
    symbolize: bool = False  # agent-side symtab names in local pprof


@dataclass
class RemoteStoreFlags:
    address: str = ""
    bearer_token: str = ""
    bearer_token_file: str = ""
    insecure: bool = False
    insecure_skip_verify: bool = False
    batch_write_interval: float = 10.0  # reference 10s flush loop
    rpc_timeout: float = 300.0
    grpc_max_call_recv_msg_size: int = 32 * 1024 * 1024
    grpc_max_call_send_msg_size: int = 32 * 1024 * 1024
    write_format: str = "arrow_v2"  # arrow_v1 | arrow_v2 | pprof
    # mTLS client auth (flags.go remote-store group).
    client_cert: str = ""
    client_key: str = ""
    grpc_connection_timeout: float = 60.0
    grpc_max_connection_retries: int = 10
    grpc_startup_backoff_time: float = 2.0
    grpc_headers: List[str] = field(default_factory=list)
    rpc_logging_enable: bool = False
    rpc_unary_timeout: float = 300.0
    # [deprecated in reference, default true] v2 Arrow schema toggle:
    # --remote-store-use-v2-schema=false maps to write_format=arrow_v1.
    use_v2_schema: bool = True


@dataclass
class DebuginfoFlags:
    upload_disable: bool = False
    strip: bool = True
    # SHF_COMPRESSED/zlib DWARF sections in the stripped upload
    # (flags.go:378; elfwriter._maybe_compress).
    compress: bool = False
    directories: List[str] = field(
        default_factory=lambda: ["/usr/lib/debug"])  # flags.go:375
    upload_max_parallel: int = 25
    upload_queue_size: int = 4096
    upload_cache_duration: float = 300.0  # flags.go:382
    temp_dir: str = "/tmp"
    disable_caching: bool = False
    upload_timeout_duration: float = 120.0


@dataclass
class OTLPFlags:
    address: str = ""
    exporter: str = "grpc"  # grpc | http | stdout


@dataclass
class TelemetryFlags:
    disable_panic_reporting: bool = False
    stderr_buffer_size_kb: int = 4096


@dataclass
class OfflineModeFlags:
    storage_path: str = ""
    rotation_interval: float = 600.0
    upload: bool = False


@dataclass
class RocmFlags:
    """GPU profiling flags — the rocprofiler-native replacement for the
    reference's CUPTI knobs (flags.go:62-66, main.go:487-491)."""

    enable: bool = True
    ring_scale_factor: int = 0  # ring = 32 MiB << n, n <= 4
    pc_sampling: bool = True
    pc_sampling_interval: int = 1048576  # cycles between PC samples
    pc_sampling_method: str = "host_trap"  # host_trap | stochastic
    kernel_batch_size: int = 100  # parcagpu.go:96 batch of kernel timings
    # log2 of the PC-bucket size in bytes for the HIP histogram:
    # 6 => 64-byte buckets (≈ a few instructions).
    bucket_bits: int = 6
    merge_node_profiles: bool = False  # RCCL all-gather across local GPUs
    shm_dir: str = "/dev/shm"


@dataclass
class HiddenFlags:
    force_panic: bool = False
    ignore_unsafe_kernel_version: bool = False


@dataclass
class Flags:
    log: LogFlags = field(default_factory=LogFlags)
    http_address: str = "127.0.0.1:7071"
    version: bool = False
    node: str = field(default_factory=socket.gethostname)
    config_path: str = ""
    environment_type: str = ""
    machine_id: str = ""
    include_env_var: List[str] = field(default_factory=list)
    tracers: str = "all"

    profiling: ProfilingFlags = field(default_factory=ProfilingFlags)
    metadata: MetadataFlags = field(default_factory=MetadataFlags)
    local_store: LocalStoreFlags = field(default_factory=LocalStoreFlags)
    remote_store: RemoteStoreFlags = field(default_factory=RemoteStoreFlags)
    debuginfo: DebuginfoFlags = field(default_factory=DebuginfoFlags)
    otlp: OTLPFlags = field(default_factory=OTLPFlags)
    telemetry: TelemetryFlags = field(default_factory=TelemetryFlags)
    offline_mode: OfflineModeFlags = field(default_factory=OfflineModeFlags)
    rocm: RocmFlags = field(default_factory=RocmFlags)
    hidden: HiddenFlags = field(default_factory=HiddenFlags)

    clock_sync_interval: float = 180.0
    off_cpu_threshold: float = 0.0
    analytics_opt_out: bool = False
    merge_gpu_profiles: bool = False
    otlp_logging: bool = False
    otel_tags: str = ""  # flags.go:132
    probe_config_file: str = ""
    enable_oom_watch: bool = False
    # Reference names (flags.go:171-172): --enable-oom-prof is an alias
    # of --enable-oom-watch; allocs gates the alloc_* heap sample types.
    enable_oom_prof: bool = False
    enable_oom_prof_allocs: bool = False
    symbolizer_jit_disable: bool = False  # flags.go:389
    dwarf_unwinding_disable: bool = False
    dwarf_unwinding_mixed: bool = True
    python_unwinding_disable: bool = False
    ruby_unwinding_disable: bool = False
    jvm_unwinding_disable: bool = False
    java_unwinding_disable: bool = False  # reference name; alias of jvm
    php_unwinding_disable: bool = False
    perl_unwinding_disable: bool = False
    mutex_profile_fraction: int = 0
    block_profile_rate: int = 0

    # -- validation (reference: flags/flags.go:201-268) -------------------

    def validate(self) -> None:
        if self.profiling.cpu_sampling_frequency <= 0:
            raise ValueError("profiling-cpu-sampling-frequency must be > 0")
        if self.profiling.duration <= 0:
            raise ValueError("profiling-duration must be > 0")
        if not 0 <= self.rocm.ring_scale_factor <= MAX_GPU_RING_SCALE:
            raise ValueError(
                f"rocm-ring-scale-factor must be in [0, {MAX_GPU_RING_SCALE}]"
            )
        if not 0.0 <= self.off_cpu_threshold <= 1.0:
            raise ValueError("off-cpu-threshold must be a probability in [0, 1]")
        if not 1 <= self.profiling.probabilistic_threshold <= 100:
            raise ValueError("probabilistic-threshold must be in [1, 100]")
        if self.offline_mode.storage_path and self.remote_store.address \
                and not self.offline_mode.upload:
            raise ValueError(
                "offline-mode-storage-path and remote-store-address are "
                "exclusive (except in offline-mode-upload replay)"
            )
        if self.offline_mode.upload and not self.offline_mode.storage_path:
            raise ValueError("offline-mode-upload requires offline-mode-storage-path")
        if self.rocm.pc_sampling_method not in ("host_trap", "stochastic"):
            raise ValueError("rocm-pc-sampling-method must be host_trap|stochastic")
        if self.remote_store.write_format not in ("arrow_v1", "arrow_v2", "pprof"):
            raise ValueError("remote-store-write-format must be arrow_v1|arrow_v2|pprof")

    @property
    def gpu_ring_bytes(self) -> int:
        return DEFAULT_GPU_RING_BYTES << self.rocm.ring_scale_factor

    def trace_cache_size(self, cores: Optional[int] = None) -> int:
        """Sample-dedup cache sizing: freq x interval x cores x 6 intervals,
        min 64 Ki, rounded up to a power of two (reference: main.go:682-703).
        """
        if cores is None:
            cores = os.cpu_count() or 1
        size = int(
            self.profiling.cpu_sampling_frequency * self.profiling.duration * cores * 6
        )
        size = max(size, 65536)
        return 1 << (size - 1).bit_length()


# -- flag registry: maps dotted field path -> CLI flag name ----------------


def _flag_name(path: str) -> str:
    # profiling.cpu_sampling_frequency -> --profiling-cpu-sampling-frequency
    # hidden fields drop their prefix (reference: Hidden prefix:"")
    parts = path.split(".")
    if parts[0] == "hidden":
        parts = parts[1:]
    if parts[0] == "log":
        parts = ["log"] + parts[1:]
    return "-".join(p.replace("_", "-") for p in parts)


def _iter_fields(obj: Any, prefix: str = ""):
    for f in dataclasses.fields(obj):
        value = getattr(obj, f.name)
        path = f"{prefix}{f.name}"
        if dataclasses.is_dataclass(value):
            yield from _iter_fields(value, path + ".")
        else:
            yield path, f, value


def _set_path(flags: Flags, path: str, value: Any) -> None:
    obj: Any = flags
    parts = path.split(".")
    for p in parts[:-1]:
        obj = getattr(obj, p)
    setattr(obj, parts[-1], value)


def _get_path(flags: Flags, path: str) -> Any:
    obj: Any = flags
    for p in path.split("."):
        obj = getattr(obj, p)
    return obj


# Help text for the most consequential flags (mirroring the reference's
# README.md flag table); everything else is self-describing by name.
_HELP = {
    "remote-store-address": "gRPC address of the Parca server "
                            "(profilestore + debuginfo + telemetry).",
    "remote-store-insecure": "Plaintext gRPC (no TLS).",
    "remote-store-bearer-token": "Bearer token attached to every RPC.",
    "remote-store-grpc-headers": "Extra metadata, key=value (repeatable).",
    "remote-store-use-v2-schema": "Arrow v2 schema (default); false "
                                  "falls back to the v1 writer.",
    "local-store-directory": "Write pprof batches to this directory "
                             "instead of (or besides) a server.",
    "offline-mode-storage-path": "Framed .padata log directory for "
                                 "air-gapped capture; replay with "
                                 "--offline-mode-upload.",
    "profiling-cpu-sampling-frequency": "Per-CPU sampling Hz "
                                        "(prime default 19).",
    "profiling-duration": "Profile batch duration in seconds.",
    "dwarf-unwinding-disable": "Frame-pointer unwinding only.",
    "rocm-enable": "GPU subsystem (rocprofiler tool rings, kernel "
                   "timings, PC sampling).",
    "rocm-pc-sampling": "gfx950 PC sampling where the driver exposes it.",
    "rocm-ring-scale-factor": "Per-process GPU ring = 32 MiB << n.",
    "rocm-merge-node-profiles": "Daemon-side merged node-level GPU "
                                "pprof across all local GPUs.",
    "merge-gpu-profiles": "Fold kernel time + PC samples into one "
                          "gpu_time profile with a gpu_view label.",
    "probe-config-file": "YAML uprobe specs (paired entry/exit spans).",
    "enable-oom-prof": "Ship allocation profiles of OOM-killed "
                       "processes (LD_PRELOAD libparca_heap.so).",
    "enable-oom-prof-allocs": "Also ship alloc_space/alloc_objects "
                              "(inuse_* always).",
    "debuginfo-compress": "SHF_COMPRESSED/zlib DWARF in uploads.",
    "debuginfo-directories": "Split-DWARF search roots "
                             "(.build-id convention).",
    "config-path": "YAML config file (flags + relabel_configs).",
    "metadata-external-labels": "key=value labels on all profiles "
                                "(repeatable).",
    "off-cpu-threshold": "Off-CPU (wallclock) sampling probability "
                         "(0..1; 0 disables).",
}


def build_parser() -> argparse.ArgumentParser:
    parser = argparse.ArgumentParser(
        prog="parca-agent-amd",
        description="MI355X-native always-on sampling profiler",
        allow_abbrev=False,
    )
    defaults = Flags()
    for path, f, value in _iter_fields(defaults):
        name = _flag_name(path)
        flag = "--" + name
        help_text = _HELP.get(name)
        if f.type in ("bool", bool) or isinstance(value, bool):
            parser.add_argument(
                flag, dest=path, default=None,
                type=_parse_bool, nargs="?", const=True, metavar="BOOL",
                help=help_text,
            )
        elif isinstance(value, int):
            parser.add_argument(flag, dest=path, default=None, type=int,
                                help=help_text)
        elif isinstance(value, float):
            parser.add_argument(flag, dest=path, default=None,
                                type=_parse_duration, help=help_text)
        elif isinstance(value, list):
            parser.add_argument(flag, dest=path, default=None,
                                action="append", help=help_text)
        elif isinstance(value, dict):
            parser.add_argument(flag, dest=path, default=None,
                                action="append", metavar="KEY=VALUE",
                                help=help_text)
        else:
            parser.add_argument(flag, dest=path, default=None, type=str,
                                help=help_text)
    return parser


def _parse_bool(v: str) -> bool:
    if isinstance(v, bool):
        return v
    return str(v).lower() in ("1", "true", "yes", "on")


_DURATION_RE = re.compile(r"^(\d+(?:\.\d+)?)(ns|us|ms|s|m|h)?$")
_DURATION_UNITS = {"ns": 1e-9, "us": 1e-6, "ms": 1e-3, "s": 1.0, "m": 60.0,
                   "h": 3600.0, None: 1.0}


def _parse_duration(v: str) -> float:
    """Accept Go-style durations ('10s', '3m') and bare numbers (seconds)."""
    m = _DURATION_RE.match(str(v).strip())
    if not m:
        raise argparse.ArgumentTypeError(f"invalid duration {v!r}")
    return float(m.group(1)) * _DURATION_UNITS[m.group(2)]


def _parse_kv_list(items: List[str]) -> Dict[str, str]:
    out = {}
    for item in items:
        if "=" not in item:
            raise ValueError(f"expected KEY=VALUE, got {item!r}")
        k, _, v = item.partition("=")
        out[k] = v
    return out


def _apply_yaml(flags: Flags, config: Dict[str, Any]) -> None:
    """YAML keys use the CLI flag names (dashes or underscores)."""
    by_flag = {_flag_name(path): path for path, _, _ in _iter_fields(Flags())}
    for key, value in config.items():
        if key == "relabel_configs":
            continue  # handled by config.load_relabel_configs
        norm = key.replace("_", "-")
        path = by_flag.get(norm)
        if path is None:
            raise ValueError(f"unknown config key {key!r}")
        current = _get_path(flags, path)
        if isinstance(current, bool):
            value = _parse_bool(value)
        elif isinstance(current, float) and isinstance(value, str):
            value = _parse_duration(value)
        elif isinstance(current, dict) and isinstance(value, list):
            value = _parse_kv_list(value)
        _set_path(flags, path, value)


def parse(argv: Optional[List[str]] = None) -> Flags:
    """Two-pass parse: defaults <- YAML config file <- CLI (CLI wins),
    mirroring the reference's kong.Configuration overlay
    (flags/flags.go:88-112)."""
    parser = build_parser()
    ns = parser.parse_args(argv)
    flags = Flags()

    config_path = getattr(ns, "config_path", None) or ""
    if config_path:
        with open(config_path) as fh:
            doc = yaml.safe_load(fh) or {}
        _apply_yaml(flags, doc)
        flags.config_path = config_path

    for path, _, default in _iter_fields(Flags()):
        cli_value = getattr(ns, path, None)
        if cli_value is None:
            continue
        if isinstance(default, dict):
            cli_value = _parse_kv_list(cli_value)
        _set_path(flags, path, cli_value)

    if flags.remote_store.bearer_token_file:
        with open(flags.remote_store.bearer_token_file) as fh:
            flags.remote_store.bearer_token = fh.read().strip()

    # Reference-name aliases.
    if flags.java_unwinding_disable:
        flags.jvm_unwinding_disable = True
    if flags.enable_oom_prof:
        flags.enable_oom_watch = True
    if not flags.remote_store.use_v2_schema and \
            flags.remote_store.write_format == "arrow_v2":
        flags.remote_store.write_format = "arrow_v1"

    flags.validate()
    return flags
