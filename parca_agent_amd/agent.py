"""Embeddable agent: wires sampler + GPU service + reporter together.

Used three ways:
- `parca-agent-amd` CLI (main.py) runs it as the host daemon.
- bench.py embeds it in the workload process to measure overhead.
- smoke()/GPU tests drive one profile cycle end-to-end.
"""

from __future__ import annotations

import ctypes
import logging
import os
from dataclasses import dataclass
from typing import List, Optional

from .flags import Flags
from .metadata import (
    AgentMetadataProvider,
    ContainerMetadataProvider,
    ProcessMetadataProvider,
    SystemMetadataProvider,
)
from .procmaps import ExecutableCache, ProcessTable
from .reporter import (
    LocalStoreDestination,
    OfflineLogDestination,
    Reporter,
)

log = logging.getLogger("parca_agent_amd.agent")


@dataclass
class AgentStats:
    cpu_samples: int = 0
    cpu_samples_lost: int = 0
    gpu_events: int = 0
    gpu_ring_dropped: int = 0
    pc_samples: int = 0
    kernels_reported: int = 0
    python_stacks: int = 0
    perl_stacks: int = 0
    native_label_samples: int = 0
    ruby_stacks: int = 0
    jvm_stacks: int = 0
    php_stacks: int = 0
    dwarf_stacks: int = 0

    @property
    def dropped_sample_pct(self) -> float:
        total = self.cpu_samples + self.cpu_samples_lost
        cpu_rate = self.cpu_samples_lost / total if total else 0.0
        gtotal = self.gpu_events + self.gpu_ring_dropped
        gpu_rate = self.gpu_ring_dropped / gtotal if gtotal else 0.0
        return 100.0 * max(cpu_rate, gpu_rate)


class Agent:
    def __init__(self, flags: Optional[Flags] = None,
                 extra_destinations: Optional[List] = None,
                 relabel_configs=(),
                 enable_cpu: bool = True,
                 enable_gpu: bool = True) -> None:
        self.flags = flags or Flags()
        f = self.flags

        destinations = list(extra_destinations or [])
        if f.local_store.directory:
            destinations.append(LocalStoreDestination(
                f.local_store.directory,
                symbolize=f.local_store.symbolize))
        if f.offline_mode.storage_path:
            destinations.append(OfflineLogDestination(
                f.offline_mode.storage_path,
                rotation_interval=f.offline_mode.rotation_interval))
        if f.remote_store.address:
            from .reporter.grpc_client import RemoteStoreDestination

            destinations.append(RemoteStoreDestination(f))
        if f.rocm.merge_node_profiles:
            from .gpu.merge import DaemonNodeProfileDestination

            node_dir = os.path.join(
                f.local_store.directory or f.offline_mode.storage_path
                or ".", "node_profiles")
            destinations.append(DaemonNodeProfileDestination(
                node_dir, node=f.node))

        self.processes = ProcessTable()
        self.executables = ExecutableCache()

        providers = [
            SystemMetadataProvider(node=f.node,
                                   external_labels=f.metadata.external_labels),
            ProcessMetadataProvider(
                executable_cache=self.executables,
                enable_cmdline=f.metadata.enable_process_cmdline,
                include_env_vars=f.include_env_var),
            AgentMetadataProvider(),
        ]
        if ContainerMetadataProvider is not None:
            # In-cluster: run the node-filtered apiserver pod informer so
            # pod labels/annotations reach the relabel pipeline
            # (reference containermetadata.go:250-470); CRI/docker/env
            # remain as fallbacks outside Kubernetes.
            self.k8s_informer = None
            if os.environ.get("KUBERNETES_SERVICE_HOST"):
                try:
                    from .metadata.kubernetes import K8sPodInformer

                    self.k8s_informer = K8sPodInformer(
                        node=os.environ.get("NODE_NAME", f.node))
                    self.k8s_informer.start()
                except Exception:
                    log.debug("k8s informer unavailable", exc_info=True)
            cri_client = None
            if f.metadata.container_runtime_socket_path:
                try:
                    from .metadata.cri import CRIClient

                    cri_client = CRIClient(
                        endpoint="unix://" +
                        f.metadata.container_runtime_socket_path)
                except Exception:
                    log.debug("CRI socket override unavailable",
                              exc_info=True)
            providers.append(ContainerMetadataProvider(
                node=f.node, k8s_informer=self.k8s_informer,
                cri_client=cri_client))

        self.reporter = Reporter(
            destinations,
            metadata_providers=providers,
            relabel_configs=list(relabel_configs),
            cpu_sampling_frequency=f.profiling.cpu_sampling_frequency,
            batch_write_interval=f.remote_store.batch_write_interval,
            merge_gpu_profiles=f.merge_gpu_profiles,
            label_ttl_seconds=(0.0 if f.metadata.disable_caching
                               else f.profiling.label_ttl),
            disable_cpu_label=f.metadata.disable_cpu_label,
            disable_thread_id_label=f.metadata.disable_thread_id_label,
            disable_thread_comm_label=f.metadata.disable_thread_comm_label,
        )

        self.cpu_service = None
        if enable_cpu:
            from .cpu import CPUSamplerService

            self.cpu_service = CPUSamplerService(
                self.reporter,
                freq=f.profiling.cpu_sampling_frequency,
                dwarf_stacks=not f.dwarf_unwinding_disable,
                poll_interval=f.profiling.perf_event_buffer_poll_interval,
                ring_pages=f.profiling.perf_ring_pages,
                processes=self.processes,
                executables=self.executables,
            )
            if not f.dwarf_unwinding_disable:
                try:
                    from .cpu.unwind import EhFrameUnwinder

                    unwinder = EhFrameUnwinder(self.executables)
                    unwinder.processes = self.cpu_service.processes
                    self.cpu_service.unwinder = unwinder
                except ImportError:
                    pass
            if not f.python_unwinding_disable:
                try:
                    from .interp.python import PythonUnwinder

                    py_unwinder = PythonUnwinder()
                    if py_unwinder.available:
                        self.cpu_service.python_unwinder = py_unwinder
                except Exception:
                    log.debug("python unwinder unavailable", exc_info=True)
            if not f.ruby_unwinding_disable:
                try:
                    from .interp.ruby import RubyUnwinder

                    rb_unwinder = RubyUnwinder()
                    if rb_unwinder.available:
                        self.cpu_service.ruby_unwinder = rb_unwinder
                except Exception:
                    log.debug("ruby unwinder unavailable", exc_info=True)
            if not f.jvm_unwinding_disable:
                try:
                    from .interp.jvm import JvmUnwinder

                    self.cpu_service.jvm_unwinder = JvmUnwinder()
                except Exception:
                    log.debug("jvm unwinder unavailable", exc_info=True)
            if not f.php_unwinding_disable:
                try:
                    from .interp.php import PhpUnwinder

                    self.cpu_service.php_unwinder = PhpUnwinder()
                except Exception:
                    log.debug("php unwinder unavailable", exc_info=True)
            if f.symbolizer_jit_disable:
                self.cpu_service.perf_maps = None
            if not f.perl_unwinding_disable:
                try:
                    from .interp.perl import PerlUnwinder

                    pl_unwinder = PerlUnwinder()
                    if pl_unwinder.available:
                        self.cpu_service.perl_unwinder = pl_unwinder
                except Exception:
                    log.debug("perl unwinder unavailable", exc_info=True)

        # Off-CPU profiling (wallclock samples) when a threshold is set.
        self.offcpu_service = None
        if enable_cpu and f.off_cpu_threshold > 0:
            try:
                from .cpu.offcpu import OffCPUService

                self.offcpu_service = OffCPUService(
                    self.reporter, threshold=f.off_cpu_threshold,
                    processes=self.processes, executables=self.executables)
            except Exception:
                log.warning("off-cpu service unavailable", exc_info=True)

        # Probabilistic profiling gates the CPU sampler fleet-wide
        # (reference: main.go:541-548).
        self._prob_controller = None
        if self.cpu_service is not None and \
                f.profiling.probabilistic_threshold < 100:
            from .probabilistic import ProbabilisticController

            self._prob_controller = ProbabilisticController(
                self.cpu_service,
                threshold=f.profiling.probabilistic_threshold,
                interval_seconds=f.profiling.probabilistic_interval,
                machine_id=f.machine_id or None)

        self.gpu_service = None
        if enable_gpu and f.rocm.enable:
            try:
                from .gpu.service import GPUProfilerService

                self.gpu_service = GPUProfilerService(
                    self.reporter,
                    shm_dir=f.rocm.shm_dir,
                    pc_flush_interval=f.profiling.duration,
                    bucket_shift=f.rocm.bucket_bits,
                    processes=self.processes,
                    executables=self.executables,
                )
            except ImportError as e:
                log.warning("GPU service unavailable: %s", e)

    def start(self) -> None:
        self.reporter.start()
        if self._prob_controller is not None:
            self._prob_controller.start()  # owns the CPU sampler lifecycle
        elif self.cpu_service is not None:
            self.cpu_service.start()
        if self.offcpu_service is not None:
            self.offcpu_service.start()
        if self.gpu_service is not None:
            self.gpu_service.start()

    def stop(self) -> None:
        if self._prob_controller is not None:
            self._prob_controller.stop()
        elif self.cpu_service is not None:
            self.cpu_service.stop()
        if self.offcpu_service is not None:
            self.offcpu_service.stop()
        if self.gpu_service is not None:
            self.gpu_service.stop()
        if getattr(self, "k8s_informer", None) is not None:
            self.k8s_informer.stop()
        self.reporter.stop()

    def stats(self) -> AgentStats:
        s = AgentStats()
        if self.cpu_service is not None:
            s.cpu_samples = self.cpu_service.total_samples
            s.cpu_samples_lost = self.cpu_service.lost_samples
            s.python_stacks = self.cpu_service.python_stacks
            s.perl_stacks = self.cpu_service.perl_stacks
            s.native_label_samples = \
                self.cpu_service.native_labels.samples_labeled
            s.ruby_stacks = self.cpu_service.ruby_stacks
            s.jvm_stacks = self.cpu_service.jvm_stacks
            s.php_stacks = self.cpu_service.php_stacks
            s.dwarf_stacks = self.cpu_service.stacks_unwound_dwarf
        if self.gpu_service is not None:
            m = self.gpu_service.metrics
            s.gpu_events = sum(m.events_by_type.values())
            s.gpu_ring_dropped = m.ring_dropped
            s.pc_samples = m.pc_samples
            s.kernels_reported = m.kernels_reported
        return s


# -- in-process rocprofiler tool control (bench/smoke) ---------------------


def tool_library_path() -> str:
    from .native import rocprof_tool_path

    return rocprof_tool_path()


def tool_env(shm_dir: str = "/dev/shm", defer_start: bool = False,
             ring_bytes: int = 32 << 20, pc_sampling: bool = True,
             pc_interval: Optional[int] = None,
             launch_stacks: bool = True,
             heap_profiling: bool = False) -> dict:
    """Environment for a target HIP process to be profiled."""
    env = {
        "ROCP_TOOL_LIBRARIES": tool_library_path(),
        # gfx950 PC sampling is behind a beta gate in rocprofiler-sdk.
        "ROCPROFILER_PC_SAMPLING_BETA_ENABLED": "ON",
        "PARCA_GPU_SHM_DIR": shm_dir,
        "PARCA_GPU_RING_BYTES": str(ring_bytes),
        "PARCA_GPU_PC_SAMPLING": "1" if pc_sampling else "0",
        "PARCA_GPU_LAUNCH_STACKS": "1" if launch_stacks else "0",
    }
    if defer_start:
        env["PARCA_GPU_DEFER_START"] = "1"
    if pc_interval is not None:
        env["PARCA_GPU_PC_INTERVAL"] = str(pc_interval)
    if heap_profiling:
        # Allocation sampler for OOM heap profiles (oom/heap.py); its
        # shm state outlives the process so the agent can ship profiles
        # of OOM-killed workloads.
        heap_lib = os.path.join(os.path.dirname(tool_library_path()),
                                "libparca_heap.so")
        if os.path.exists(heap_lib):
            prior = os.environ.get("LD_PRELOAD", "")
            env["LD_PRELOAD"] = f"{heap_lib}:{prior}" if prior else heap_lib
            env["PARCA_HEAP_DIR"] = shm_dir
    return env


class InProcessToolControl:
    """Controls the rocprofiler tool already loaded in THIS process via
    ROCP_TOOL_LIBRARIES (dlopen returns the existing handle)."""

    def __init__(self) -> None:
        self._lib = ctypes.CDLL(tool_library_path())
        self._lib.parca_rocprof_start.restype = ctypes.c_int
        self._lib.parca_rocprof_stop.restype = ctypes.c_int
        self._lib.parca_rocprof_stats.restype = ctypes.c_int
        self._lib.parca_rocprof_stats.argtypes = [
            ctypes.POINTER(ctypes.c_uint64)]

    def start(self) -> int:
        return self._lib.parca_rocprof_start()

    def stop(self) -> int:
        return self._lib.parca_rocprof_stop()

    def flush(self) -> int:
        return self._lib.parca_rocprof_flush()

    def stats(self) -> dict:
        buf = (ctypes.c_uint64 * 3)()
        if self._lib.parca_rocprof_stats(buf) != 0:
            return {"written": 0, "dropped": 0, "launch_stacks": 0}
        return {"written": buf[0], "dropped": buf[1], "launch_stacks": buf[2]}
