"""parca-agent-amd: an MI355X-native always-on sampling profiler.

A from-scratch rebuild of the capabilities of parca-dev/parca-agent
(reference: /root/reference) for AMD MI355X (gfx950) nodes:

- CPU sampling at 19 Hz via perf_event_open (native C++ core, frame-pointer
  and .eh_frame unwinding) instead of the reference's eBPF fork
  (reference: main.go:496-607).
- GPU profiling via rocprofiler-sdk (kernel dispatch intercept + gfx950 PC
  sampling) instead of CUPTI (reference: parcagpu/parcagpu.go).
- PC-sample bucketing on-device with a hand-written CDNA4 HIP kernel.
- Node-level multi-GPU profile merge over RCCL/xGMI (no reference analog).
- Same egress surface: pprof wire format, Arrow v1/v2 sample records,
  ProfileStore Write/WriteArrow gRPC, debuginfo upload protocol
  (reference: reporter/parca_reporter.go, reporter/arrow_v2.go).
"""

from .version import __version__

__all__ = ["__version__"]
