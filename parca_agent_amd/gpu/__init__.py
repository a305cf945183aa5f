from . import events
from .codeobj import CodeObjectRegistry, parse_code_object_uri
from .fixer import CompletedKernel, GpuTraceFixer
from .pcbuckets import BucketLayout, HostAccumulator
from .service import GPUProfilerService

__all__ = [
    "events",
    "CodeObjectRegistry",
    "parse_code_object_uri",
    "CompletedKernel",
    "GpuTraceFixer",
    "BucketLayout",
    "HostAccumulator",
    "GPUProfilerService",
]
