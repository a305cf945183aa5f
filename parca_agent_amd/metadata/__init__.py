"""Metadata providers enriching per-PID label sets.

Provider protocol mirrors the reference's
`MetadataProvider.AddMetadata(pid, labels)` (reference:
reporter/metadata/containermetadata.go:98-103): each provider mutates the
label dict; meta labels (`__meta_*`) are relabel-input only and stripped
before export.
"""

from .agent import AgentMetadataProvider
from .process import ProcessMetadataProvider
from .system import SystemMetadataProvider

try:  # container runtime metadata needs nothing exotic, but keep it isolated
    from .container import ContainerMetadataProvider
except ImportError:  # pragma: no cover
    ContainerMetadataProvider = None  # type: ignore

__all__ = [
    "AgentMetadataProvider",
    "ProcessMetadataProvider",
    "SystemMetadataProvider",
    "ContainerMetadataProvider",
]
