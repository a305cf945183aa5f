"""Agent metadata provider (reference: reporter/metadata/agent.go:14-21)."""

from __future__ import annotations

from typing import Dict

from ..version import REVISION


class AgentMetadataProvider:
    name = "agent"

    def add_metadata(self, pid: int, labels: Dict[str, str]) -> bool:
        labels.setdefault("profiler_agent_revision", REVISION)
        labels.setdefault("profiler_agent", "parca-agent-amd")
        return True
