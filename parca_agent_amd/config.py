"""YAML config file loading: relabel_configs section.

The same file that overlays CLI flags also carries Prometheus
`relabel_configs` applied to every PID's label set (reference:
config/config.go:27-55, config.yaml)."""

from __future__ import annotations

from typing import List

import yaml

from .relabel import RelabelConfig


def load_relabel_configs(path: str) -> List[RelabelConfig]:
    with open(path) as fh:
        doc = yaml.safe_load(fh) or {}
    return parse_relabel_configs(doc)


def parse_relabel_configs(doc: dict) -> List[RelabelConfig]:
    raw = doc.get("relabel_configs") or []
    if not isinstance(raw, list):
        raise ValueError("relabel_configs must be a list")
    return [RelabelConfig.from_dict(item) for item in raw]
