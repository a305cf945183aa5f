"""Prometheus-style relabeling engine.

The reference applies `relabel_configs` from its YAML config to the label
set of every PID (and per-sample for probe-origin traces) before shipping
samples (reference: config/config.go:27-55,
reporter/parca_reporter.go:779-841). This module reimplements the
Prometheus relabel semantics natively: actions replace, keep, drop,
keepequal, dropequal, hashmod, labelmap, labeldrop, labelkeep,
lowercase, uppercase.
"""

from __future__ import annotations

import hashlib
import re
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence

DEFAULT_SEPARATOR = ";"

# Labels prefixed with this are internal-only and stripped before export,
# matching Prometheus meta-label behaviour (reference strips __meta_*
# labels after relabeling, parca_reporter.go:789-798).
META_PREFIX = "__"


@dataclass
class RelabelConfig:
    source_labels: List[str] = field(default_factory=list)
    separator: str = DEFAULT_SEPARATOR
    target_label: str = ""
    regex: str = "(.*)"
    modulus: int = 0
    replacement: str = "$1"
    action: str = "replace"

    _compiled: Optional[re.Pattern] = field(default=None, repr=False, compare=False)

    def __post_init__(self) -> None:
        # Prometheus anchors relabel regexes on both ends.
        self._compiled = re.compile("^(?:" + self.regex + ")$")
        self.action = self.action.lower()
        valid = {
            "replace", "keep", "drop", "keepequal", "dropequal", "hashmod",
            "labelmap", "labeldrop", "labelkeep", "lowercase", "uppercase",
        }
        if self.action not in valid:
            raise ValueError(f"unknown relabel action {self.action!r}")
        if self.action in ("replace", "hashmod", "lowercase", "uppercase",
                           "keepequal", "dropequal") and not self.target_label:
            raise ValueError(f"relabel action {self.action!r} requires target_label")

    @classmethod
    def from_dict(cls, d: dict) -> "RelabelConfig":
        return cls(
            source_labels=list(d.get("source_labels", []) or []),
            separator=d.get("separator", DEFAULT_SEPARATOR),
            target_label=d.get("target_label", ""),
            regex=str(d.get("regex", "(.*)")),
            modulus=int(d.get("modulus", 0) or 0),
            replacement=str(d.get("replacement", "$1")),
            action=str(d.get("action", "replace")),
        )


def _expand(template: str, match: re.Match) -> str:
    """Expand $1 / ${name} references like Prometheus does."""

    def repl(m: re.Match) -> str:
        ref = m.group(1) or m.group(2)
        try:
            if ref.isdigit():
                return match.group(int(ref)) or ""
            return match.group(ref) or ""
        except (IndexError, re.error):
            return ""

    return re.sub(r"\$(?:\{(\w+)\}|(\w+))", repl, template)


def relabel(
    labels: Dict[str, str], configs: Sequence[RelabelConfig]
) -> Optional[Dict[str, str]]:
    """Apply relabel configs. Returns None if the series is dropped."""
    lb = dict(labels)
    for cfg in configs:
        assert cfg._compiled is not None
        value = cfg.separator.join(lb.get(name, "") for name in cfg.source_labels)
        action = cfg.action
        if action == "drop":
            if cfg._compiled.match(value):
                return None
        elif action == "keep":
            if not cfg._compiled.match(value):
                return None
        elif action == "dropequal":
            if lb.get(cfg.target_label, "") == value:
                return None
        elif action == "keepequal":
            if lb.get(cfg.target_label, "") != value:
                return None
        elif action == "replace":
            m = cfg._compiled.match(value)
            if m is None:
                continue
            target = _expand(cfg.target_label, m)
            replacement = _expand(cfg.replacement, m)
            if not target:
                continue
            if replacement == "":
                lb.pop(target, None)
            else:
                lb[target] = replacement
        elif action == "lowercase":
            lb[cfg.target_label] = value.lower()
        elif action == "uppercase":
            lb[cfg.target_label] = value.upper()
        elif action == "hashmod":
            h = int.from_bytes(hashlib.md5(value.encode()).digest()[-8:], "big")
            lb[cfg.target_label] = str(h % cfg.modulus)
        elif action == "labelmap":
            for name in list(lb):
                m = cfg._compiled.match(name)
                if m:
                    lb[_expand(cfg.replacement, m)] = lb[name]
        elif action == "labeldrop":
            for name in list(lb):
                if cfg._compiled.match(name):
                    del lb[name]
        elif action == "labelkeep":
            for name in list(lb):
                if not cfg._compiled.match(name):
                    del lb[name]
    return lb


def strip_meta_labels(labels: Dict[str, str]) -> Dict[str, str]:
    return {k: v for k, v in labels.items() if not k.startswith(META_PREFIX)}
