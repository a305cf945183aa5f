"""System metadata provider: uname + node labels.

Reference: reporter/metadata/system.go:29-45.
"""

from __future__ import annotations

import os
from typing import Dict


class SystemMetadataProvider:
    name = "system"

    def __init__(self, node: str = "", external_labels: Dict[str, str] = None) -> None:
        uname = os.uname()
        self._static = {
            "node": node or uname.nodename,
            "__meta_system_kernel_release": uname.release,
            "__meta_system_kernel_machine": uname.machine,
            "__meta_system_sysname": uname.sysname,
        }
        if external_labels:
            self._static.update(external_labels)

    def add_metadata(self, pid: int, labels: Dict[str, str]) -> bool:
        for k, v in self._static.items():
            labels.setdefault(k, v)
        return True
