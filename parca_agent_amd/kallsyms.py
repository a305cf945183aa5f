"""Kernel symbolization via /proc/kallsyms.

The reference resolves kernel frames through the fork's kallsyms reader;
here kernel IPs are symbolized agent-side (kernel symbols never leave the
host as debuginfo)."""

from __future__ import annotations

import bisect
from typing import List, Optional


class Kallsyms:
    def __init__(self, path: str = "/proc/kallsyms") -> None:
        addrs: List[int] = []
        names: List[str] = []
        try:
            with open(path) as fh:
                entries = []
                for line in fh:
                    parts = line.split()
                    if len(parts) < 3:
                        continue
                    addr_s, kind, name = parts[0], parts[1], parts[2]
                    if kind.lower() not in ("t", "w"):
                        continue
                    try:
                        addr = int(addr_s, 16)
                    except ValueError:
                        continue
                    if addr == 0:
                        continue  # kptr_restrict hides addresses
                    entries.append((addr, name))
            entries.sort()
            addrs = [a for a, _ in entries]
            names = [n for _, n in entries]
        except OSError:
            pass
        self._addrs = addrs
        self._names = names

    def __len__(self) -> int:
        return len(self._addrs)

    def lookup(self, addr: int) -> Optional[str]:
        i = bisect.bisect_right(self._addrs, addr) - 1
        if i < 0:
            return None
        return self._names[i]
