"""Agent-side .eh_frame unwinder wrapper.

Lazily builds stack-delta tables (native parse, csrc/sampler/ehframe.cc)
per executable and keeps per-process mapping tables in sync so
`CPUSamplerService` can recover stacks of frame-pointer-less binaries
from the DWARF-mode regs+stack captures (SURVEY.md §7 stage 4).
"""

from __future__ import annotations

import logging
import threading
from typing import Dict, List, Optional, Tuple

from ..elf import ELFFile
from ..procmaps import ExecutableCache, ProcessTable

log = logging.getLogger("parca_agent_amd.unwind")

_NO_LIMIT = 1 << 62


def cgroup_memory_limit() -> int:
    """The agent's cgroup memory limit in bytes, or a huge sentinel when
    unlimited/unreadable (v2 memory.max, v1 memory.limit_in_bytes)."""
    for path in ("/sys/fs/cgroup/memory.max",
                 "/sys/fs/cgroup/memory/memory.limit_in_bytes"):
        try:
            raw = open(path).read().strip()
        except OSError:
            continue
        if raw == "max":
            return _NO_LIMIT
        try:
            v = int(raw)
        except ValueError:
            continue
        return v if 0 < v < _NO_LIMIT else _NO_LIMIT
    return _NO_LIMIT


def default_row_budget(mem_limit: Optional[int] = None,
                       default_rows: int = 24_000_000) -> int:
    """Scale the stack-delta table budget to the agent's memory limit —
    the automemlimit analog (reference go.mod: automemlimit sets
    GOMEMLIMIT from the cgroup). Tables are 8 B/row after the round-2
    packing (ehframe.cc PackedRow), so the 24M-row default now costs
    ~190 MB instead of ~400 MB — within the <300 MB steady-state target
     while covering the same executables as the
    reference's 512 MiB DWARF memlock budget (flags.go:41-42). Tables
    may use ~40 % of the container limit; floor 1M rows."""
    if mem_limit is None:
        mem_limit = cgroup_memory_limit()
    from_limit = int(mem_limit * 0.4) // 8
    return max(min(default_rows, from_limit), 1_000_000)

# Index of registers in SampleEvent.regs (see _sampler.REGS_ORDER).
_IP_IDX = 8
_SP_IDX = 7
_BP_IDX = 6


class EhFrameUnwinder:
    def __init__(self, executables: ExecutableCache,
                 max_modules: int = 512,
                 max_total_rows: Optional[int] = None,
                 synchronous: bool = False) -> None:
        from ..native import sampler as native_sampler

        self._native = native_sampler().Unwinder()
        self.executables = executables
        self.max_modules = max_modules
        # Memory budget: 8 B/row packed -> 24M rows ~= 190 MB, the
        # analog of the reference's 512 MiB DWARF memlock budget
        # (flags.go:41-42), shrunk automatically in memory-limited
        # containers.
        self.max_total_rows = max_total_rows if max_total_rows is not None \
            else default_row_budget()
        self.total_rows = 0
        self.synchronous = synchronous
        self._mu = threading.Lock()
        # path -> module id; -1 = no table / failed; -2 = build pending
        # (large .eh_frame sections — libtorch-scale — parse in a worker
        # so the sample drain never stalls).
        self._modules: Dict[str, int] = {}
        self._build_queue: List[str] = []
        self._builder: Optional[threading.Thread] = None
        # pid -> mapping signature we last pushed natively
        self._pushed: Dict[int, Tuple] = {}
        # pid -> monotonic time of the last sync attempt while tables
        # were still building: re-syncing every sample would stat every
        # mapping at sample rate.
        self._pending_retry: Dict[int, float] = {}
        self.processes: Optional[ProcessTable] = None
        self.tables_built = 0
        self.table_failures = 0

    # -- module tables -----------------------------------------------------

    def _build_table(self, path: str) -> int:
        if self.total_rows >= self.max_total_rows:
            log.info("eh_frame row budget exhausted; skipping %s", path)
            return -1
        mid = -1
        try:
            with ELFFile.open(path) as elf:
                sec = elf.section(".eh_frame")
                if sec is not None and sec.size > 0:
                    data = elf.section_data(sec)
                    mid = self._native.add_module_from_eh_frame(
                        data, sec.addr,
                        max_rows=self.max_total_rows - self.total_rows)
                    self.total_rows += self._native.module_rows(mid)
                    self.tables_built += 1
        except (OSError, ValueError) as e:
            log.debug("eh_frame table failed for %s: %s", path, e)
            self.table_failures += 1
        return mid

    def _builder_loop(self) -> None:
        while True:
            with self._mu:
                if not self._build_queue:
                    self._builder = None
                    return
                path = self._build_queue.pop(0)
            mid = self._build_table(path)
            with self._mu:
                self._modules[path] = mid
                # Invalidate pushed mappings so processes pick up the
                # freshly built module on their next sample.
                self._pushed.clear()

    def _module_id(self, path: str) -> int:
        with self._mu:
            mid = self._modules.get(path)
            if mid is not None:
                return mid
            if len(self._modules) >= self.max_modules:
                return -1
            if self.synchronous:
                self._modules[path] = -1  # placeholder against recursion
            else:
                self._modules[path] = -2
                self._build_queue.append(path)
                if self._builder is None:
                    self._builder = threading.Thread(
                        target=self._builder_loop, name="ehframe-build",
                        daemon=True)
                    self._builder.start()
                return -2
        # synchronous path (tests / tools)
        mid = self._build_table(path)
        with self._mu:
            self._modules[path] = mid
        return mid

    def sync_process(self, pid: int) -> bool:
        """Push the process's executable mappings (with table module ids)
        into the native unwinder. Returns False if nothing usable."""
        if self.processes is None:
            return False
        proc = self.processes.ensure_maps(pid)
        if proc is None:
            return False
        maps = proc.mappings
        sig = tuple((m.start, m.path) for m in maps)
        if self._pushed.get(pid) == sig:
            return True
        import time as _time

        now = _time.monotonic()
        if now - self._pending_retry.get(pid, 0.0) < 1.0:
            return True  # tables still building; native maps unchanged
        self._pending_retry[pid] = now
        native_maps: List[Tuple[int, int, int, int]] = []
        pending = False
        for m in maps:
            if not m.path.startswith("/"):
                continue
            mid = self._module_id(m.path)
            if mid == -2:
                pending = True
                continue
            if mid < 0:
                continue
            info = self.executables.get(m.path)
            # bias = runtime_addr - link_vaddr for this mapping:
            # runtime ip -> vaddr via info.normalize; bias is constant per
            # PT_LOAD segment. Compute from the mapping start.
            vaddr_of_start = info.normalize(m.start, m.start, m.file_offset)
            bias = m.start - vaddr_of_start
            native_maps.append((m.start, m.end, bias, mid))
        if not native_maps:
            return False
        self._native.set_mappings(pid, native_maps)
        if not pending:  # re-push once pending tables finish building
            self._pushed[pid] = sig
        return True

    # -- unwinding ---------------------------------------------------------

    def unwind(self, pid: int, regs, stack: bytes) -> List[int]:
        if not regs or not stack:
            return []
        if not self.sync_process(pid):
            return []
        ip = regs[_IP_IDX]
        sp = regs[_SP_IDX]
        bp = regs[_BP_IDX]
        return list(self._native.unwind(pid, ip, sp, bp, stack))

    def drop_process(self, pid: int) -> None:
        self._pushed.pop(pid, None)
        self._pending_retry.pop(pid, None)
        self._native.drop_process(pid)
