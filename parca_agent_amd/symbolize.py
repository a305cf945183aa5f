"""Shared frame resolution: runtime IP -> normalized Frame.

Used by both the CPU sampler service and the GPU launch-stack resolver so
host frames from either source intern identically (same FileID/address →
same pprof location)."""

from __future__ import annotations

from .model import Frame, FrameType, MappingFile
from .procmaps import ExecutableCache, ProcessTable


class FrameResolver:
    def __init__(self, processes: ProcessTable, executables: ExecutableCache,
                 on_executable=None) -> None:
        self.processes = processes
        self.executables = executables
        self.on_executable = on_executable
        self._seen: set = set()
        self.no_mapping = 0

    def resolve(self, pid: int, ip: int) -> Frame:
        proc = self.processes.ensure_maps(pid)
        mapping = proc.find_mapping(ip) if proc else None
        if mapping is None or not mapping.path.startswith("/"):
            self.no_mapping += 1
            path = mapping.path if mapping else ""
            return Frame(kind=FrameType.UNKNOWN, address=ip,
                         mapping=MappingFile(path=path) if path else None)
        info = self.executables.get(mapping.path)
        if info.error:
            return Frame(kind=FrameType.NATIVE, address=ip,
                         mapping=MappingFile(path=mapping.path))
        if self.on_executable is not None and \
                info.file_id not in self._seen:
            self._seen.add(info.file_id)
            try:
                self.on_executable(info)
            except Exception:
                pass
        addr = info.normalize(ip, mapping.start, mapping.file_offset)
        return Frame(
            kind=FrameType.NATIVE, address=addr,
            mapping=MappingFile(file_id=info.file_id, path=mapping.path,
                                build_id=info.build_id or ""))
