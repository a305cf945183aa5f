"""Process metadata provider: /proc-derived labels.

Reference: reporter/metadata/process.go:199-246 (cmdline/comm/cgroup/stat)
and 152-197 (main executable identity + compiler/static/stripped labels).
"""

from __future__ import annotations

from typing import Dict, Optional

from .. import procmaps
from ..elf import ELFFile


class ProcessMetadataProvider:
    name = "process"

    def __init__(self, executable_cache=None, enable_cmdline: bool = False,
                 include_env_vars=()) -> None:
        self._exe_cache = executable_cache
        self._enable_cmdline = enable_cmdline
        # Env var names to surface as labels (reference flag
        # --include-env-var, flags.go IncludeEnvVar).
        self._include_env = tuple(include_env_vars)

    def add_metadata(self, pid: int, labels: Dict[str, str]) -> bool:
        comm = procmaps.read_comm(pid)
        if comm is None:
            return False
        labels.setdefault("comm", comm)
        exe = procmaps.read_exe(pid)
        if exe:
            labels["__meta_process_executable"] = exe
            labels.setdefault("executable", exe.rsplit("/", 1)[-1])
            if self._exe_cache is not None:
                info = self._exe_cache.get(exe)
                if info.build_id:
                    labels["__meta_process_executable_build_id"] = info.build_id
                if info.file_id:
                    labels["__meta_process_executable_file_id"] = info.file_id
            self._add_elf_labels(exe, labels)
        cgroup = procmaps.read_cgroup(pid)
        if cgroup:
            labels["__meta_process_cgroup"] = cgroup
        if self._enable_cmdline:
            cmdline = procmaps.read_cmdline(pid)
            if cmdline:
                labels["cmdline"] = cmdline
        ppid = _read_ppid(pid)
        if ppid is not None:
            labels["__meta_process_ppid"] = str(ppid)
        if self._include_env:
            for name, value in _read_environ(pid, self._include_env):
                labels[f"env_{name.lower()}"] = value
        return True

    @staticmethod
    def _add_elf_labels(exe: str, labels: Dict[str, str]) -> None:
        """stripped/static hints (the ainur-analog, process.go:152-197)."""
        try:
            with ELFFile.open(exe) as elf:
                labels["__meta_executable_stripped"] = str(
                    elf.is_stripped()).lower()
                has_interp = any(s.name == ".interp" for s in elf.sections)
                labels["__meta_executable_static"] = str(not has_interp).lower()
        except (OSError, ValueError):
            pass


def _read_environ(pid: int, names) -> list:
    out = []
    try:
        with open(f"/proc/{pid}/environ", "rb") as fh:
            data = fh.read(131072)
    except OSError:
        return out
    wanted = set(names)
    for chunk in data.split(b"\x00"):
        key, sep, value = chunk.partition(b"=")
        if not sep:
            continue
        k = key.decode("utf-8", "replace")
        if k in wanted:
            out.append((k, value.decode("utf-8", "replace")))
    return out


def _read_ppid(pid: int) -> Optional[int]:
    try:
        with open(f"/proc/{pid}/status") as fh:
            for line in fh:
                if line.startswith("PPid:"):
                    return int(line.split()[1])
    except (OSError, ValueError):
        pass
    return None
