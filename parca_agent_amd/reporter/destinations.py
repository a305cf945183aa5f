"""Flush destinations: local store directory, offline framed log.

- LocalStoreDestination writes one gzipped pprof per sample type per flush
  into `--local-store-directory` (reference flag: flags.go LocalStore).
- OfflineLogDestination writes framed Arrow batches with the same
  magic/patched-count/fsync discipline as the reference's .padata log
  (reference: parca_reporter.go:1366-1438, 1527-1665), rotated with zstd.
"""

from __future__ import annotations

import logging
import os
import struct
import threading
import time
from typing import Dict, List, Optional, Tuple

from ..pprof import FrameKey, MappingKey, ProfileBuilder, ValueType
from .reporter import PendingSample, build_arrow_record

log = logging.getLogger("parca_agent_amd.destinations")

OFFLINE_MAGIC = 0xA6E7CCCA
OFFLINE_VERSION = 2  # arrow_v2 frames


class LocalSymbolizer:
    """Best-effort agent-side symbolization for locally stored pprof.

    The remote path leaves native frames as (file_id, address) for the
    Parca server's symbolizer (reference behaviour), but local-store
    profiles have no server — resolve what the on-host ELF symtab /
    dynsym can name. Indexes build in a background thread (libtorch's
    symtab takes >1 s to index — that must never sit on the flush
    path, which bench.py measures as agent CPU); a module's frames stay
    address-only until its index lands, typically one flush interval.
    """

    def __init__(self, max_modules: int = 256) -> None:
        import threading

        from ..lru import LRU

        self._indexes: LRU = LRU(max_modules)
        self._mu = threading.Lock()
        self._queue: List[str] = []
        self._builder = None

    def _build_loop(self) -> None:
        from ..elf import ELFFile, SymbolIndex

        while True:
            with self._mu:
                if not self._queue:
                    self._builder = None
                    return
                path = self._queue.pop(0)
            try:
                with ELFFile.open(path) as elf:
                    idx = SymbolIndex(elf.symbols())
            except (OSError, ValueError):
                idx = False  # negative cache
            self._indexes.put(path, idx)

    def name_for(self, path: str, vaddr: int) -> str:
        import threading

        idx = self._indexes.get(path)
        if idx is None:
            with self._mu:
                if path not in self._queue:
                    self._queue.append(path)
                if self._builder is None:
                    self._builder = threading.Thread(
                        target=self._build_loop, name="local-symbolize",
                        daemon=True)
                    self._builder.start()
            return ""
        if not idx:
            return ""
        sym = idx.lookup(vaddr)
        return sym.name if sym is not None else ""

    def wait_idle(self, timeout: float = 10.0) -> None:
        """Tests/shutdown: block until queued builds finish."""
        import time as _time

        deadline = _time.monotonic() + timeout
        while _time.monotonic() < deadline:
            with self._mu:
                if self._builder is None:
                    return
            _time.sleep(0.02)


def samples_to_pprof(samples: List[PendingSample],
                     symbolizer: Optional[LocalSymbolizer] = None
                     ) -> Dict[str, bytes]:
    """Group a flush batch by sample type and encode each as pprof."""
    groups: Dict[Tuple[str, str, str, str, int], List[PendingSample]] = {}
    for s in samples:
        key = (s.sample_type.sample_type, s.sample_type.sample_unit,
               s.sample_type.period_type, s.sample_type.period_unit, s.period)
        groups.setdefault(key, []).append(s)

    # Two groups can share a sample type while differing in period (e.g.
    # gpu_pcsample from two processes with different PC-sampling
    # intervals): keep BOTH by suffixing the output key with a group
    # index instead of silently overwriting (one file/payload per group).
    stype_counts: Dict[str, int] = {}
    for (stype, _, _, _, _) in groups:
        stype_counts[stype] = stype_counts.get(stype, 0) + 1
    stype_seq: Dict[str, int] = {}

    out: Dict[str, bytes] = {}
    for (stype, sunit, ptype, punit, period), group in groups.items():
        builder = ProfileBuilder(
            sample_types=[ValueType(stype, sunit)],
            period_type=ValueType(ptype, punit),
            period=period,
        )
        # Frames and traces are interned upstream (cpu/service.py frame
        # intern + reporter stack LRU), so key the expensive
        # FrameKey/location work on object identity: repeated stacks
        # skip straight to pre-resolved location-id tuples. This took a
        # 59k-sample batch encode from ~15 s to well under a second.
        mk_cache: Dict[int, Optional[MappingKey]] = {}
        fk_cache: Dict[int, FrameKey] = {}
        trace_locs: Dict[int, tuple] = {}
        for s in group:
            loc_ids = trace_locs.get(id(s.trace))
            if loc_ids is None:
                frames = []
                for f in s.trace.frames:
                    fk = fk_cache.get(id(f))
                    if fk is None:
                        mapping = None
                        if f.mapping is not None:
                            mapping = mk_cache.get(id(f.mapping))
                            if mapping is None:
                                mapping = MappingKey(
                                    memory_start=0, memory_limit=0,
                                    file_offset=0,
                                    filename=f.mapping.path,
                                    build_id=f.mapping.id_label)
                                mk_cache[id(f.mapping)] = mapping
                        name = f.function_name
                        if not name and symbolizer is not None and \
                                f.mapping is not None and \
                                f.mapping.path.startswith("/"):
                            name = symbolizer.name_for(
                                f.mapping.path, f.address)
                        fk = FrameKey(
                            address=f.address, mapping=mapping,
                            function_name=name,
                            source_file=f.source_file,
                            line=f.source_line)
                        fk_cache[id(f)] = fk
                    frames.append(fk)
                loc_ids = tuple(builder.location_id(fk) for fk in frames)
                trace_locs[id(s.trace)] = loc_ids
            builder.add_sample_by_ids(loc_ids, [s.value],
                                      labels=sorted(s.labels.items()))
        key = stype
        if stype_counts[stype] > 1:
            n = stype_seq.get(stype, 0)
            stype_seq[stype] = n + 1
            if n:
                key = f"{stype}.{n}"
        out[key] = builder.serialize_gzip()
    return out


class LocalStoreDestination:
    # symbolize defaults OFF: indexing libtorch-scale symtabs costs
    # seconds of agent CPU (measured 0.13% -> 0.29% node overhead), and
    # the reference's local store is unsymbolized too. Opt in with
    # --local-store-symbolize when there is no Parca server to do it.
    def __init__(self, directory: str, symbolize: bool = False) -> None:
        self.directory = directory
        os.makedirs(directory, exist_ok=True)
        self._seq = 0
        self.symbolizer = LocalSymbolizer() if symbolize else None

    def write_batch(self, samples: List[PendingSample]) -> None:
        profiles = samples_to_pprof(samples, self.symbolizer)
        ts = int(time.time())
        for stype, data in profiles.items():
            path = os.path.join(
                self.directory, f"{ts}.{self._seq:06d}.{stype}.pb.gz")
            tmp = path + ".tmp"
            with open(tmp, "wb") as fh:
                fh.write(data)
            os.replace(tmp, path)
        self._seq += 1

    def close(self) -> None:
        pass


class OfflineLogDestination:
    """Framed offline log: header `magic u32 | version u16 | count u32`,
    then frames `length u64 | arrow-ipc-bytes`. The count at offset 6 is
    patched after each frame is fsynced so a torn final frame is detectable
    on replay (reference ordering discipline: parca_reporter.go:1366-1380).
    """

    HEADER = struct.Struct("<IHI")

    def __init__(self, storage_path: str,
                 rotation_interval: float = 600.0) -> None:
        self.dir = storage_path
        os.makedirs(storage_path, exist_ok=True)
        self.rotation_interval = rotation_interval
        self._mu = threading.Lock()
        self._fh: Optional[object] = None
        self._count = 0
        self._opened_at = 0.0
        self._path = ""

    def _open_new(self) -> None:
        ts = time.strftime("%Y%m%dT%H%M%S")
        self._path = os.path.join(self.dir, f"profiles-{ts}-{os.getpid()}.padata")
        self._fh = open(self._path, "wb")
        self._fh.write(self.HEADER.pack(OFFLINE_MAGIC, OFFLINE_VERSION, 0))
        self._fh.flush()
        os.fsync(self._fh.fileno())
        self._count = 0
        self._opened_at = time.monotonic()

    def write_batch(self, samples: List[PendingSample]) -> None:
        from .arrow_v2 import serialize_record

        record = build_arrow_record(samples)
        payload = serialize_record(record)
        with self._mu:
            if self._fh is None:
                self._open_new()
            assert self._fh is not None
            self._fh.write(struct.pack("<Q", len(payload)))
            self._fh.write(payload)
            self._fh.flush()
            os.fsync(self._fh.fileno())
            # Patch batch count at offset 6 only after the frame is durable.
            self._count += 1
            pos = self._fh.tell()
            self._fh.seek(6)
            self._fh.write(struct.pack("<I", self._count))
            self._fh.seek(pos)
            self._fh.flush()
            os.fsync(self._fh.fileno())
            if time.monotonic() - self._opened_at > self.rotation_interval:
                self._rotate_locked()

    def _rotate_locked(self) -> None:
        assert self._fh is not None
        self._fh.close()
        self._fh = None
        try:
            self._compress(self._path)
        except Exception:
            log.warning("offline log compression failed for %s",
                        self._path, exc_info=True)

    @staticmethod
    def _compress(path: str) -> None:
        import pyarrow as pa

        codec = pa.Codec("zstd")
        with open(path, "rb") as fh:
            data = fh.read()
        compressed = codec.compress(data, asbytes=True)
        tmp = path + ".zst.tmp"
        with open(tmp, "wb") as fh:
            # Store the decompressed size so replay can allocate.
            fh.write(struct.pack("<Q", len(data)))
            fh.write(compressed)
        os.replace(tmp, path + ".zst")
        os.unlink(path)

    def rotate(self) -> None:
        with self._mu:
            if self._fh is not None:
                self._rotate_locked()

    def close(self) -> None:
        self.rotate()


def read_offline_log(path: str) -> List[bytes]:
    """Read the framed batches (arrow IPC payloads) from a .padata or
    .padata.zst file, honouring the patched count."""
    data: bytes
    if path.endswith(".zst"):
        import pyarrow as pa

        with open(path, "rb") as fh:
            raw = fh.read()
        (dec_size,) = struct.unpack_from("<Q", raw, 0)
        data = pa.Codec("zstd").decompress(raw[8:], dec_size, asbytes=True)
    else:
        with open(path, "rb") as fh:
            data = fh.read()
    if len(data) < OfflineLogDestination.HEADER.size:
        raise ValueError(f"{path}: truncated header")
    magic, version, count = OfflineLogDestination.HEADER.unpack_from(data, 0)
    if magic != OFFLINE_MAGIC:
        raise ValueError(f"{path}: bad magic {magic:#x}")
    if version != OFFLINE_VERSION:
        raise ValueError(f"{path}: unsupported version {version}")
    out = []
    pos = OfflineLogDestination.HEADER.size
    for _ in range(count):
        if pos + 8 > len(data):
            break  # torn frame beyond the patched count
        (length,) = struct.unpack_from("<Q", data, pos)
        pos += 8
        if pos + length > len(data):
            break
        out.append(data[pos : pos + length])
        pos += length
    return out
