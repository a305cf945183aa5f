"""Native custom-labels reader: join per-thread labels onto CPU samples.

Programs opt in by including ``include/parca_custom_labels.h`` and
calling ``parca_label_set(key, value)``; the header publishes a
per-thread label table in ``/dev/shm/parca_labels_<pid>`` (fixed
layout, per-slot seqlock). This module is the agent half: per sample
it maps the pid's table (cached) and reads the tid's slot with seqlock
retries, returning the label tuple the reporter attaches to the sample
(reference capability: native custom labels, written out at
parca_reporter.go:362-374 with per-label counters metrics/all.go:1442-
1477 — transport re-designed for a userspace perf agent, see header).
"""

from __future__ import annotations

import mmap
import os
import struct
import time
from typing import Dict, Optional, Tuple

MAGIC = 0x53424C4350  # "PCLBS"
HEADER = struct.Struct("<QIIIIII32x")  # magic, ver, nslots, max, klen, vlen, pad
SLOT_HEAD = struct.Struct("<IIII")     # tid, seq, count, pad

# Re-probe an absent file at most this often per pid (most processes
# never publish labels; the negative cache keeps the per-sample cost to
# one dict hit).
NEGATIVE_TTL = 5.0


class _Table:
    __slots__ = ("mm", "nslots", "max_labels", "key_len", "val_len",
                 "slot_size", "base")

    def __init__(self, mm: mmap.mmap):
        magic, ver, nslots, maxl, klen, vlen, _ = HEADER.unpack_from(mm, 0)
        if magic != MAGIC or ver != 1 or not (0 < nslots <= 65536) or \
                not (0 < maxl <= 64) or not (0 < klen <= 256) or \
                not (0 < vlen <= 1024):
            raise ValueError("bad label table header")
        self.mm = mm
        self.nslots = nslots
        self.max_labels = maxl
        self.key_len = klen
        self.val_len = vlen
        self.slot_size = SLOT_HEAD.size + maxl * (klen + vlen)
        self.base = HEADER.size

    def read_slot(self, tid: int) -> Tuple[Tuple[str, str], ...]:
        idx = tid % self.nslots
        for probe in range(self.nslots):
            off = self.base + ((idx + probe) % self.nslots) * self.slot_size
            if off + self.slot_size > len(self.mm):
                return ()
            slot_tid, _, _, _ = SLOT_HEAD.unpack_from(self.mm, off)
            if slot_tid == 0:
                return ()  # free slot terminates the probe chain
            if slot_tid == tid:
                return self._consistent_read(off)
        return ()

    def _consistent_read(self, off: int) -> Tuple[Tuple[str, str], ...]:
        for _ in range(4):
            _, seq1, count, _ = SLOT_HEAD.unpack_from(self.mm, off)
            if seq1 & 1:  # writer mid-update
                continue
            labels = []
            lo = off + SLOT_HEAD.size
            pair = self.key_len + self.val_len
            for i in range(self.max_labels):
                if len(labels) >= count:
                    break
                p = lo + i * pair
                raw_k = self.mm[p:p + self.key_len]
                k = raw_k.split(b"\0", 1)[0]
                if not k:
                    continue
                raw_v = self.mm[p + self.key_len:p + pair]
                v = raw_v.split(b"\0", 1)[0]
                labels.append((k.decode("utf-8", "replace"),
                               v.decode("utf-8", "replace")))
            _, seq2, _, _ = SLOT_HEAD.unpack_from(self.mm, off)
            if seq1 == seq2:
                return tuple(labels)
        return ()  # writer kept racing us: drop labels, never ship torn ones


class NativeLabelReader:
    """Per-pid table cache + per-sample (pid, tid) -> labels join."""

    def __init__(self, directory: Optional[str] = None):
        self.directory = directory or os.environ.get(
            "PARCA_LABELS_DIR", "/dev/shm")
        self._tables: Dict[int, _Table] = {}
        self._negative: Dict[int, float] = {}
        self.samples_labeled = 0
        self.read_errors = 0

    def _path(self, pid: int) -> str:
        return os.path.join(self.directory, f"parca_labels_{pid}")

    def _table_for(self, pid: int) -> Optional[_Table]:
        t = self._tables.get(pid)
        if t is not None:
            return t
        now = time.monotonic()
        exp = self._negative.get(pid)
        if exp is not None and now < exp:
            return None
        try:
            with open(self._path(pid), "rb") as fh:
                # Spoof guard: only trust a table created by the process
                # it claims to describe — the file owner must match the
                # profiled process's uid (or be root).
                st = os.fstat(fh.fileno())
                try:
                    proc_uid = os.stat(f"/proc/{pid}").st_uid
                except OSError:
                    proc_uid = None  # process already gone: accept
                if proc_uid is not None and \
                        st.st_uid not in (proc_uid, 0):
                    raise ValueError("label table owner mismatch")
                mm = mmap.mmap(fh.fileno(), 0, mmap.MAP_SHARED,
                               mmap.PROT_READ)
            t = _Table(mm)
        except (OSError, ValueError):
            self._negative[pid] = now + NEGATIVE_TTL
            if len(self._negative) > 16384:
                self._negative.clear()
            return None
        self._tables[pid] = t
        return t

    def labels_for(self, pid: int, tid: int) -> Tuple[Tuple[str, str], ...]:
        t = self._table_for(pid)
        if t is None:
            return ()
        try:
            labels = t.read_slot(tid)
        except (ValueError, IndexError):
            self.read_errors += 1
            return ()
        if labels:
            self.samples_labeled += 1
        return labels

    def forget(self, pid: int) -> None:
        """Drop a dead pid's mapping (called from process-exit sweeps)."""
        t = self._tables.pop(pid, None)
        if t is not None:
            try:
                t.mm.close()
            except Exception:
                pass
        self._negative.pop(pid, None)

    def sweep(self) -> None:
        """Unmap tables whose process is gone and unlink their files
        (the agent is the janitor for crashed processes' tables)."""
        for pid in list(self._tables):
            if not os.path.exists(f"/proc/{pid}"):
                self.forget(pid)
                try:
                    os.unlink(self._path(pid))
                except OSError:
                    pass
