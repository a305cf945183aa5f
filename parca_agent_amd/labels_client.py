"""Pure-Python writer for the native custom-labels table.

The C half lives in ``include/parca_custom_labels.h``; this module
writes the same fixed layout so Python services can publish per-thread
labels the agent joins onto their CPU samples (and, through the
interpreter unwinder, onto their Python stacks). Standalone by design:
copy this one file into any application — it has no dependencies on
the rest of the package.

Usage:
    from parca_agent_amd.labels_client import label_set
    label_set("endpoint", "/checkout")
    ...
    label_set("endpoint", "")   # empty value deletes
"""

from __future__ import annotations

import mmap
import os
import struct
import threading

MAGIC = 0x53424C4350  # "PCLBS"
NSLOTS = 512
MAX_LABELS = 8
KEY_LEN = 32
VAL_LEN = 64
HEADER = struct.Struct("<QIIIIII32x")
SLOT_HEAD = struct.Struct("<IIII")
SLOT_SIZE = SLOT_HEAD.size + MAX_LABELS * (KEY_LEN + VAL_LEN)

_lock = threading.Lock()
_mm: mmap.mmap | None = None
_pid = -1
_slot_off = threading.local()


def _table() -> mmap.mmap | None:
    global _mm, _pid
    pid = os.getpid()
    if _mm is not None and _pid == pid:
        return _mm
    with _lock:
        if _mm is not None and _pid == pid:
            return _mm
        directory = os.environ.get("PARCA_LABELS_DIR", "/dev/shm")
        path = os.path.join(directory, f"parca_labels_{pid}")
        size = HEADER.size + NSLOTS * SLOT_SIZE
        try:
            fd = os.open(path, os.O_RDWR | os.O_CREAT, 0o644)
            try:
                if os.fstat(fd).st_size < size:
                    os.ftruncate(fd, size)
                mm = mmap.mmap(fd, size, mmap.MAP_SHARED,
                               mmap.PROT_READ | mmap.PROT_WRITE)
            finally:
                os.close(fd)
        except OSError:
            return None
        magic = struct.unpack_from("<Q", mm, 0)[0]
        if magic != MAGIC:
            HEADER.pack_into(mm, 0, MAGIC, 1, NSLOTS, MAX_LABELS,
                             KEY_LEN, VAL_LEN, 0)
        _mm, _pid = mm, pid
        return mm


def _my_slot(mm: mmap.mmap) -> int | None:
    off = getattr(_slot_off, "off", None)
    if off is not None and getattr(_slot_off, "pid", -1) == os.getpid():
        return off
    tid = threading.get_native_id()
    idx = tid % NSLOTS
    for probe in range(NSLOTS):
        off = HEADER.size + ((idx + probe) % NSLOTS) * SLOT_SIZE
        cur = struct.unpack_from("<I", mm, off)[0]
        if cur in (0, tid):
            # The GIL (or, under free threading, benign same-process
            # races resolved by identical tid writes) makes a plain
            # store sufficient here, unlike the cross-process C side.
            struct.pack_into("<I", mm, off, tid)
            _slot_off.off = off
            _slot_off.pid = os.getpid()
            return off
    return None


def label_set(key: str, value: str) -> None:
    """Set (or with value='' delete) one label on the calling thread."""
    if not key:
        return
    mm = _table()
    if mm is None:
        return
    off = _my_slot(mm)
    if off is None:
        return
    kb = key.encode("utf-8")[:KEY_LEN - 1]
    vb = value.encode("utf-8")[:VAL_LEN - 1]
    _, seq, count, _ = SLOT_HEAD.unpack_from(mm, off)
    struct.pack_into("<I", mm, off + 4, seq | 1)  # odd: writing
    try:
        found = free = -1
        lo = off + SLOT_HEAD.size
        pair = KEY_LEN + VAL_LEN
        for i in range(MAX_LABELS):
            p = lo + i * pair
            cur = mm[p:p + KEY_LEN].split(b"\0", 1)[0]
            if not cur:
                if free < 0:
                    free = i
            elif cur == kb:
                found = i
                break
        if vb:
            at = found if found >= 0 else free
            if at >= 0:
                p = lo + at * pair
                mm[p:p + KEY_LEN] = kb.ljust(KEY_LEN, b"\0")
                mm[p + KEY_LEN:p + pair] = vb.ljust(VAL_LEN, b"\0")
                if found < 0:
                    count += 1
        elif found >= 0:
            p = lo + found * pair
            mm[p:p + pair] = b"\0" * pair
            count = max(0, count - 1)
        struct.pack_into("<I", mm, off + 8, count)
    finally:
        struct.pack_into("<I", mm, off + 4, (seq | 1) + 1)  # even again


def labels_clear() -> None:
    """Remove every label on the calling thread."""
    mm = _table()
    if mm is None:
        return
    off = _my_slot(mm)
    if off is None:
        return
    _, seq, _, _ = SLOT_HEAD.unpack_from(mm, off)
    struct.pack_into("<I", mm, off + 4, seq | 1)
    try:
        lo = off + SLOT_HEAD.size
        mm[lo:lo + MAX_LABELS * (KEY_LEN + VAL_LEN)] = \
            b"\0" * (MAX_LABELS * (KEY_LEN + VAL_LEN))
        struct.pack_into("<I", mm, off + 8, 0)
    finally:
        struct.pack_into("<I", mm, off + 4, (seq | 1) + 1)
