"""Debuginfo symbol uploader.

The reference runs a 25-worker pool over a 4096-deep queue with a
retry-LRU and an in-progress map, speaking the 4-RPC protocol
ShouldInitiateUpload -> InitiateUpload -> (signed-URL PUT | gRPC chunk
stream) -> MarkUploadFinished, optionally stripping to debug-only first
(reference: reporter/parca_uploader.go:109-406). Same behaviour here;
GPU code objects upload through the same path with their bytes already
in memory (the cubin-upload analog, parcagpu.go:262).
"""

from __future__ import annotations

import io
import logging
import os
import queue
import threading
from dataclasses import dataclass
from typing import Optional

from ..lru import LRU
from . import protos
from .elfwriter import only_keep_debug

log = logging.getLogger("parca_agent_amd.uploader")


@dataclass
class UploadItem:
    build_id: str          # server-side identity (GNU build-id or FileID)
    hash: str              # content hash advertised to the server
    path: str = ""         # file-backed executables
    data: Optional[bytes] = None  # in-memory (GPU code objects)
    type: int = 0


class DebuginfoUploader:
    def __init__(self, client, max_parallel: int = 25,
                 queue_size: int = 4096, strip: bool = True,
                 compress: bool = False,
                 temp_dir: str = "/tmp",
                 debug_directories: Optional[list] = None,
                 retry_cache_size: int = 8192,
                 retry_cache_ttl: float = 600.0) -> None:
        self.client = client
        self.strip = strip
        self.compress = compress
        self.temp_dir = temp_dir
        # Ordered external-debuginfo search roots (reference
        # --debuginfo-directories, default /usr/lib/debug): distro
        # debug packages install split DWARF under
        # <root>/.build-id/ab/cdef...debug.
        self.debug_directories = list(debug_directories
                                      if debug_directories is not None
                                      else ["/usr/lib/debug"])
        self._queue: "queue.Queue[Optional[UploadItem]]" = \
            queue.Queue(maxsize=queue_size)
        self._workers = [
            threading.Thread(target=self._worker, daemon=True,
                             name=f"debuginfo-upload-{i}")
            for i in range(max_parallel)
        ]
        # build_id -> True once attempted recently (retry suppression LRU,
        # parca_uploader.go:109-157).
        self._attempted: LRU[str, bool] = LRU(retry_cache_size,
                                              ttl_seconds=retry_cache_ttl)
        self._in_progress: set = set()
        self._mu = threading.Lock()
        self._started = False
        self.uploaded = 0
        self.skipped = 0
        self.errors = 0

    def start(self) -> None:
        if self._started:
            return
        self._started = True
        for w in self._workers:
            w.start()

    def stop(self) -> None:
        if not self._started:
            return
        for _ in self._workers:
            self._queue.put(None)
        for w in self._workers:
            w.join(timeout=10)
        self._started = False

    def enqueue(self, item: UploadItem) -> bool:
        """Non-blocking; drops when the queue is full (accepted-loss
        semantics like the reference)."""
        with self._mu:
            if item.build_id in self._in_progress or \
                    self._attempted.get(item.build_id):
                self.skipped += 1
                return False
            self._in_progress.add(item.build_id)
        try:
            self._queue.put_nowait(item)
            return True
        except queue.Full:
            with self._mu:
                self._in_progress.discard(item.build_id)
            self.skipped += 1
            return False

    # -- worker ------------------------------------------------------------

    def _worker(self) -> None:
        while True:
            item = self._queue.get()
            if item is None:
                return
            try:
                self._attempt(item)
            except Exception:
                self.errors += 1
                log.warning("debuginfo upload failed for %s",
                            item.build_id, exc_info=True)
            finally:
                with self._mu:
                    self._in_progress.discard(item.build_id)
                    self._attempted.put(item.build_id, True)

    def _attempt(self, item: UploadItem) -> None:
        should, reason = self.client.should_initiate_upload(
            item.build_id, item.hash, type_=item.type)
        if not should:
            self.skipped += 1
            return

        data = item.data
        if data is None and item.path:
            external = self._find_external_debug(item.build_id)
            if external is not None:
                try:
                    with open(external, "rb") as fh:
                        data = fh.read()
                except OSError:
                    data = None
            if data is None:
                data = self._prepare_file(item.path)
        if not data:
            return

        ins = self.client.initiate_upload(item.build_id, item.hash,
                                          len(data), type_=item.type)
        if ins.upload_strategy == protos.UPLOAD_STRATEGY_SIGNED_URL and \
                ins.signed_url:
            self._signed_url_put(ins.signed_url, data)
        else:
            self.client.upload(ins.upload_id, data, type_=item.type)
        self.client.mark_upload_finished(item.build_id, ins.upload_id,
                                         type_=item.type)
        self.uploaded += 1

    def _find_external_debug(self, build_id: str) -> Optional[str]:
        """GNU build-id debug-file convention:
        <dir>/.build-id/<first2>/<rest>.debug. Split debuginfo carries
        the full DWARF the stripped binary lacks, so it wins over
        re-stripping the binary itself."""
        if not build_id or len(build_id) < 4 or not all(
                c in "0123456789abcdef" for c in build_id.lower()):
            return None
        b = build_id.lower()
        for root in self.debug_directories:
            cand = os.path.join(root, ".build-id", b[:2], b[2:] + ".debug")
            if os.path.exists(cand):
                return cand
        return None

    def _prepare_file(self, path: str) -> Optional[bytes]:
        try:
            if self.strip:
                buf = io.BytesIO()
                only_keep_debug(path, buf, compress=self.compress)
                return buf.getvalue()
            with open(path, "rb") as fh:
                return fh.read()
        except (OSError, ValueError, AssertionError) as e:
            log.debug("debuginfo extraction failed for %s: %s", path, e)
            try:
                with open(path, "rb") as fh:
                    return fh.read()
            except OSError:
                return None

    @staticmethod
    def _signed_url_put(url: str, data: bytes) -> None:
        import urllib.request

        req = urllib.request.Request(url, data=data, method="PUT")
        req.add_header("Content-Type", "application/octet-stream")
        with urllib.request.urlopen(req, timeout=120) as resp:
            if resp.status not in (200, 201, 204):
                raise RuntimeError(f"signed-url PUT failed: {resp.status}")
