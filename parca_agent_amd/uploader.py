"""Offline-mode replay: upload recorded .padata(.zst) logs to a server.

Reference: uploader/log_uploader.go — header check, per-batch upload,
file deletion after success. The v2 log frames are self-contained Arrow
records, so no stacktrace-resolution round trip is needed (the v1
handshake of log_uploader.go:510-654 is obsolete against WriteArrow).
"""

from __future__ import annotations

import glob
import logging
import os

log = logging.getLogger("parca_agent_amd.offline_upload")


def offline_upload(flags) -> int:
    from .reporter.destinations import read_offline_log
    from .reporter.grpc_client import ParcaClient, build_channel

    if not flags.remote_store.address:
        # The validate() exclusivity rule blocks address+storage_path
        # together; replay mode relaxes it via direct invocation.
        print("offline upload requires --remote-store-address")
        return 1
    client = ParcaClient(build_channel(flags))
    pattern_dir = flags.offline_mode.storage_path
    files = sorted(
        glob.glob(os.path.join(pattern_dir, "*.padata")) +
        glob.glob(os.path.join(pattern_dir, "*.padata.zst")))
    if not files:
        log.info("no offline logs under %s", pattern_dir)
        return 0
    failed = 0
    for path in files:
        try:
            payloads = read_offline_log(path)
        except (ValueError, OSError) as e:
            log.warning("skipping corrupt log %s: %s", path, e)
            failed += 1
            continue
        ok = True
        for payload in payloads:
            try:
                client.write_arrow(payload)
            except Exception:
                log.error("upload failed for %s", path, exc_info=True)
                ok = False
                failed += 1
                break
        if ok:
            os.unlink(path)
            log.info("uploaded and removed %s (%d batches)",
                     path, len(payloads))
    return 0 if failed == 0 else 1
