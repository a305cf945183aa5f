"""Probabilistic profiling: fleet-wide load shedding.

Reference semantics (fork StartProbabilisticProfiling, main.go:541-548;
flags.go:324-325): with threshold T in [1,100], each agent decides once
per interval (default 1 m) whether to profile during that interval, with
probability T/100. The decision is a hash of (machine identity, interval
number) so it is stable within an interval, uncorrelated across
intervals, and needs no coordination — across a large fleet, T% of
agents are profiling at any moment.
"""

from __future__ import annotations

import hashlib
import logging
import threading
import time
from typing import Callable, Optional

log = logging.getLogger("parca_agent_amd.probabilistic")

THRESHOLD_MAX = 100


def should_profile(machine_id: str, interval_index: int,
                   threshold: int) -> bool:
    if threshold >= THRESHOLD_MAX:
        return True
    if threshold <= 0:
        return False
    h = hashlib.blake2b(
        f"{machine_id}:{interval_index}".encode(), digest_size=8)
    return int.from_bytes(h.digest(), "big") % THRESHOLD_MAX < threshold


class ProbabilisticController:
    """Starts/stops a profiling service at interval boundaries."""

    def __init__(self, service, threshold: int, interval_seconds: float,
                 machine_id: Optional[str] = None,
                 clock: Callable[[], float] = time.time) -> None:
        self.service = service
        self.threshold = threshold
        self.interval = interval_seconds
        self.clock = clock
        if machine_id is None:
            from .analytics.sender import _machine_id

            machine_id = _machine_id()
        self.machine_id = machine_id
        self._running_service = False
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def _apply(self) -> None:
        idx = int(self.clock() // self.interval)
        want = should_profile(self.machine_id, idx, self.threshold)
        if want and not self._running_service:
            self.service.start()
            self._running_service = True
            log.info("probabilistic profiling: ON for interval %d", idx)
        elif not want and self._running_service:
            self.service.stop()
            self._running_service = False
            log.info("probabilistic profiling: OFF for interval %d", idx)

    def start(self) -> None:
        self._apply()
        self._stop.clear()
        self._thread = threading.Thread(target=self._run,
                                        name="probabilistic", daemon=True)
        self._thread.start()

    def _run(self) -> None:
        while True:
            now = self.clock()
            next_boundary = (int(now // self.interval) + 1) * self.interval
            if self._stop.wait(max(next_boundary - now, 0.05)):
                return
            try:
                self._apply()
            except Exception:
                log.error("probabilistic switch failed", exc_info=True)

    def stop(self) -> None:
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=5)
            self._thread = None
        if self._running_service:
            self.service.stop()
            self._running_service = False
