"""LRU caches with optional TTL.

The reference keeps per-PID label sets and per-trace-hash stacks in
freelru caches with lifetimes (reference: reporter/parca_reporter.go:326-331,
762-798). This is the equivalent, tuned for the agent's access pattern:
hot lookups on the sample hot path, eviction callbacks used for metrics.
"""

from __future__ import annotations

import time
from collections import OrderedDict
from typing import Callable, Generic, Hashable, Optional, Tuple, TypeVar

K = TypeVar("K", bound=Hashable)
V = TypeVar("V")


class LRU(Generic[K, V]):
    def __init__(
        self,
        max_size: int,
        ttl_seconds: Optional[float] = None,
        on_evict: Optional[Callable[[K, V], None]] = None,
        clock: Callable[[], float] = time.monotonic,
    ) -> None:
        if max_size <= 0:
            raise ValueError("max_size must be positive")
        self.max_size = max_size
        self.ttl = ttl_seconds
        self.on_evict = on_evict
        self._clock = clock
        self._data: "OrderedDict[K, Tuple[float, V]]" = OrderedDict()
        self.hits = 0
        self.misses = 0
        self.evictions = 0

    def __len__(self) -> int:
        return len(self._data)

    def __contains__(self, key: K) -> bool:
        return self.get(key) is not None

    def get(self, key: K, default: Optional[V] = None) -> Optional[V]:
        entry = self._data.get(key)
        if entry is None:
            self.misses += 1
            return default
        ts, value = entry
        if self.ttl is not None and self._clock() - ts > self.ttl:
            del self._data[key]
            self.misses += 1
            if self.on_evict:
                self.on_evict(key, value)
            return default
        self._data.move_to_end(key)
        self.hits += 1
        return value

    def put(self, key: K, value: V) -> None:
        if key in self._data:
            del self._data[key]
        self._data[key] = (self._clock(), value)
        while len(self._data) > self.max_size:
            old_key, (_, old_val) = self._data.popitem(last=False)
            self.evictions += 1
            if self.on_evict:
                self.on_evict(old_key, old_val)

    def remove(self, key: K) -> bool:
        if key in self._data:
            del self._data[key]
            return True
        return False

    def purge(self) -> None:
        self._data.clear()

    def keys(self):
        return list(self._data.keys())
