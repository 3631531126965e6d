"""Keyed serialization of filter/bind (reference pkg/scheduler/serial:
hashed key-rwmutex with min-lock-duration metrics)."""
from __future__ import annotations

import threading
import time


class KeyedLocker:
    def __init__(self, buckets: int = 64):
        self._locks = [threading.Lock() for _ in range(buckets)]
        self._buckets = buckets
        self.wait_seconds: float = 0.0

    def _lock_for(self, key: str) -> threading.Lock:
        return self._locks[hash(key) % self._buckets]

    def acquire(self, key: str) -> None:
        t0 = time.monotonic()
        self._lock_for(key).acquire()
        self.wait_seconds += time.monotonic() - t0

    def release(self, key: str) -> None:
        self._lock_for(key).release()
