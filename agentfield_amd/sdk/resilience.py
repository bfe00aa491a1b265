"""Client-side resilience: rate limiter with circuit breaker + TTL/LRU
result cache (reference parity: rate_limiter.py P18, result_cache.py P19)."""
from __future__ import annotations

import hashlib
import json
import random
import threading
import time
from collections import OrderedDict


class CircuitOpenError(RuntimeError):
    pass


class StatelessRateLimiter:
    """Exponential backoff with jitter on overload signals (429-style) and a
    circuit breaker on consecutive failures."""

    def __init__(self, base_delay: float = 0.5, max_delay: float = 30.0,
                 breaker_threshold: int = 5, breaker_reset: float = 30.0,
                 jitter: float = 0.25, seed: int | None = None):
        self.base_delay = base_delay
        self.max_delay = max_delay
        self.breaker_threshold = breaker_threshold
        self.breaker_reset = breaker_reset
        self.jitter = jitter
        self._rng = random.Random(seed)
        self._consecutive = 0
        self._opened_at: float | None = None
        self._lock = threading.Lock()

    @property
    def is_open(self) -> bool:
        with self._lock:
            if self._opened_at is None:
                return False
            if time.time() - self._opened_at >= self.breaker_reset:
                self._opened_at = None  # half-open: allow a probe
                self._consecutive = self.breaker_threshold - 1
                return False
            return True

    def delay_for(self, attempt: int) -> float:
        d = min(self.max_delay, self.base_delay * (2 ** attempt))
        return d * (1.0 + self._rng.uniform(-self.jitter, self.jitter))

    def record_success(self):
        with self._lock:
            self._consecutive = 0
            self._opened_at = None

    def record_failure(self):
        with self._lock:
            self._consecutive += 1
            if self._consecutive >= self.breaker_threshold:
                self._opened_at = time.time()

    def call(self, fn, *args, retries: int = 3,
             retryable=(Exception,), **kwargs):
        if self.is_open:
            raise CircuitOpenError("circuit breaker open")
        last = None
        for attempt in range(retries + 1):
            try:
                out = fn(*args, **kwargs)
                self.record_success()
                return out
            except retryable as e:
                last = e
                self.record_failure()
                if self.is_open or attempt == retries:
                    raise
                time.sleep(self.delay_for(attempt))
        raise last


class ResultCache:
    """TTL + LRU cache for execution/ai results (P19)."""

    def __init__(self, max_entries: int = 1024, ttl: float = 300.0):
        self.max_entries = max_entries
        self.ttl = ttl
        self._data: OrderedDict[str, tuple[float, object]] = OrderedDict()
        self._lock = threading.Lock()
        self.hits = 0
        self.misses = 0

    @staticmethod
    def key_for(*parts) -> str:
        raw = json.dumps(parts, sort_keys=True, default=str)
        return hashlib.sha256(raw.encode()).hexdigest()

    def get(self, key: str):
        with self._lock:
            ent = self._data.get(key)
            if ent is None:
                self.misses += 1
                return None
            ts, value = ent
            if time.time() - ts > self.ttl:
                del self._data[key]
                self.misses += 1
                return None
            self._data.move_to_end(key)
            self.hits += 1
            return value

    def put(self, key: str, value) -> None:
        with self._lock:
            self._data[key] = (time.time(), value)
            self._data.move_to_end(key)
            while len(self._data) > self.max_entries:
                self._data.popitem(last=False)

    def purge_expired(self) -> int:
        now = time.time()
        with self._lock:
            dead = [k for k, (ts, _) in self._data.items()
                    if now - ts > self.ttl]
            for k in dead:
                del self._data[k]
            return len(dead)

    def stats(self) -> dict:
        return {"entries": len(self._data), "hits": self.hits,
                "misses": self.misses}
