"""Workqueue + client rate limiting, mirroring the reference's behavior:

- ``TokenBucket`` — client-go flowcontrol-style token bucket used both for
  the --kube-api-qps/--kube-api-burst client throttle (reference
  options.go:87-88) and as the overall-queue half of the MaxOf limiter
  (--controller-queue-rate-limit/-burst, options.go:90-91).
- ``ItemExponentialFailureRateLimiter`` — per-key exponential backoff
  5 ms → 1000 s (reference mpi_job_controller.go:121-124).
- ``MaxOfRateLimiter`` — workqueue.NewMaxOfRateLimiter semantics: the
  requeue delay is the max over the member limiters.
"""
from __future__ import annotations

import threading
import time


class TokenBucket:
    """Blocking token bucket: ``wait()`` sleeps until a token is available.
    ``delay()`` returns the non-blocking wait a new item would incur."""

    def __init__(self, qps: float, burst: int):
        self.qps = max(qps, 1e-9)
        self.burst = max(burst, 1)
        self.tokens = float(self.burst)
        self.last = time.monotonic()
        self._lock = threading.Lock()

    def _refill(self):
        now = time.monotonic()
        self.tokens = min(self.burst, self.tokens + (now - self.last) * self.qps)
        self.last = now

    def delay(self) -> float:
        """Reserve one token; return seconds to wait before acting on it."""
        with self._lock:
            self._refill()
            self.tokens -= 1.0
            if self.tokens >= 0:
                return 0.0
            return -self.tokens / self.qps

    def wait(self):
        d = self.delay()
        if d > 0:
            time.sleep(d)


class ItemExponentialFailureRateLimiter:
    def __init__(self, base_delay: float = 0.005, max_delay: float = 1000.0):
        self.base = base_delay
        self.max = max_delay
        self._fails: dict = {}
        self._lock = threading.Lock()

    def when(self, item) -> float:
        with self._lock:
            n = self._fails.get(item, 0)
            self._fails[item] = n + 1
        return min(self.base * (2 ** n), self.max)

    def forget(self, item):
        with self._lock:
            self._fails.pop(item, None)

    def retries(self, item) -> int:
        with self._lock:
            return self._fails.get(item, 0)


class MaxOfRateLimiter:
    """workqueue.NewMaxOfRateLimiter(exponential, bucket): delay is the max
    over members (reference mpi_job_controller.go:121-124)."""

    def __init__(self, *limiters):
        self.limiters = limiters

    def when(self, item) -> float:
        out = 0.0
        for l in self.limiters:
            if isinstance(l, TokenBucket):
                out = max(out, l.delay())
            else:
                out = max(out, l.when(item))
        return out

    def forget(self, item):
        for l in self.limiters:
            if hasattr(l, "forget"):
                l.forget(item)


def default_controller_limiter(rate: float = 10.0, burst: int = 100) -> MaxOfRateLimiter:
    """The reference's controller-queue limiter: MaxOf{exponential 5ms→1000s,
    bucket(rate, burst)} with flag defaults 10/100 (options.go:90-91)."""
    return MaxOfRateLimiter(ItemExponentialFailureRateLimiter(),
                            TokenBucket(rate, burst))
