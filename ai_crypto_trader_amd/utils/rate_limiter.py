"""Rate limiters (reference parity: services/utils/rate_limiter.py:140-352 —
sliding window / fixed window / token bucket / leaky bucket; in-process
state instead of Redis; defaults 10k market-data, 1k API, 50 AI req/min
per the reference CHANGELOG Phase 1C)."""

from __future__ import annotations

import asyncio
import functools
import time
from collections import deque


class RateLimitExceeded(RuntimeError):
    pass


class SlidingWindowLimiter:
    def __init__(self, limit: int, window_s: float = 60.0):
        self.limit = limit
        self.window_s = window_s
        self.events: deque[float] = deque()

    def allow(self) -> bool:
        now = time.monotonic()
        cutoff = now - self.window_s
        while self.events and self.events[0] < cutoff:
            self.events.popleft()
        if len(self.events) >= self.limit:
            return False
        self.events.append(now)
        return True

    def remaining(self) -> int:
        now = time.monotonic()
        cutoff = now - self.window_s
        while self.events and self.events[0] < cutoff:
            self.events.popleft()
        return max(self.limit - len(self.events), 0)


class FixedWindowLimiter:
    def __init__(self, limit: int, window_s: float = 60.0):
        self.limit = limit
        self.window_s = window_s
        self.window_start = 0.0
        self.count = 0

    def allow(self) -> bool:
        now = time.monotonic()
        if now - self.window_start >= self.window_s:
            self.window_start = now
            self.count = 0
        if self.count >= self.limit:
            return False
        self.count += 1
        return True


class TokenBucketLimiter:
    def __init__(self, rate_per_s: float, burst: int):
        self.rate = rate_per_s
        self.burst = burst
        self.tokens = float(burst)
        self.last = time.monotonic()

    def allow(self, n: float = 1.0) -> bool:
        now = time.monotonic()
        self.tokens = min(self.burst, self.tokens + (now - self.last) * self.rate)
        self.last = now
        if self.tokens >= n:
            self.tokens -= n
            return True
        return False


class LeakyBucketLimiter:
    def __init__(self, rate_per_s: float, capacity: int):
        self.rate = rate_per_s
        self.capacity = capacity
        self.level = 0.0
        self.last = time.monotonic()

    def allow(self) -> bool:
        now = time.monotonic()
        self.level = max(0.0, self.level - (now - self.last) * self.rate)
        self.last = now
        if self.level + 1 > self.capacity:
            return False
        self.level += 1
        return True


DEFAULT_LIMITS = {           # reference CHANGELOG.md:63-66
    "market_data": 10_000,
    "api": 1_000,
    "ai": 50,
}

_limiters: dict[str, SlidingWindowLimiter] = {}


def get_limiter(name: str, limit: int | None = None,
                window_s: float = 60.0) -> SlidingWindowLimiter:
    if name not in _limiters:
        _limiters[name] = SlidingWindowLimiter(
            limit or DEFAULT_LIMITS.get(name, 1000), window_s)
    return _limiters[name]


def rate_limited(name: str, limit: int | None = None, wait: bool = True):
    """Decorator (reference rate_limiter.py:448): blocks (async sleeps)
    until allowed, or raises when wait=False."""

    def deco(fn):
        lim = get_limiter(name, limit)
        if asyncio.iscoroutinefunction(fn):
            @functools.wraps(fn)
            async def aw(*a, **k):
                while not lim.allow():
                    if not wait:
                        raise RateLimitExceeded(name)
                    await asyncio.sleep(0.05)
                return await fn(*a, **k)
            return aw

        @functools.wraps(fn)
        def w(*a, **k):
            if not lim.allow():
                if not wait:
                    raise RateLimitExceeded(name)
                while not lim.allow():
                    time.sleep(0.05)
            return fn(*a, **k)
        return w

    return deco
