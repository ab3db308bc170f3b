"""Circuit breaker + retry-with-backoff (reference parity:
services/utils/circuit_breaker.py:31-331 — CLOSED/OPEN/HALF_OPEN state
machine, sync+async call paths, global registry, decorators)."""

from __future__ import annotations

import asyncio
import functools
import random
import time
from enum import Enum


class State(str, Enum):
    CLOSED = "closed"
    OPEN = "open"
    HALF_OPEN = "half_open"


class CircuitOpenError(RuntimeError):
    pass


class CircuitBreaker:
    def __init__(self, name: str, failure_threshold: int = 5,
                 recovery_timeout: float = 30.0, half_open_max_calls: int = 2):
        self.name = name
        self.failure_threshold = failure_threshold
        self.recovery_timeout = recovery_timeout
        self.half_open_max_calls = half_open_max_calls
        self.state = State.CLOSED
        self.failures = 0
        self.successes = 0
        self.last_failure_time = 0.0
        self._half_open_calls = 0
        self.total_calls = 0
        self.total_failures = 0

    def _check_transition(self):
        if (self.state == State.OPEN
                and time.monotonic() - self.last_failure_time
                >= self.recovery_timeout):
            self.state = State.HALF_OPEN
            self._half_open_calls = 0

    def _on_success(self):
        self.successes += 1
        if self.state == State.HALF_OPEN:
            self.state = State.CLOSED
        self.failures = 0

    def _on_failure(self):
        self.failures += 1
        self.total_failures += 1
        self.last_failure_time = time.monotonic()
        if self.state == State.HALF_OPEN or \
                self.failures >= self.failure_threshold:
            self.state = State.OPEN

    def _admit(self):
        self._check_transition()
        self.total_calls += 1
        if self.state == State.OPEN:
            raise CircuitOpenError(f"circuit '{self.name}' is open")
        if self.state == State.HALF_OPEN:
            if self._half_open_calls >= self.half_open_max_calls:
                raise CircuitOpenError(
                    f"circuit '{self.name}' half-open limit reached")
            self._half_open_calls += 1

    def call(self, fn, *args, **kwargs):
        self._admit()
        try:
            out = fn(*args, **kwargs)
        except Exception:
            self._on_failure()
            raise
        self._on_success()
        return out

    async def call_async(self, fn, *args, **kwargs):
        self._admit()
        try:
            out = await fn(*args, **kwargs)
        except Exception:
            self._on_failure()
            raise
        self._on_success()
        return out

    def reset(self):
        self.state = State.CLOSED
        self.failures = 0
        self._half_open_calls = 0

    def status(self) -> dict:
        return {
            "name": self.name, "state": self.state.value,
            "failures": self.failures, "total_calls": self.total_calls,
            "total_failures": self.total_failures,
        }


_registry: dict[str, CircuitBreaker] = {}


def get_breaker(name: str, **kw) -> CircuitBreaker:
    if name not in _registry:
        _registry[name] = CircuitBreaker(name, **kw)
    return _registry[name]


def all_breakers() -> dict[str, CircuitBreaker]:
    return dict(_registry)


def circuit_breaker(name: str, **kw):
    """Decorator (sync or async), reference circuit_breaker.py:297-331."""

    def deco(fn):
        br = get_breaker(name, **kw)
        if asyncio.iscoroutinefunction(fn):
            @functools.wraps(fn)
            async def aw(*a, **k):
                return await br.call_async(fn, *a, **k)
            aw.breaker = br
            return aw

        @functools.wraps(fn)
        def w(*a, **k):
            return br.call(fn, *a, **k)
        w.breaker = br
        return w

    return deco


def retry_with_backoff(max_retries: int = 3, base_delay: float = 0.5,
                       max_delay: float = 30.0, jitter: float = 0.1,
                       exceptions=(Exception,)):
    """Exponential backoff + jitter (reference circuit_breaker.py:227-279)."""

    def deco(fn):
        if asyncio.iscoroutinefunction(fn):
            @functools.wraps(fn)
            async def aw(*a, **k):
                delay = base_delay
                for attempt in range(max_retries + 1):
                    try:
                        return await fn(*a, **k)
                    except exceptions:
                        if attempt == max_retries:
                            raise
                        await asyncio.sleep(
                            min(delay, max_delay)
                            * (1 + jitter * random.random()))
                        delay *= 2
            return aw

        @functools.wraps(fn)
        def w(*a, **k):
            delay = base_delay
            for attempt in range(max_retries + 1):
                try:
                    return fn(*a, **k)
                except exceptions:
                    if attempt == max_retries:
                        raise
                    time.sleep(min(delay, max_delay)
                               * (1 + jitter * random.random()))
                    delay *= 2
        return w

    return deco
