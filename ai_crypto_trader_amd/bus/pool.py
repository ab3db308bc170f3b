"""Bus connection pooling (reference parity: services/utils/redis_pool.py
:18-406 — pooled async Redis, max connections, health checks, op
wrappers; the reference shipped it but its services never used it.
Here the pool IS the standard way services get a bus handle).

For the in-process backend the "pool" is a shared singleton (one bus per
process by construction); for the redis backend it manages a bounded set
of RedisBus connections with ping health checks."""

from __future__ import annotations

import asyncio
import time

from .message_bus import InProcessBus, RedisBus, make_bus


class BusPoolManager:
    def __init__(self, backend: str = "inprocess", max_connections: int = 20,
                 health_check_interval: float = 30.0, **conn_kw):
        self.backend = backend
        self.max_connections = max_connections
        self.health_check_interval = health_check_interval
        self.conn_kw = conn_kw
        self._shared: InProcessBus | None = None
        self._pool: list[RedisBus] = []
        self._in_use: set[int] = set()
        self._last_health: dict[int, float] = {}
        self._lock = asyncio.Lock()
        self.stats = {"acquired": 0, "released": 0, "health_checks": 0,
                      "failures": 0}

    async def acquire(self):
        self.stats["acquired"] += 1
        if self.backend == "inprocess":
            if self._shared is None:
                self._shared = InProcessBus()
            return self._shared
        async with self._lock:
            for i, bus in enumerate(self._pool):
                if i not in self._in_use:
                    if await self._healthy(i, bus):
                        self._in_use.add(i)
                        return bus
            if len(self._pool) >= self.max_connections:
                raise RuntimeError("bus pool exhausted")
            bus = make_bus(self.backend, **self.conn_kw)
            self._pool.append(bus)
            self._in_use.add(len(self._pool) - 1)
            return bus

    async def release(self, bus):
        self.stats["released"] += 1
        if self.backend == "inprocess":
            return
        async with self._lock:
            for i, b in enumerate(self._pool):
                if b is bus:
                    self._in_use.discard(i)
                    return

    async def _healthy(self, i: int, bus) -> bool:
        now = time.monotonic()
        if now - self._last_health.get(i, 0.0) < self.health_check_interval:
            return True
        self.stats["health_checks"] += 1
        try:
            ok = await bus.ping()
        except Exception:
            ok = False
        if not ok:
            self.stats["failures"] += 1
            return False
        self._last_health[i] = now
        return True

    async def close(self):
        if self._shared is not None:
            await self._shared.close()
        for b in self._pool:
            try:
                await b.close()
            except Exception:
                pass
        self._pool.clear()
        self._in_use.clear()


_default_pool: BusPoolManager | None = None


def get_pool(backend: str = "inprocess", **kw) -> BusPoolManager:
    global _default_pool
    if _default_pool is None:
        _default_pool = BusPoolManager(backend, **kw)
    return _default_pool
