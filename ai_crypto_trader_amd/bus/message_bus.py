"""Message bus — the control-plane transport.

The reference's services communicate exclusively through Redis pub/sub
channels and keys (docker-compose.yml:405-407). This bus
preserves that contract — same channel names, same payload shapes, same
key/hash semantics — behind an interface with two backends:

  InProcessBus  asyncio queues; the default in this repo (single-process
                run_trader.py topology; no external broker needed)
  RedisBus      drop-in when a Redis server + redis-py are present, wire-
                compatible with the reference's services

The data plane (GA fitness, MC stats, gradients) does NOT go through this
bus: it moves over RCCL/xGMI collectives (parallel/dist.py).
"""

from __future__ import annotations

import asyncio
import fnmatch
import json
import time
from collections import defaultdict
from typing import Any, AsyncIterator


class Subscription:
    def __init__(self, bus: "InProcessBus", channels: tuple[str, ...]):
        self.bus = bus
        self.channels = channels
        self.queue: asyncio.Queue = asyncio.Queue(maxsize=10_000)

    async def __aiter__(self) -> AsyncIterator[tuple[str, dict]]:
        while True:
            yield await self.queue.get()

    async def get(self, timeout: float | None = None):
        if timeout is None:
            return await self.queue.get()
        return await asyncio.wait_for(self.queue.get(), timeout)

    async def get_batch(self, max_items: int = 512) -> list:
        """Block for the first message, then drain what is immediately
        available — one loop wake-up per burst instead of per message
        (per-message wait_for starves consumers under a saturated loop)."""
        out = [await self.queue.get()]
        while len(out) < max_items and not self.queue.empty():
            out.append(self.queue.get_nowait())
        return out

    def close(self):
        self.bus._unsubscribe(self)


class InProcessBus:
    """Redis-semantics bus: pub/sub + string keys + hashes, in one process."""

    def __init__(self):
        self._subs: list[Subscription] = []
        self._kv: dict[str, str] = {}
        self._hashes: dict[str, dict[str, str]] = defaultdict(dict)
        self._expiry: dict[str, float] = {}
        self.published_counts: dict[str, int] = defaultdict(int)

    # --- pub/sub ---------------------------------------------------------
    async def publish(self, channel: str, payload: dict | str) -> int:
        if isinstance(payload, (dict, list)):
            payload = json.dumps(payload)
        n = 0
        self.published_counts[channel] += 1
        for sub in list(self._subs):
            if any(fnmatch.fnmatch(channel, pat) for pat in sub.channels):
                try:
                    data = json.loads(payload)
                except (json.JSONDecodeError, TypeError):
                    data = payload
                try:
                    sub.queue.put_nowait((channel, data))
                    n += 1
                except asyncio.QueueFull:
                    pass          # slow consumer: drop (Redis would buffer)
        return n

    def subscribe(self, *channels: str) -> Subscription:
        sub = Subscription(self, channels)
        self._subs.append(sub)
        return sub

    def _unsubscribe(self, sub: Subscription):
        if sub in self._subs:
            self._subs.remove(sub)

    # --- key/value (Redis string keys) -----------------------------------
    def _check_expired(self, key: str):
        exp = self._expiry.get(key)
        if exp is not None and time.time() > exp:
            self._kv.pop(key, None)
            self._expiry.pop(key, None)

    async def set(self, key: str, value: Any, ex: float | None = None):
        if isinstance(value, (dict, list)):
            value = json.dumps(value)
        self._kv[key] = str(value)
        if ex is not None:
            self._expiry[key] = time.time() + ex

    async def get(self, key: str) -> str | None:
        self._check_expired(key)
        return self._kv.get(key)

    async def get_json(self, key: str):
        v = await self.get(key)
        return None if v is None else json.loads(v)

    async def delete(self, *keys: str):
        for k in keys:
            self._kv.pop(k, None)
            self._hashes.pop(k, None)

    async def keys(self, pattern: str = "*") -> list[str]:
        names = list(self._kv) + list(self._hashes)
        return [k for k in names if fnmatch.fnmatch(k, pattern)]

    # --- hashes ----------------------------------------------------------
    async def hset(self, key: str, field: str, value: Any):
        if isinstance(value, (dict, list)):
            value = json.dumps(value)
        self._hashes[key][field] = str(value)

    async def hget(self, key: str, field: str) -> str | None:
        return self._hashes.get(key, {}).get(field)

    async def hgetall(self, key: str) -> dict[str, str]:
        return dict(self._hashes.get(key, {}))

    async def ping(self) -> bool:
        return True

    async def close(self):
        self._subs.clear()


class RedisSubscription:
    """Subscription adapter over redis pubsub with the Subscription
    interface (get / get_batch / aiter / close). A lazy reader task pumps
    the wire into a local queue so get_batch() can drain bursts exactly
    like the in-process bus."""

    def __init__(self, r, channels: tuple[str, ...]):
        self.r = r
        self.channels = channels
        self.queue: asyncio.Queue = asyncio.Queue(maxsize=10_000)
        self._pubsub = None
        self._task: asyncio.Task | None = None

    async def _ensure(self):
        if self._task is None:
            self._pubsub = self.r.pubsub()
            pats = [c for c in self.channels if any(x in c for x in "*?[")]
            plain = [c for c in self.channels if c not in pats]
            if plain:
                await self._pubsub.subscribe(*plain)
            if pats:
                await self._pubsub.psubscribe(*pats)
            self._task = asyncio.create_task(self._pump())

    async def _pump(self):
        async for msg in self._pubsub.listen():
            if msg["type"] not in ("message", "pmessage"):
                continue
            data = msg["data"]
            try:
                data = json.loads(data)
            except (json.JSONDecodeError, TypeError):
                pass
            try:
                self.queue.put_nowait((msg["channel"], data))
            except asyncio.QueueFull:
                pass                      # slow consumer: drop

    async def __aiter__(self) -> AsyncIterator[tuple[str, Any]]:
        await self._ensure()
        while True:
            yield await self.queue.get()

    async def get(self, timeout: float | None = None):
        await self._ensure()
        if timeout is None:
            return await self.queue.get()
        return await asyncio.wait_for(self.queue.get(), timeout)

    async def get_batch(self, max_items: int = 512) -> list:
        await self._ensure()
        out = [await self.queue.get()]
        while len(out) < max_items and not self.queue.empty():
            out.append(self.queue.get_nowait())
        return out

    def close(self):
        if self._task is not None:
            self._task.cancel()
            self._task = None


class RedisBus:
    """Thin async-redis adapter with the same interface (requires a redis
    server; wire-compatible with the reference's channel schema)."""

    def __init__(self, host="localhost", port=6379):
        import redis.asyncio as aioredis   # gated: not in the base image

        self.r = aioredis.Redis(host=host, port=port, decode_responses=True)

    async def publish(self, channel, payload):
        if isinstance(payload, (dict, list)):
            payload = json.dumps(payload)
        return await self.r.publish(channel, payload)

    def subscribe(self, *channels):
        """Same interface as InProcessBus.subscribe: returns an object
        with get/get_batch/close backed by a redis pubsub reader task
        (glob patterns map to PSUBSCRIBE)."""
        return RedisSubscription(self.r, channels)

    async def set(self, key, value, ex=None):
        if isinstance(value, (dict, list)):
            value = json.dumps(value)
        await self.r.set(key, value, ex=ex)

    async def get(self, key):
        return await self.r.get(key)

    async def get_json(self, key):
        v = await self.get(key)
        return None if v is None else json.loads(v)

    async def hset(self, key, field, value):
        if isinstance(value, (dict, list)):
            value = json.dumps(value)
        await self.r.hset(key, field, value)

    async def hget(self, key, field):
        return await self.r.hget(key, field)

    async def hgetall(self, key):
        return await self.r.hgetall(key)

    async def ping(self):
        return await self.r.ping()

    async def close(self):
        await self.r.aclose()


def make_bus(backend: str = "inprocess", **kw):
    if backend == "redis":
        return RedisBus(**kw)
    return InProcessBus()
