"""Service base class.

Every reference service repeats the same scaffolding: redis connect/retry
loops, a health-check TCP server, rotating-file logging, a run loop
(e.g. market_monitor_service.py:49-145, :635-680). This base centralizes
it: bus wiring, logging, health state, graceful stop, task supervision.
"""

from __future__ import annotations

import asyncio
import logging
import time

from ..bus.message_bus import InProcessBus
from ..config import AppConfig, get_config
from ..utils.metrics import get_metrics


def setup_logger(name: str, log_dir: str = "logs") -> logging.Logger:
    logger = logging.getLogger(name)
    if not logger.handlers:
        logger.setLevel(logging.INFO)
        h = logging.StreamHandler()
        h.setFormatter(logging.Formatter(
            "%(asctime)s %(name)s %(levelname)s %(message)s"))
        logger.addHandler(h)
    return logger


class Service:
    name = "service"

    def __init__(self, bus: InProcessBus, config: AppConfig | None = None):
        self.bus = bus
        self.config = config or get_config()
        self.log = setup_logger(self.name)
        self.metrics = get_metrics(self.name)
        self.running = False
        self.healthy = True
        self.started_at = 0.0
        self._tasks: list[asyncio.Task] = []

    # --- lifecycle -------------------------------------------------------
    async def start(self):
        self.running = True
        self.started_at = time.time()
        self._tasks = [asyncio.create_task(coro, name=f"{self.name}:{i}")
                       for i, coro in enumerate(self.run_tasks())]
        for t in self._tasks:
            t.add_done_callback(self._task_done)
        self.log.info("started (%d tasks)", len(self._tasks))

    def _task_done(self, task: asyncio.Task):
        if task.cancelled():
            return
        exc = task.exception()
        if exc is not None:
            self.healthy = False
            self.log.error("task %s died: %r", task.get_name(), exc,
                           exc_info=exc)

    def run_tasks(self):
        """Override: yield coroutines to supervise."""
        return [self.run()]

    async def run(self):
        raise NotImplementedError

    async def stop(self):
        self.running = False
        for t in self._tasks:
            t.cancel()
        await asyncio.gather(*self._tasks, return_exceptions=True)
        self._tasks.clear()
        self.log.info("stopped")

    async def wait(self):
        if self._tasks:
            await asyncio.gather(*self._tasks, return_exceptions=True)

    def health(self) -> dict:
        return {
            "service": self.name,
            # healthy = no task died with an exception (a task that ran to
            # completion, e.g. a finished replay feed, is still healthy)
            "healthy": self.healthy,
            "uptime_s": time.time() - self.started_at if self.started_at
            else 0.0,
        }

    # --- helpers ---------------------------------------------------------
    async def consume(self, sub, handler):
        """Drive `handler(channel, msg)` (sync or async) over a
        subscription with batched draining; exceptions are logged, not
        fatal. Runs until cancelled (Service.stop) or self.running drops."""
        while self.running:
            batch = await sub.get_batch()
            for chan, msg in batch:
                try:
                    r = handler(chan, msg)
                    if asyncio.iscoroutine(r):
                        await r
                except Exception as e:
                    self.log.warning("handler error on %s: %r", chan, e)

    async def sleep(self, seconds: float):
        """Interruptible sleep honoring self.running."""
        end = time.monotonic() + seconds
        while self.running and time.monotonic() < end:
            await asyncio.sleep(min(0.1, end - time.monotonic()))
