"""Service supervisor: failure detection + elastic recovery.

The reference's recovery story is Docker `restart: unless-stopped` +
`nc -z` healthchecks (SURVEY.md §5 'Failure detection'). In the
single-process MI355X topology the supervisor takes that role: it polls
every service's health, restarts services whose tasks died (with
exponential backoff and a restart budget), and exposes the same signals
the Prometheus ServiceDown alert consumes. Fault injection for tests:
`inject_failure(service)` kills a service's tasks the way a real crash
would."""

from __future__ import annotations

import asyncio
import time

from .base import Service


class ServiceSupervisor(Service):
    name = "supervisor"

    def __init__(self, bus, services: list[Service], config=None,
                 check_interval: float = 1.0, max_restarts: int = 5,
                 backoff_base: float = 1.0):
        super().__init__(bus, config)
        self.services = services
        self.check_interval = check_interval
        self.max_restarts = max_restarts
        self.backoff_base = backoff_base
        self.restarts: dict[str, int] = {}
        self.last_restart: dict[str, float] = {}
        self.events: list[dict] = []

    def _needs_restart(self, s: Service) -> bool:
        if not s.running:
            return False            # deliberately stopped / finished
        if s.healthy:
            return False
        n = self.restarts.get(s.name, 0)
        if n >= self.max_restarts:
            return False
        backoff = self.backoff_base * (2 ** n)
        return time.monotonic() - self.last_restart.get(s.name, 0.0) \
            >= backoff

    async def restart(self, s: Service):
        self.restarts[s.name] = self.restarts.get(s.name, 0) + 1
        self.last_restart[s.name] = time.monotonic()
        self.events.append({"at": time.time(), "service": s.name,
                            "restart": self.restarts[s.name]})
        self.log.warning("restarting %s (attempt %d)", s.name,
                         self.restarts[s.name])
        try:
            await s.stop()
        except Exception:
            pass
        s.healthy = True
        await s.start()
        await self.bus.publish("service_restarts", {
            "service": s.name, "attempt": self.restarts[s.name],
        })

    async def run(self):
        while self.running:
            for s in self.services:
                self.metrics.service_health.labels(s.name).set(
                    1.0 if s.healthy else 0.0)
                if self._needs_restart(s):
                    try:
                        await self.restart(s)
                    except Exception as e:
                        self.log.error("restart of %s failed: %r",
                                       s.name, e)
            await self.sleep(self.check_interval)

    def status(self) -> dict:
        return {
            "services": {s.name: s.health() for s in self.services},
            "restarts": dict(self.restarts),
        }

    # --- fault injection (tests) ----------------------------------------
    @staticmethod
    async def inject_failure(s: Service):
        """Simulate a crash: cancel the service's tasks and mark it
        unhealthy, as _task_done would after an unhandled exception."""
        for t in s._tasks:
            t.cancel()
        await asyncio.gather(*s._tasks, return_exceptions=True)
        s.healthy = False
