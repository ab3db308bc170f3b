"""Lightweight span tracer -> Chrome trace-event JSON.

The reference ships elasticapm in requirements but never imports it
(SURVEY.md §5 'Tracing'). This tracer is what the new framework uses:
nested spans (sync/async context managers) across services and GPU ops,
dumped as chrome://tracing / Perfetto-compatible JSON. GPU kernel wall
times recorded via utils.metrics.record_kernel_time can be mirrored here
with `instant` events."""

from __future__ import annotations

import json
import threading
import time
from contextlib import contextmanager
from pathlib import Path


class Tracer:
    def __init__(self, service: str = "app", max_events: int = 100_000):
        self.service = service
        self.events: list[dict] = []
        self.max_events = max_events
        self._lock = threading.Lock()
        self.t0 = time.perf_counter()

    def _us(self) -> float:
        return (time.perf_counter() - self.t0) * 1e6

    def _emit(self, ev: dict):
        with self._lock:
            if len(self.events) < self.max_events:
                self.events.append(ev)

    @contextmanager
    def span(self, name: str, **args):
        t0 = self._us()
        try:
            yield
        finally:
            self._emit({
                "name": name, "ph": "X", "ts": t0,
                "dur": self._us() - t0, "pid": self.service,
                "tid": threading.current_thread().name,
                "args": args,
            })

    def instant(self, name: str, **args):
        self._emit({
            "name": name, "ph": "i", "ts": self._us(),
            "pid": self.service, "tid": threading.current_thread().name,
            "s": "p", "args": args,
        })

    def counter(self, name: str, **values):
        self._emit({
            "name": name, "ph": "C", "ts": self._us(),
            "pid": self.service, "args": values,
        })

    def dump(self, path: str | Path) -> Path:
        path = Path(path)
        path.parent.mkdir(parents=True, exist_ok=True)
        with self._lock:
            path.write_text(json.dumps(
                {"traceEvents": self.events,
                 "displayTimeUnit": "ms"}, indent=None))
        return path


_tracers: dict[str, Tracer] = {}


def get_tracer(service: str = "app") -> Tracer:
    if service not in _tracers:
        _tracers[service] = Tracer(service)
    return _tracers[service]
