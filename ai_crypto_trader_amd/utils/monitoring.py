"""Structured logging + timing helpers (reference parity:
services/utils/monitoring.py — structlog-style JSON logging :29-97,
metric helper factories with dummy fallbacks :123-250, @timed decorator
:252)."""

from __future__ import annotations

import asyncio
import functools
import json
import logging
import time
from pathlib import Path


class JsonFormatter(logging.Formatter):
    def format(self, record: logging.LogRecord) -> str:
        out = {
            "ts": round(record.created, 3),
            "level": record.levelname.lower(),
            "logger": record.name,
            "event": record.getMessage(),
        }
        extra = getattr(record, "ctx", None)
        if extra:
            out.update(extra)
        if record.exc_info:
            out["exc"] = self.formatException(record.exc_info)
        return json.dumps(out)


def setup_json_logging(name: str, log_dir: str | None = "logs",
                       level=logging.INFO,
                       max_bytes: int = 10 * 1024 * 1024,
                       backups: int = 5) -> logging.Logger:
    """JSON logs to logs/<name>.log (RotatingFileHandler 10MB x 5 — the
    reference's per-service pattern, e.g. monte_carlo_service.py:24-39)."""
    logger = logging.getLogger(f"json.{name}")
    if logger.handlers:
        return logger
    logger.setLevel(level)
    if log_dir:
        from logging.handlers import RotatingFileHandler

        Path(log_dir).mkdir(parents=True, exist_ok=True)
        h = RotatingFileHandler(Path(log_dir) / f"{name}.log",
                                maxBytes=max_bytes, backupCount=backups)
    else:
        h = logging.StreamHandler()
    h.setFormatter(JsonFormatter())
    logger.addHandler(h)
    logger.propagate = False
    return logger


def log_event(logger: logging.Logger, event: str, **ctx):
    logger.info(event, extra={"ctx": ctx})


def timed(metrics=None, op: str | None = None):
    """@timed decorator (reference :252): records wall time into the
    metrics latency histogram (utils/metrics.py) or logs it."""

    def deco(fn):
        name = op or fn.__name__

        def record(dt):
            if metrics is not None:
                metrics.latency.labels(name).observe(dt)

        if asyncio.iscoroutinefunction(fn):
            @functools.wraps(fn)
            async def aw(*a, **k):
                t0 = time.perf_counter()
                try:
                    return await fn(*a, **k)
                finally:
                    record(time.perf_counter() - t0)
            return aw

        @functools.wraps(fn)
        def w(*a, **k):
            t0 = time.perf_counter()
            try:
                return fn(*a, **k)
            finally:
                record(time.perf_counter() - t0)
        return w

    return deco
