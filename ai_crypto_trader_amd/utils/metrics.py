"""Prometheus metrics (reference parity: services/utils/metrics.py:18-221 —
~20 trading metrics + /metrics + /health endpoints; gauges/counters/
histograms via prometheus_client, aiohttp server per service).

Adds the GPU-era metrics the reference lacks (SURVEY.md §5 calls rocprof
captures + per-kernel timings a deliverable): per-kernel hipEvent timings
surfaced as histograms via record_kernel_time()."""

from __future__ import annotations

import os

try:
    from prometheus_client import (
        CollectorRegistry, Counter, Gauge, Histogram, generate_latest,
    )
    HAVE_PROM = True
except ImportError:          # pragma: no cover
    HAVE_PROM = False

ENABLE_METRICS = os.environ.get("ENABLE_METRICS", "1") == "1"


class _Dummy:
    def labels(self, *a, **k):
        return self

    def inc(self, *a):
        pass

    def set(self, *a):
        pass

    def observe(self, *a):
        pass


class Metrics:
    def __init__(self, service: str = "app"):
        self.service = service
        if not (HAVE_PROM and ENABLE_METRICS):
            self.registry = None
            d = _Dummy()
            for name in ("signals", "executions", "portfolio_value",
                         "ai_confidence", "social_sentiment", "var",
                         "latency", "errors", "kernel_time",
                         "candles_per_sec", "mc_paths_per_sec",
                         "active_trades", "equity"):
                setattr(self, name, d)
            return
        self.registry = CollectorRegistry()
        r = self.registry
        self.signals = Counter(
            "trading_signals_total", "Trading signals",
            ["symbol", "decision"], registry=r)
        self.executions = Counter(
            "trade_executions_total", "Trade executions",
            ["symbol", "side"], registry=r)
        self.portfolio_value = Gauge(
            "portfolio_value_usd", "Portfolio value", registry=r)
        self.equity = Gauge("equity", "Account equity", registry=r)
        self.service_health = Gauge(
            "crypto_trader_service_health",
            "1 = service healthy (reference PRODUCTION_READINESS metric)",
            ["service_name"], registry=r)
        self.active_trades = Gauge(
            "active_trades", "Open positions", registry=r)
        self.ai_confidence = Gauge(
            "ai_confidence", "Last signal confidence", ["symbol"],
            registry=r)
        self.social_sentiment = Gauge(
            "social_sentiment", "Weighted sentiment", ["symbol"], registry=r)
        self.var = Gauge("portfolio_var", "Portfolio VaR", registry=r)
        self.latency = Histogram(
            "request_duration_seconds", "Operation latency", ["op"],
            registry=r)
        self.errors = Counter(
            "errors_total", "Errors", ["service", "kind"], registry=r)
        self.kernel_time = Histogram(
            "hip_kernel_seconds", "HIP kernel wall time", ["kernel"],
            registry=r,
            buckets=(1e-5, 1e-4, 1e-3, 1e-2, 0.1, 0.5, 1.0, 5.0))
        self.candles_per_sec = Gauge(
            "backtest_candles_per_sec", "Backtest throughput", registry=r)
        self.mc_paths_per_sec = Gauge(
            "mc_paths_per_sec", "Monte-Carlo throughput", registry=r)

    def record_kernel_time(self, kernel: str, seconds: float):
        self.kernel_time.labels(kernel).observe(seconds)

    def export(self) -> bytes:
        if self.registry is None:
            return b""
        return generate_latest(self.registry)


_metrics: dict[str, Metrics] = {}


def get_metrics(service: str = "app") -> Metrics:
    if service not in _metrics:
        _metrics[service] = Metrics(service)
    return _metrics[service]


class GpuTimer:
    """hipEvent-based kernel timing surfaced as Prometheus histograms
    ('per-kernel timing via hipEvents surfaced as Prometheus
    metrics'). torch.cuda.Event IS a hipEvent on ROCm.

        with GpuTimer(metrics, "mc_paths"):
            ...launch kernels...
    Falls back to wall time on CPU."""

    def __init__(self, metrics: Metrics, kernel: str):
        self.metrics = metrics
        self.kernel = kernel
        self._ev = None
        self._t0 = None

    def __enter__(self):
        try:
            import torch

            if torch.cuda.is_available():
                self._ev = (torch.cuda.Event(enable_timing=True),
                            torch.cuda.Event(enable_timing=True))
                self._ev[0].record()
                return self
        except Exception:
            pass
        import time

        self._t0 = time.perf_counter()
        return self

    def __exit__(self, *exc):
        if self._ev is not None:
            import torch

            self._ev[1].record()
            self._ev[1].synchronize()
            secs = self._ev[0].elapsed_time(self._ev[1]) / 1e3
        else:
            import time

            secs = time.perf_counter() - self._t0
        self.metrics.record_kernel_time(self.kernel, secs)
        self.seconds = secs
        return False
