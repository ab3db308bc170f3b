#!/usr/bin/env python3
"""Dashboard (reference parity: dashboard.py:47-2315 — Dash/Plotly app
subscribing 7+ channels into an in-memory DataStore with ~20 views).

MI355X-image stack: FastAPI + uvicorn serving (a) a JSON API exposing the
same state views (portfolio, signals, trades, risk, regime, MC,
predictions, patterns, explanations) and (b) a self-refreshing HTML
overview. Runs embedded in run_trader (same process, same bus) or
standalone against a Redis bus.

  python dashboard.py            # standalone demo on synthetic replay
"""

from __future__ import annotations

import asyncio
import json
from collections import deque

from ai_crypto_trader_amd.bus.message_bus import InProcessBus
from ai_crypto_trader_amd.bus.schema import Channels, Keys


class DataStore:
    """In-memory rolling store fed by a bus listener (reference
    dashboard.py:47-137)."""

    CHANNELS = (
        Channels.MARKET_UPDATES, Channels.SOCIAL_UPDATES,
        Channels.TRADING_SIGNALS, Channels.TRADE_EXECUTIONS,
        Channels.RISK_ALERTS, Channels.PATTERN_SIGNALS,
        Channels.NN_PREDICTIONS, Channels.EXPLAINED_TRADING_SIGNALS,
        Channels.STRATEGY_SWITCH, Channels.STRATEGY_EVOLUTION_UPDATES,
    )

    def __init__(self, bus, maxlen: int = 500):
        self.bus = bus
        self.buffers: dict[str, deque] = {
            c: deque(maxlen=maxlen) for c in self.CHANNELS
        }
        self.prices: dict[str, deque] = {}
        self._task = None

    async def start(self):
        sub = self.bus.subscribe(*self.CHANNELS)

        async def listen():
            while True:
                batch = await sub.get_batch()
                for chan, msg in batch:
                    self.buffers[chan].append(msg)
                    if chan == Channels.MARKET_UPDATES and \
                            isinstance(msg, dict) and msg.get("symbol"):
                        d = self.prices.setdefault(
                            msg["symbol"], deque(maxlen=2000))
                        d.append(msg["current_price"])

        self._task = asyncio.create_task(listen())

    def recent(self, channel: str, n: int = 50) -> list:
        return list(self.buffers.get(channel, []))[-n:]


def build_app(bus, store: DataStore):
    from contextlib import asynccontextmanager

    from fastapi import FastAPI
    from fastapi.responses import HTMLResponse

    @asynccontextmanager
    async def lifespan(_app):
        await store.start()
        yield

    app = FastAPI(title="ai-crypto-trader-amd dashboard",
                  lifespan=lifespan)

    @app.get("/api/portfolio")
    async def portfolio():
        return {
            "holdings": await bus.get_json(Keys.HOLDINGS),
            "active_trades": await bus.get_json(Keys.ACTIVE_TRADES),
            "trailing_stops": await bus.get_json(Keys.TRAILING_STOPS),
        }

    @app.get("/api/risk")
    async def risk():
        return {
            "portfolio_risk": await bus.get_json(Keys.PORTFOLIO_RISK),
            "diversification":
                await bus.get_json(Keys.PORTFOLIO_DIVERSIFICATION),
            "adaptive_stops":
                await bus.get_json(Keys.ADAPTIVE_STOP_LOSSES),
            "alerts": store.recent(Channels.RISK_ALERTS),
        }

    @app.get("/api/signals")
    async def signals():
        return store.recent(Channels.TRADING_SIGNALS)

    @app.get("/api/trades")
    async def trades():
        return store.recent(Channels.TRADE_EXECUTIONS)

    @app.get("/api/regime")
    async def regime():
        return {
            "current": await bus.get_json(Keys.CURRENT_MARKET_REGIME),
            "history": await bus.get_json(Keys.MARKET_REGIME_HISTORY),
            "switches": store.recent(Channels.STRATEGY_SWITCH),
        }

    @app.get("/api/monte_carlo")
    async def monte_carlo():
        return {
            "results": await bus.get_json(Keys.MONTE_CARLO_RESULTS),
            "fan_chart": await bus.get_json(Keys.MC_FAN_CHART),
        }

    @app.get("/api/predictions")
    async def predictions():
        return store.recent(Channels.NN_PREDICTIONS)

    @app.get("/api/patterns")
    async def patterns():
        return {
            "report": await bus.get_json(Keys.PATTERN_ANALYSIS_REPORT),
            "signals": store.recent(Channels.PATTERN_SIGNALS),
        }

    @app.get("/api/explanations")
    async def explanations():
        return store.recent(Channels.EXPLAINED_TRADING_SIGNALS, 10)

    @app.get("/api/prices/{symbol}")
    async def prices(symbol: str):
        return list(store.prices.get(symbol, []))

    @app.get("/api/evolution")
    async def evolution():
        return {
            "updates": store.recent(Channels.STRATEGY_EVOLUTION_UPDATES),
            "params": await bus.get_json(Keys.STRATEGY_PARAMS),
        }

    @app.get("/api/social")
    async def social():
        # social panel (reference dashboard.py:759)
        metrics = await bus.hgetall(Keys.SOCIAL_METRICS)
        return {
            "metrics": {k: json.loads(v) for k, v in metrics.items()},
            "updates": store.recent(Channels.SOCIAL_UPDATES, 20),
            "risk_adjustments":
                await bus.get_json(Keys.SOCIAL_RISK_REPORT),
        }

    @app.get("/api/correlation")
    async def correlation():
        # correlation heatmap data (reference dashboard.py:1712)
        risk_d = await bus.get_json(Keys.PORTFOLIO_RISK) or {}
        return {
            "correlation_matrix": risk_d.get("correlation_matrix"),
            "symbols": risk_d.get("symbols"),
            "avg_correlation": risk_d.get("avg_correlation"),
        }

    @app.get("/api/models")
    async def models():
        # AI model performance/comparison panels (reference :1180-1479)
        return {
            "registry": await bus.get_json(Keys.MODEL_REGISTRY),
            "events": store.recent(Channels.MODEL_REGISTRY_EVENTS, 20),
            "performance":
                store.recent(Channels.MODEL_PERFORMANCE_UPDATES, 20),
        }

    @app.get("/", response_class=HTMLResponse)
    async def index():
        holdings = await bus.get_json(Keys.HOLDINGS) or {}
        regime = await bus.get_json(Keys.CURRENT_MARKET_REGIME) or {}
        risk_d = await bus.get_json(Keys.PORTFOLIO_RISK) or {}
        sigs = store.recent(Channels.TRADING_SIGNALS, 10)
        rows = "".join(
            f"<tr><td>{s.get('symbol')}</td><td>{s.get('decision')}</td>"
            f"<td>{s.get('confidence', 0):.2f}</td></tr>"
            for s in reversed(sigs) if isinstance(s, dict))
        return f"""<html><head><title>ai-crypto-trader-amd</title>
<meta http-equiv=refresh content=5>
<style>body{{font-family:monospace;background:#111;color:#0f0}}
table{{border-collapse:collapse}}td,th{{border:1px solid #333;padding:4px}}
</style></head><body>
<h2>ai-crypto-trader-amd</h2>
<p>portfolio: ${holdings.get('total_value', 0):,.2f} |
regime: {regime.get('regime', '?')} |
VaR: {risk_d.get('portfolio_var_pct', 0):.2f}%</p>
<h3>recent signals</h3>
<table><tr><th>symbol</th><th>decision</th><th>conf</th></tr>{rows}</table>
<p>JSON API: /api/portfolio /api/risk /api/signals /api/trades /api/regime
/api/monte_carlo /api/predictions /api/patterns /api/explanations
/api/evolution</p>
</body></html>"""

    return app


def main():
    import uvicorn

    bus = InProcessBus()
    store = DataStore(bus)
    app = build_app(bus, store)

    async def run_all():
        # standalone demo: replay a synthetic market through the monitor
        # on the SAME event loop as the server (the in-process bus is
        # single-loop by design)
        from ai_crypto_trader_amd.config import AppConfig
        from ai_crypto_trader_amd.data.feed import SyntheticFeed
        from ai_crypto_trader_amd.data.synthetic import (
            candles_chl_v, generate_ohlcv,
        )
        from ai_crypto_trader_amd.services.market_monitor import (
            MarketMonitorService,
        )
        symbols = ["BTCUSDC", "ETHUSDC"]
        feed = SyntheticFeed(candles_chl_v(generate_ohlcv(50_000, 2)),
                             symbols, speed=60.0)
        svc = MarketMonitorService(bus, feed, AppConfig())
        await svc.start()
        server = uvicorn.Server(uvicorn.Config(
            app, host="127.0.0.1", port=8050, log_level="warning"))
        await server.serve()

    asyncio.run(run_all())


if __name__ == "__main__":
    main()
