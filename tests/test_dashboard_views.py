"""Dashboard view tests: every view endpoint renders
from a replay run and every JSON endpoint honors its data contract
(reference views: dashboard.py:509-1937)."""

import asyncio

import pytest

pytest.importorskip("fastapi")
from fastapi.testclient import TestClient  # noqa: E402

from ai_crypto_trader_amd.bus.message_bus import InProcessBus  # noqa: E402
from ai_crypto_trader_amd.bus.schema import Channels, Keys  # noqa: E402
from dashboard import VIEWS, DataStore, build_app  # noqa: E402


@pytest.fixture(scope="module")
def client():
    bus = InProcessBus()
    store = DataStore(bus, poll_s=0.05)

    async def seed():
        # (store already started by the app lifespan)
        # replay: market updates feed the chart series
        for i in range(40):
            px = 100 + i * 0.1
            await bus.publish(Channels.MARKET_UPDATES, {
                "symbol": "BTCUSDC", "current_price": px,
                "rsi": 50 + i % 20, "macd": 0.01 * (i % 5 - 2),
                "bb_position": 0.5,
            })
        await bus.publish(Channels.TRADING_SIGNALS, {
            "symbol": "BTCUSDC", "decision": "BUY", "confidence": 0.9,
            "risk_info": {"optimal_position_pct": 0.11,
                          "var_pct": 2.0},
        })
        await bus.publish(Channels.TRADE_EXECUTIONS, {
            "symbol": "BTCUSDC", "side": "BUY", "qty": 0.1,
            "price": 100.0,
        })
        await bus.publish(Channels.EXPLAINED_TRADING_SIGNALS, {
            "symbol": "BTCUSDC",
            "factor_weights": {"rsi": 0.4, "macd": 0.3, "social": 0.3},
            "explanation": {"summary": "momentum + social tailwind"},
        })
        await bus.publish(Channels.SOCIAL_UPDATES, {
            "symbol": "BTCUSDC",
            "data": {"weighted_sentiment": 0.62, "metrics": {}},
        })
        await bus.publish(Channels.PATTERN_SIGNALS, {
            "symbol": "BTCUSDC", "pattern": "double_top",
            "signal": "SELL", "confidence": 0.6, "completion": 80.0,
        })
        await bus.set(Keys.HOLDINGS, {
            "holdings": {"USDC": {"qty": 1000, "value": 1000}},
            "total_value": 1000.0, "timestamp": 0,
        })
        await bus.set(Keys.PORTFOLIO_RISK, {
            "portfolio_var_pct": 2.5, "cvar_pct": 3.4,
            "correlation_matrix": [[1.0, 0.4], [0.4, 1.0]],
            "symbols": ["BTCUSDC", "ETHUSDC"],
            "avg_correlation": 0.4,
        })
        await bus.set(Keys.ACTIVE_TRADES, {
            "BTCUSDC": {"symbol": "BTCUSDC", "qty": 0.1,
                        "entry_price": 100.0, "stop_price": 95.0,
                        "tp_price": 110.0, "stop_order_id": "1"},
        })
        await bus.set(Keys.TRAILING_STOPS, {
            "BTCUSDC": {"entry": 100.0, "stop": 95.0, "peak": 103.0,
                        "active": True, "updates": 1},
        })
        await bus.set(Keys.CURRENT_MARKET_REGIME,
                      {"regime": "bull", "confidence": 0.8})
        await bus.set(Keys.MC_FAN_CHART, {
            "scenario": "base", "symbols": ["BTCUSDC"],
            "horizon_days": 5, "n_paths": 1000,
            "percentiles": {"5": [1, 0.99], "50": [1, 1.01],
                            "95": [1, 1.04]},
        })
        await bus.set(Keys.MODEL_REGISTRY, {
            "models": {"lstm_1m": {"score": 0.61, "status": "active"}},
        })
        await asyncio.sleep(0.2)      # poll loop captures key history

    app = build_app(bus, store)
    with TestClient(app) as c:
        # TestClient starts the lifespan (store.start) in its own loop;
        # seed through that loop via the portal
        c.portal.call(lambda: asyncio.get_event_loop().create_task(
            seed()))
        import time
        time.sleep(0.6)
        yield c


def test_all_views_render(client):
    for v in VIEWS:
        r = client.get(f"/view/{v}")
        assert r.status_code == 200, v
        assert "<canvas" in r.text or "<div id=" in r.text, v
        assert "getJSON" in r.text or "load()" in r.text, v


def test_index_links_views(client):
    r = client.get("/")
    assert r.status_code == 200
    for v in VIEWS:
        assert f"/view/{v}" in r.text


def test_chart_contract(client):
    d = client.get("/api/chart/BTCUSDC").json()
    assert d["symbol"] == "BTCUSDC"
    assert len(d["candles"]) == 40
    c0 = d["candles"][1]
    assert set(c0) == {"open", "close", "high", "low"}
    assert c0["low"] <= c0["open"] <= c0["high"]
    assert len(d["rsi"]) == len(d["macd"]) == len(d["bb_position"]) == 40


def test_var_and_equity_history_contract(client):
    vh = client.get("/api/var_history").json()
    assert vh and {"t", "var_pct", "cvar_pct"} <= set(vh[0])
    assert vh[-1]["var_pct"] == pytest.approx(2.5)
    eh = client.get("/api/equity_history").json()
    assert eh and eh[-1]["total_value"] == pytest.approx(1000.0)


def test_correlation_contract(client):
    d = client.get("/api/correlation").json()
    assert d["correlation_matrix"] == [[1.0, 0.4], [0.4, 1.0]]
    assert d["symbols"] == ["BTCUSDC", "ETHUSDC"]


def test_portfolio_and_stops_contract(client):
    d = client.get("/api/portfolio").json()
    at = d["active_trades"]["BTCUSDC"]
    assert at["stop_price"] < at["entry_price"] < at["tp_price"]
    assert d["trailing_stops"]["BTCUSDC"]["peak"] == pytest.approx(103.0)


def test_signals_sizing_contract(client):
    sigs = client.get("/api/signals").json()
    assert sigs and sigs[-1]["risk_info"]["optimal_position_pct"] == \
        pytest.approx(0.11)


def test_explanations_contract(client):
    ex = client.get("/api/explanations").json()
    assert ex and "factor_weights" in ex[-1]
    assert sum(ex[-1]["factor_weights"].values()) == pytest.approx(1.0)


def test_mc_fan_contract(client):
    d = client.get("/api/monte_carlo").json()
    fc = d["fan_chart"]
    assert "percentiles" in fc and "50" in fc["percentiles"]


def test_models_contract(client):
    d = client.get("/api/models").json()
    assert d["registry"]["models"]["lstm_1m"]["score"] == \
        pytest.approx(0.61)


def test_social_patterns_regime_contract(client):
    s = client.get("/api/social").json()
    assert s["updates"][-1]["data"]["weighted_sentiment"] == \
        pytest.approx(0.62)
    p = client.get("/api/patterns").json()
    assert p["signals"][-1]["pattern"] == "double_top"
    r = client.get("/api/regime").json()
    assert r["current"]["regime"] == "bull"
