"""Offline backtesting vertical: data manager, engine, analyzer, CLI."""

import json
import subprocess
import sys

import numpy as np
import pytest

from ai_crypto_trader_amd.backtesting.data_manager import (
    HistoricalDataManager, SocialDataProvider,
)
from ai_crypto_trader_amd.backtesting.engine import (
    BacktestEngine,
)
from ai_crypto_trader_amd.backtesting.result_analyzer import ResultAnalyzer


@pytest.fixture()
def dm(tmp_path):
    return HistoricalDataManager(str(tmp_path / "data"))


def test_fetch_and_load_roundtrip(dm):
    df = dm.fetch_market_data("BTCUSDC", "1m", 500)
    assert len(df) == 500
    df2 = dm.load_market_data("BTCUSDC", "1m")
    np.testing.assert_allclose(df2["close"], df["close"], rtol=1e-6)
    # deterministic per (symbol, interval)
    df3 = dm.fetch_market_data("BTCUSDC", "1m", 500)
    np.testing.assert_array_equal(df3["close"], df["close"])


def test_social_merge(dm):
    m = dm.fetch_market_data("ETHUSDC", "1m", 2000)
    s = dm.fetch_social_data("ETHUSDC", n_days=2)
    merged = dm.merge_market_and_social_data(m, s)
    assert len(merged) == len(m)
    assert "sentiment" in merged
    assert merged["sentiment"].notna().all()
    # provider: neutral defaults before any social row
    prov = SocialDataProvider(dm)
    assert prov.at("NOSUCH", 0)["sentiment"] == 0.5
    at = prov.at("ETHUSDC", int(s["timestamp"].iloc[-1]) + 1)
    assert at["sentiment"] == pytest.approx(float(s["sentiment"].iloc[-1]))


def test_backtest_engine_cpu(tmp_path):
    eng = BacktestEngine(str(tmp_path / "d"), device="cpu")
    stats = eng.run_backtest("BTCUSDC", "dca_strategy", n_candles=3000)
    for k in ("total_return_pct", "win_rate", "profit_factor", "sharpe",
              "max_drawdown_pct", "n_trades", "candles_per_sec"):
        assert k in stats
    assert stats["engine"] == "cpu"
    # determinism
    stats2 = eng.run_backtest("BTCUSDC", "dca_strategy", n_candles=3000)
    assert stats2["final_equity"] == stats["final_equity"]


def test_backtest_sweep_and_analyzer(tmp_path):
    eng = BacktestEngine(str(tmp_path / "d"), device="cpu")
    out = eng.run_multiple_backtests(
        ["BTCUSDC", "ETHUSDC"], ["momentum", "conservative"],
        n_candles=2000)
    assert len(out["results"]) == 4
    ra = ResultAnalyzer(str(tmp_path / "an"))
    rep = ra.summary_report(out["results"])
    assert rep["n"] == 4 and "best_by" in rep
    assert ra.comparison_chart(out["results"]) is not None
    stats = eng.run_backtest("BTCUSDC", "momentum", n_candles=2000,
                             record_equity=True)
    assert ra.plot_equity_curve(stats) is not None


def test_ga_optimize_improves(tmp_path):
    eng = BacktestEngine(str(tmp_path / "d"), device="cpu")
    base = eng.run_backtest("BTCUSDC", "default", n_candles=1500)
    opt = eng.optimize("BTCUSDC", pop_size=24, generations=4,
                       n_candles=1500)
    assert opt["optimize"]["best_fitness"] >= base["fitness"]


def test_cli_smoke(tmp_path):
    d = str(tmp_path / "cli")
    r = subprocess.run(
        [sys.executable, "run_backtest.py", "--data-dir", d, "fetch",
         "--symbol", "SOLUSDC", "--candles", "1200"],
        capture_output=True, text=True, timeout=180)
    assert r.returncode == 0, r.stderr
    r = subprocess.run(
        [sys.executable, "run_backtest.py", "--data-dir", d, "backtest",
         "--symbol", "SOLUSDC", "--strategy", "momentum", "--candles",
         "1200", "--cpu"],
        capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr
    out = json.loads(r.stdout)
    assert out["symbol"] == "SOLUSDC"
    r = subprocess.run(
        [sys.executable, "run_backtest.py", "--data-dir", d, "list"],
        capture_output=True, text=True, timeout=120)
    assert r.returncode == 0 and "SOLUSDC" in r.stdout


def test_custom_strategy_tester():
    from ai_crypto_trader_amd.backtesting.strategy_tester import (
        StrategyTester, rsi_threshold_strategy,
    )
    from ai_crypto_trader_amd.data.synthetic import (
        candles_chl_v, generate_ohlcv,
    )

    candles = candles_chl_v(generate_ohlcv(4000, 1, seed=31, sigma=1.5))[0]
    tester = StrategyTester()
    res = tester.backtest_strategy(candles, rsi_threshold_strategy(35, 65))
    assert res.stats["n_trades"] > 0
    assert len(res.trades) == res.stats["n_trades"]
    assert res.equity_curve is not None and len(res.equity_curve) == 4000
    # every closed trade honors the stop/take mechanics
    for tr in res.trades:
        assert tr.exit_t > tr.entry_t
        assert tr.reason in ("stop_loss", "take_profit", "signal")
        if tr.reason == "stop_loss":
            assert tr.exit_price < tr.entry_price
        if tr.reason == "take_profit":
            assert tr.exit_price > tr.entry_price

    # arbitrary callable: never trades -> flat equity
    res2 = tester.backtest_strategy(candles, lambda ctx: "HOLD")
    assert res2.stats["n_trades"] == 0
    assert res2.stats["final_equity"] == 1.0

    # context carries indicators + social seam
    seen = {}

    def probe(ctx):
        seen.update(ctx)
        return "HOLD"

    tester.backtest_strategy(candles, probe)
    for k in ("rsi14", "macd", "bb_up", "close", "in_position"):
        assert k in seen


def test_interval_aware_sharpe(tmp_path):
    """Sharpe annualizes by the interval's bars-per-year (recomputed
    host-side from the kernel's raw return moments): the same per-bar
    return stream annualizes sqrt(525600/8760) ~ 7.75x higher at 1m than
    at 1h."""
    import numpy as np

    from ai_crypto_trader_amd.backtesting.engine import metrics_to_stats
    from ai_crypto_trader_amd.backtesting.engine_cpu import METRIC_NAMES

    m = np.zeros(len(METRIC_NAMES), np.float32)
    d = dict(zip(METRIC_NAMES, range(len(METRIC_NAMES))))
    m[d["final_equity"]] = 1.1
    m[d["n_trades"]] = 5
    m[d["wins"]] = 3
    m[d["sum_ret"]] = 0.10          # T=1000 bars of identical moments
    m[d["sum_ret2"]] = 0.001
    s1m = metrics_to_stats(m, 1000, "1m")["sharpe"]
    s1h = metrics_to_stats(m, 1000, "1h")["sharpe"]
    assert s1m == pytest.approx(s1h * np.sqrt(525_600 / 8_760), rel=1e-6)
    # default matches the kernel's 1m convention closely
    assert s1m > 0


def test_multi_interval_backtest(tmp_path):
    """Backtests run at non-1m intervals end-to-end with interval-aware
    annualization."""
    eng = BacktestEngine(str(tmp_path / "d"), device="cpu")
    s1h = eng.run_backtest("BTCUSDC", "momentum", interval="1h",
                           n_candles=1500)
    assert s1h["interval"] == "1h" and s1h["n_candles"] == 1500
    assert np.isfinite(s1h["sharpe"])
    s1m = eng.run_backtest("BTCUSDC", "momentum", interval="1m",
                           n_candles=1500)
    # different interval data => independent results, both well-formed
    assert s1m["interval"] == "1m"
    assert np.isfinite(s1m["sharpe"])
