"""Analysis toolkit, evaluation system, security/monitoring/checkpoint."""

import asyncio

import numpy as np
import pytest

from ai_crypto_trader_amd.analysis import (
    CryptoScanner, PositionSizer, TechnicalAnalyzer, TradingSignalVotes,
)
from ai_crypto_trader_amd.backtesting.engine import STRATEGY_PRESETS
from ai_crypto_trader_amd.backtesting.evaluation import (
    StrategyEvaluationSystem, calculate_advanced_metrics, calculate_metrics,
    generate_condition_market,
)
from ai_crypto_trader_amd.data.synthetic import candles_chl_v, generate_ohlcv
from ai_crypto_trader_amd.utils.api_security import (
    AccessLevel, APISecurityManager,
)
from ai_crypto_trader_amd.utils.checkpoint import CheckpointManager
from ai_crypto_trader_amd.utils.monitoring import (
    log_event, setup_json_logging, timed,
)


@pytest.fixture(scope="module")
def candles():
    return candles_chl_v(generate_ohlcv(3000, 1, seed=5))[0]


def test_technical_analyzer(candles):
    ta = TechnicalAnalyzer(candles)
    assert 0 <= ta.rsi() <= 100
    macd, sig, hist = ta.macd()
    assert abs(hist - (macd - sig)) < 1e-6
    bb = ta.bollinger()
    assert bb["lower"] <= bb["middle"] <= bb["upper"]
    assert ta.trend() in ("uptrend", "downtrend", "neutral")
    assert ta.volatility() > 0
    sr = ta.support_resistance()
    assert sr["support"] < sr["resistance"]
    ich = ta.ichimoku()
    assert ich["senkou_a"] == pytest.approx(
        (ich["tenkan"] + ich["kijun"]) / 2)
    assert ich["position"] in ("above_cloud", "below_cloud", "in_cloud")


def test_position_sizer():
    ps = PositionSizer(base_pct=0.1)
    low_vol = ps.calculate_position_size(10_000, 0.3)
    high_vol = ps.calculate_position_size(10_000, 1.5)
    assert high_vol["position_pct"] < low_vol["position_pct"]
    assert low_vol["take_profit_pct"] == pytest.approx(
        2 * low_vol["stop_loss_pct"])


def test_signal_votes(candles):
    sv = TradingSignalVotes(TechnicalAnalyzer(candles))
    sig = sv.signal()
    assert sig["decision"] in ("BUY", "SELL", "NEUTRAL")
    assert 0 <= sig["strength"] <= 100
    assert len(sig["votes"]) == 6


def test_scanner(candles):
    market = {"AUSDC": candles, "BUSDC": candles * 1.1}
    out = CryptoScanner().scan_market(market, top_k=2)
    assert len(out) == 2
    assert all(0 <= r["score"] <= 100 for r in out)
    assert out[0]["score"] >= out[1]["score"]


def test_evaluation_cv_and_requirements():
    ses = StrategyEvaluationSystem("cpu")
    mkt = generate_condition_market("bull", 3000, seed=2)
    cv = ses.cross_validate(mkt, STRATEGY_PRESETS["momentum"], k=3)
    assert len(cv["folds"]) == 3
    assert 0 <= cv["consistency"] <= 1
    from ai_crypto_trader_amd.config import AppConfig
    ok, fails = ses.meets_requirements(
        {"sharpe": 2.0, "win_rate": 0.6, "profit_factor": 1.5,
         "max_drawdown_pct": 5.0}, AppConfig().evolution)
    assert ok and not fails
    ok2, fails2 = ses.meets_requirements(
        {"sharpe": 0.0, "win_rate": 0.3, "profit_factor": 0.5,
         "max_drawdown_pct": 50.0}, AppConfig().evolution)
    assert not ok2 and len(fails2) == 4


def test_advanced_metrics_streaks():
    eq = np.linspace(1.0, 1.2, 100)
    trades = [{"pnl": 1}, {"pnl": 1}, {"pnl": -1}, {"pnl": 1},
              {"pnl": -1}, {"pnl": -1}, {"pnl": -1}]
    adv = calculate_advanced_metrics(eq, trades)
    assert adv["max_win_streak"] == 2
    assert adv["max_loss_streak"] == 3
    m = calculate_metrics(eq, trades)
    assert m["total_return_pct"] == pytest.approx(20.0, rel=1e-3)


def test_evaluator_improve_cycle():
    from ai_crypto_trader_amd.services.strategy_evaluator import (
        AIStrategyEvaluator,
    )
    ev = AIStrategyEvaluator("cpu")
    mkt = generate_condition_market("ranging", 2000, seed=9)
    res = ev.improve_cycle(dict(STRATEGY_PRESETS["momentum"]), mkt,
                           rounds=1)
    assert res["final_quality"] >= res["initial_quality"]
    assert res["history_len"] >= 1


def test_api_security(tmp_path):
    mgr = APISecurityManager(str(tmp_path / "keys.json"))
    kid, secret = mgr.issue_key("alice", AccessLevel.TRADE,
                                ip_whitelist=["1.2.3.4"])
    assert mgr.authenticate(kid, secret, AccessLevel.READ_ONLY, "1.2.3.4")
    assert not mgr.authenticate(kid, secret, AccessLevel.ADMIN, "1.2.3.4")
    assert not mgr.authenticate(kid, secret, AccessLevel.TRADE, "9.9.9.9")
    assert not mgr.authenticate(kid, "wrong", ip="1.2.3.4")
    new_secret = mgr.rotate_key(kid)
    assert not mgr.authenticate(kid, secret, ip="1.2.3.4")
    assert mgr.authenticate(kid, new_secret, ip="1.2.3.4")
    mgr.revoke_key(kid)
    assert not mgr.authenticate(kid, new_secret, ip="1.2.3.4")
    assert any(a["action"] == "auth_fail" for a in mgr.audit)
    # rotation scheduler
    mgr2 = APISecurityManager()
    k2, _ = mgr2.issue_key("bob")
    mgr2.keys[k2]["issued_at"] -= 31 * 86400
    assert k2 in mgr2.keys_needing_rotation()


def test_json_logging_and_timed(tmp_path, caplog):
    lg = setup_json_logging("testsvc", str(tmp_path))
    log_event(lg, "hello", foo=1)
    text = (tmp_path / "testsvc.log").read_text()
    assert '"event": "hello"' in text and '"foo": 1' in text

    calls = []

    class FakeMetrics:
        class latency:
            @staticmethod
            def labels(op):
                class H:
                    @staticmethod
                    def observe(v):
                        calls.append((op, v))
                return H

    @timed(FakeMetrics, "op1")
    def f():
        return 42

    assert f() == 42
    assert calls and calls[0][0] == "op1"


def test_checkpoint_roundtrip(tmp_path):
    import torch

    from ai_crypto_trader_amd.backtesting.ga_engine import GAEngine
    from ai_crypto_trader_amd.bus.message_bus import InProcessBus

    cm = CheckpointManager(str(tmp_path / "ck"), keep=2)
    candles = candles_chl_v(generate_ohlcv(1000, 1, seed=0))
    eng = GAEngine(candles, pop_per_rank=16, device="cpu", seed=4)
    eng.step()
    cm.save_ga(eng)
    eng2 = GAEngine(candles, pop_per_rank=16, device="cpu", seed=4)
    assert cm.load_ga(eng2)
    np.testing.assert_array_equal(eng2.pop_t.numpy(), eng.pop_t.numpy())
    assert eng2.gen == eng.gen

    model = torch.nn.Linear(4, 2)
    opt = torch.optim.Adam(model.parameters())
    cm.save_model(model, opt, tag="lin", meta={"v": 1})
    model2 = torch.nn.Linear(4, 2)
    meta = cm.load_model(model2, tag="lin")
    assert meta["v"] == 1
    torch.testing.assert_close(model2.weight, model.weight)

    async def bus_roundtrip():
        bus = InProcessBus()
        await bus.set("holdings", {"total_value": 5.0})
        await cm.save_bus_state(bus, ["holdings"], tag="b")
        bus2 = InProcessBus()
        n = await cm.load_bus_state(bus2, tag="b")
        assert n == 1
        assert (await bus2.get_json("holdings"))["total_value"] == 5.0

    asyncio.run(bus_roundtrip())


def test_env_overrides(monkeypatch, tmp_path):
    """Env-var flag layer (reference: EVOLUTION_METHOD / GA_POPULATION_SIZE
    env flags + generic ACT_section__field form)."""
    from ai_crypto_trader_amd.config import AppConfig

    monkeypatch.setenv("EVOLUTION_METHOD", "ga")
    monkeypatch.setenv("GA_POPULATION_SIZE", "64")
    monkeypatch.setenv("TRADING_SYMBOLS", "BTCUSDC, ETHUSDC")
    monkeypatch.setenv("ACT_risk__max_portfolio_var_pct", "0.123")
    monkeypatch.setenv("ACT_nosuch__field", "ignored")   # unknown: no crash
    cfg = AppConfig.load()
    assert cfg.evolution.method == "ga"
    assert cfg.evolution.population_size == 64
    assert cfg.trading.symbols == ["BTCUSDC", "ETHUSDC"]
    assert cfg.risk.max_portfolio_var_pct == 0.123


def test_params_from_code():
    """Code-to-params bridge (reference regex extraction from generated
    strategy code): JS and python spellings, percent normalization, and
    the extracted dict evaluates on the native engine."""
    from ai_crypto_trader_amd.backtesting.strategy import (
        DEFAULT_PARAMS, clip_params, params_to_dict,
    )
    from ai_crypto_trader_amd.services.strategy_evaluator import (
        params_from_code,
    )

    code = """
    // LLM-generated strategy
    const stopLoss = 0.03;
    const takeProfit = 6;       // percent
    let rsiPeriod = 21;
    "bollinger_period": 24,
    oversold: 22
    entry_votes = 3
    """
    got = params_from_code(code)
    assert got["stop_loss_pct"] == 0.03
    assert got["take_profit_pct"] == 0.06
    assert got["rsi_period"] == 21.0
    assert got["bb_window"] == 24.0
    assert got["rsi_oversold"] == 22.0
    assert got["entry_votes"] == 3.0
    # merge onto defaults -> valid native strategy vector
    base = params_to_dict(DEFAULT_PARAMS)
    base.update(got)
    import numpy as np
    from ai_crypto_trader_amd.backtesting.strategy import dict_to_params
    vec = clip_params(dict_to_params(base)[None])[0]
    assert np.isfinite(vec).all()


def test_ga_checkpoint_resume_deterministic(tmp_path):
    """Checkpoint/resume reproduces the uninterrupted GA trajectory
    exactly (Philox-seeded evolution is a pure function of (pop, gen))."""
    from ai_crypto_trader_amd.backtesting.ga_engine import GAEngine
    from ai_crypto_trader_amd.data.synthetic import (
        candles_chl_v, generate_ohlcv,
    )
    from ai_crypto_trader_amd.utils.checkpoint import CheckpointManager

    candles = candles_chl_v(generate_ohlcv(1200, 2, seed=6))

    ref = GAEngine(candles, pop_per_rank=16, device="cpu", seed=9)
    for _ in range(6):
        ref.step()

    a = GAEngine(candles, pop_per_rank=16, device="cpu", seed=9)
    for _ in range(3):
        a.step()
    cm = CheckpointManager(str(tmp_path / "ck"))
    cm.save_ga(a, tag="resume_test")

    b = GAEngine(candles, pop_per_rank=16, device="cpu", seed=9)
    assert cm.load_ga(b, tag="resume_test")
    assert b.gen == 3
    for _ in range(3):
        b.step()

    np.testing.assert_array_equal(b.pop_t.numpy(), ref.pop_t.numpy())


def test_period_return_buckets():
    """Daily/monthly return buckets (reference :180-188)."""
    from ai_crypto_trader_amd.backtesting.evaluation import (
        calculate_metrics, period_returns,
    )

    # 3 exact days of 0.1%/day compounding at 1-minute resolution
    daily_r = 0.001
    per_min = (1 + daily_r) ** (1 / 1440)
    eq = per_min ** np.arange(3 * 1440 + 1)
    d = period_returns(eq, 1440)
    assert d["n_periods"] == 3
    np.testing.assert_allclose(d["returns_pct"], [0.1] * 3, rtol=1e-5)
    assert d["positive_share"] == 1.0
    m = calculate_metrics(eq)
    assert m["daily"]["n_periods"] == 3
    assert m["monthly"]["n_periods"] == 0    # < one month of candles


def test_strategy_validator():
    from ai_crypto_trader_amd.backtesting.strategy import (
        DEFAULT_PARAMS, params_to_dict,
    )
    from ai_crypto_trader_amd.services.strategy_evaluator import (
        validate_strategy,
    )

    ok, issues = validate_strategy(params_to_dict(DEFAULT_PARAMS))
    assert ok, issues
    bad = params_to_dict(DEFAULT_PARAMS)
    bad["stop_loss_pct"] = 0.0
    bad["rsi_oversold"] = 500.0
    ok2, issues2 = validate_strategy(bad)
    assert not ok2 and len(issues2) >= 2
