"""CPU reference implementations: Monte-Carlo, covariance, indicators, GA."""

import numpy as np

from ai_crypto_trader_amd.data.synthetic import candles_chl_v, generate_ohlcv
from ai_crypto_trader_amd.ops.covar import (
    corr_from_cov, cov_cpu, historical_var_cvar, is_positive_definite,
    portfolio_var,
)
from ai_crypto_trader_amd.ops.ga import (
    ga_evolve_cpu, population_diversity,
)
from ai_crypto_trader_amd.ops.indicators import NIND, indicators_cpu
from ai_crypto_trader_amd.ops.montecarlo import (
    mc_paths_cpu, philox4x32_np, philox_normal4_np, risk_stats,
)
from ai_crypto_trader_amd.backtesting.strategy import (
    PARAM_BOUNDS, random_population,
)


# --------------------------- Philox / Monte-Carlo --------------------------

def test_philox_deterministic():
    c = np.arange(16, dtype=np.uint64)
    h = np.zeros(16, dtype=np.uint64)
    a1 = philox4x32_np(42, c, h)
    a2 = philox4x32_np(42, c, h)
    for x, y in zip(a1, a2):
        np.testing.assert_array_equal(x, y)
    b = philox4x32_np(43, c, h)
    assert any((x != y).any() for x, y in zip(a1, b))


def test_philox_normals_are_standard():
    n = 200_000
    z = philox_normal4_np(
        7, np.arange(n, dtype=np.uint64), np.zeros(n, dtype=np.uint64)
    ).ravel()
    assert abs(z.mean()) < 0.01
    assert abs(z.std() - 1.0) < 0.01
    assert abs((z**3).mean()) < 0.05            # skew ~ 0
    assert abs((z**4).mean() - 3.0) < 0.1       # kurtosis ~ 3


def test_mc_cpu_matches_analytic_moments():
    A, n_steps, n_paths = 4, 16, 40_000
    rho = 0.5
    corr = np.full((A, A), rho) + (1 - rho) * np.eye(A)
    chol = np.linalg.cholesky(corr)
    mu = np.full(A, 0.10)
    sigma = np.full(A, 0.4)
    w = np.full(A, 1.0 / A)
    dt = 1.0 / 252.0
    fv, mdd = mc_paths_cpu(
        chol, mu, sigma, w, n_steps=n_steps, n_paths=n_paths, dt=dt, seed=3
    )
    horizon = n_steps * dt
    expected_mean = np.exp(mu[0] * horizon)       # E[S_t] = e^{mu t}
    assert abs(fv.mean() - expected_mean) < 0.01
    assert (mdd >= 0).all() and (mdd < 1).all()
    assert fv.min() > 0
    # portfolio vol with correlation rho: sigma_p = sigma*sqrt((1+(A-1)rho)/A)
    sig_p = sigma[0] * np.sqrt((1 + (A - 1) * rho) / A)
    lv = np.log(fv)
    assert abs(lv.std() - sig_p * np.sqrt(horizon)) < 0.01


def test_risk_stats():
    rng = np.random.default_rng(0)
    fv = (1.0 + 0.1 * rng.standard_normal(100_000)).astype(np.float32)
    s = risk_stats(fv, v0=1.0)
    assert abs(s["var_95"] - 0.1 * 1.645) < 0.01
    assert s["cvar_95"] > s["var_95"]
    assert s["var_99"] > s["var_95"]
    assert abs(s["prob_profit"] - 0.5) < 0.01
    assert s["p5"] < s["p50"] < s["p95"]


# ------------------------------- covariance --------------------------------

def test_cov_cpu_matches_numpy():
    rng = np.random.default_rng(1)
    X = rng.standard_normal((5000, 16)).astype(np.float32) * 0.01
    np.testing.assert_allclose(
        cov_cpu(X), np.cov(X.T).astype(np.float32), rtol=1e-4, atol=1e-10
    )


def test_portfolio_var_quadratic_form():
    rng = np.random.default_rng(2)
    X = rng.standard_normal((2000, 8)) * 0.01
    cov = cov_cpu(X)
    corr = corr_from_cov(cov)
    assert is_positive_definite(corr)
    values = np.full(8, 1000.0)
    vols = np.sqrt(np.diagonal(cov))
    pv = portfolio_var(values, vols, corr)
    # uncorrelated-ish assets: portfolio VaR below sum of individual VaRs
    assert 0 < pv < (values * vols * 1.645).sum()
    v, cv = historical_var_cvar(X[:, 0], 1000.0)
    assert cv >= v > 0


# ------------------------------- indicators --------------------------------

def test_indicators_constant_price():
    T = 256
    c = np.ones((1, T, 4), dtype=np.float32)
    out = indicators_cpu(c)
    last = out[0, -1]
    assert np.allclose(last[0], 1.0)     # ema12
    assert np.allclose(last[1], 1.0)     # ema26
    assert np.allclose(last[2], 0.0, atol=1e-6)      # macd
    assert np.allclose(last[6], 1.0)     # bb_mid
    assert np.allclose(last[9], 0.0, atol=1e-6)      # atr
    assert np.allclose(last[12], 1.0)    # vwap


def test_indicators_ranges(small_market):
    out = indicators_cpu(small_market)
    rsi = out[..., 5]
    stoch = out[..., 10]
    will = out[..., 11]
    assert (rsi >= 0).all() and (rsi <= 100).all()
    assert (stoch >= -1e-3).all() and (stoch <= 100.001).all()
    assert (will >= -100.001).all() and (will <= 1e-3).all()
    assert (out[..., 7] >= out[..., 6]).all()   # bb_up >= mid
    assert (out[..., 6] >= out[..., 8]).all()   # mid >= lo
    assert np.isfinite(out).all()


def test_indicators_shape(small_market):
    out = indicators_cpu(small_market)
    assert out.shape == small_market.shape[:2] + (NIND,)


# ----------------------------------- GA ------------------------------------

def test_ga_cpu_invariants():
    P = 64
    pop = random_population(P, seed=0)
    rng = np.random.default_rng(0)
    fitness = rng.standard_normal(P).astype(np.float32)
    child = ga_evolve_cpu(pop, fitness, elite_k=4, seed=1, gen=0)
    assert child.shape == pop.shape
    order = np.argsort(-fitness)
    np.testing.assert_array_equal(child[:4], pop[order[:4]])   # elitism
    lo, hi = PARAM_BOUNDS[:, 0], PARAM_BOUNDS[:, 1]
    assert (child >= lo - 1e-6).all() and (child <= hi + 1e-6).all()
    assert (child[:, 4] > child[:, 3]).all()    # ema_slow > ema_fast
    assert population_diversity(child) > 0


def test_ga_improves_on_simple_objective():
    """GA must optimize a toy objective over the param space."""
    P = 128
    pop = random_population(P, seed=2)
    target = PARAM_BOUNDS[:, 0] + 0.7 * (PARAM_BOUNDS[:, 1] - PARAM_BOUNDS[:, 0])

    def fit(p):
        return -np.abs((p - target) /
                       (PARAM_BOUNDS[:, 1] - PARAM_BOUNDS[:, 0])).mean(axis=1)

    best0 = fit(pop).max()
    for g in range(30):
        f = fit(pop).astype(np.float32)
        pop = ga_evolve_cpu(pop, f, elite_k=8, seed=5, gen=g)
    bestN = fit(pop).max()
    assert bestN > best0
    assert bestN > -0.05


def test_indicators_fast_matches_golden():
    """Vectorized serving-path indicators == the per-candle golden loop
    (same formulas; only f32-vs-f64 recurrence rounding differs)."""
    from ai_crypto_trader_amd.data.synthetic import (
        candles_chl_v, generate_ohlcv,
    )
    from ai_crypto_trader_amd.ops.indicators import (
        indicators_cpu, indicators_fast,
    )

    c = candles_chl_v(generate_ohlcv(2500, 3, seed=11, sigma=2.0))
    ref = indicators_cpu(c)
    fast = indicators_fast(c)
    np.testing.assert_allclose(fast, ref, rtol=1e-4, atol=1e-3)
