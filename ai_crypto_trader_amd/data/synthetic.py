"""Deterministic synthetic OHLCV generation (seeded GBM).

The reference has no offline data fixtures — its tests hit live Binance
(SURVEY.md §4). Here synthetic candles are first-class: the same seeded
GBM math as the Monte-Carlo kernel (monte_carlo_service.py:264-273
semantics: S_t = S_{t-1} * exp((mu - sigma^2/2) dt + sigma sqrt(dt) Z)),
used for backtests, GA fitness, RL environments and benchmarks.

Prices are returned NORMALIZED (close[0] == 1.0): strategies are
scale-invariant and normalized prices keep fp32 rolling sums on the GPU
well-conditioned (strategy.py docstring).
"""

from __future__ import annotations

import numpy as np

OHLCV_FIELDS = ("open", "high", "low", "close", "volume")


def generate_ohlcv(
    n_candles: int,
    n_symbols: int = 1,
    *,
    seed: int = 0,
    mu: float = 0.10,
    sigma: float = 0.60,
    dt: float = 1.0 / 525_600.0,   # one 1m candle in years
    base_volume: float = 1.0,
    dtype=np.float32,
) -> np.ndarray:
    """Generate (n_symbols, n_candles, 5) OHLCV, close[sym, 0] ~= 1.0.

    GBM closes; open = previous close; high/low bracket open/close with a
    half-range drawn from |N(0, sigma*sqrt(dt))|; volume lognormal around
    base_volume. Fully determined by `seed`.
    """
    rng = np.random.default_rng(seed)
    z = rng.standard_normal((n_symbols, n_candles))
    step = (mu - 0.5 * sigma * sigma) * dt + sigma * np.sqrt(dt) * z
    logp = np.cumsum(step, axis=1)
    close = np.exp(logp - logp[:, :1])          # normalize: close[:,0]==1.0
    open_ = np.empty_like(close)
    open_[:, 0] = 1.0
    open_[:, 1:] = close[:, :-1]
    wick = np.abs(rng.standard_normal((n_symbols, n_candles))) * (
        sigma * np.sqrt(dt)
    )
    hi = np.maximum(open_, close) * (1.0 + wick)
    lo = np.minimum(open_, close) * (1.0 - wick)
    vol = base_volume * np.exp(
        0.5 * rng.standard_normal((n_symbols, n_candles))
    )
    out = np.stack([open_, hi, lo, close, vol], axis=-1).astype(dtype)
    return out


def candles_chl_v(ohlcv: np.ndarray) -> np.ndarray:
    """Pack (sym, T, 5) OHLCV into the (sym, T, 4) [close, high, low, volume]
    float32 layout the backtest engines consume (contiguous per symbol)."""
    out = np.stack(
        [ohlcv[..., 3], ohlcv[..., 1], ohlcv[..., 2], ohlcv[..., 4]], axis=-1
    )
    return np.ascontiguousarray(out, dtype=np.float32)


def generate_regime_ohlcv(
    n_candles: int,
    n_symbols: int = 1,
    *,
    seed: int = 0,
    episode: int = 2000,
    dtype=np.float32,
) -> np.ndarray:
    """Regime-switching GBM market: episodes cycle through bull / calm /
    bear / volatile (mu, sigma annualized) in a per-symbol phase-shifted
    order. Produces the sustained trends and volatility bursts a CALM
    single-(mu,sigma) GBM lacks — the live-soak market for exercising the
    DEFAULT confidence gate (0.7): the analyzer's momentum/trend/oscillator
    factors only line up strongly enough during real episodes.
    Deterministic in `seed`."""
    regimes = [(3.0, 1.2), (0.1, 0.5), (-3.0, 1.5), (0.0, 2.5)]
    rng = np.random.default_rng(seed)
    dt = 1.0 / 525_600.0
    z = rng.standard_normal((n_symbols, n_candles))
    mu = np.empty((n_symbols, n_candles))
    sigma = np.empty((n_symbols, n_candles))
    for s in range(n_symbols):
        for e in range(0, n_candles, episode):
            m, sg = regimes[(e // episode + s) % len(regimes)]
            mu[s, e:e + episode] = m
            sigma[s, e:e + episode] = sg
    step = (mu - 0.5 * sigma * sigma) * dt + sigma * np.sqrt(dt) * z
    logp = np.cumsum(step, axis=1)
    close = np.exp(logp - logp[:, :1])
    open_ = np.empty_like(close)
    open_[:, 0] = 1.0
    open_[:, 1:] = close[:, :-1]
    wick = np.abs(rng.standard_normal((n_symbols, n_candles))) * (
        sigma * np.sqrt(dt))
    hi = np.maximum(open_, close) * (1.0 + wick)
    lo = np.minimum(open_, close) / (1.0 + wick)
    vol = np.exp(rng.standard_normal((n_symbols, n_candles)) * 0.3)
    return np.stack([open_, hi, lo, close, vol], axis=-1).astype(dtype)
