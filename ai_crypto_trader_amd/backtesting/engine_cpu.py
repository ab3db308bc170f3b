"""CPU reference backtest engine (numpy, float32).

Implements EXACTLY the per-candle state machine specified in strategy.py —
it is the golden reference the HIP backtest kernel
(ops/hip/backtest.hip) is validated against, and the engine behind the
"CPU backtest_engine (plumbing, no GPU)" baseline config.

Replaces the reference's per-candle loop (strategy_tester.py:190-300)
with deterministic parameterized strategies instead of a per-candle OpenAI
call, and computes indicators per candle position rather
than once per window (the reference's known shortcut, strategy_tester.py:63-125).

Vectorized over lanes = (param-set x symbol); loops over candles — the
same loop order as one GPU lane.
"""

from __future__ import annotations

import numpy as np

from .strategy import FEE, MAX_WIN, NPARAM, RESNAP, WARMUP

NMETRIC = 10
METRIC_NAMES = [
    "final_equity", "n_trades", "wins", "gross_profit", "gross_loss",
    "max_drawdown", "sum_ret", "sum_ret2", "sharpe", "fitness",
]

ANNUALIZE = np.float32(np.sqrt(525_600.0))   # 1m candles per year
EPS = np.float32(1e-9)


def shared_series(candles: np.ndarray):
    """Param-independent per-symbol series shared by every lane (the GPU
    kernel computes these cooperatively during tile staging — see
    strategy.py step-1 spec): rolling hmax14/lmin14 and sma20/sma50 with
    min(t+1, W) windows; f64 window sums -> f64 divide -> f32.

    candles: (nsym, T, 4) -> dict of (nsym, T) f32 arrays."""
    from numpy.lib.stride_tricks import sliding_window_view

    f32 = np.float32
    high = candles[:, :, 1]
    low = candles[:, :, 2]
    close64 = candles[:, :, 0].astype(np.float64)
    nsym, T = high.shape

    def roll_extreme(x, W, fn, pad):
        xp = np.concatenate(
            [np.full((nsym, W - 1), pad, x.dtype), x], axis=1)
        return fn(sliding_window_view(xp, W, axis=1), axis=-1)

    hmax = roll_extreme(high, 14, np.max, -np.inf).astype(f32)
    lmin = roll_extreme(low, 14, np.min, np.inf).astype(f32)

    def roll_mean(W):
        cp = np.concatenate(
            [np.zeros((nsym, W - 1), np.float64), close64], axis=1)
        s = sliding_window_view(cp, W, axis=1).sum(axis=-1)
        L = np.minimum(np.arange(T) + 1, W).astype(np.float64)
        return (s / L).astype(f32)

    return {"hmax": hmax, "lmin": lmin,
            "sma20": roll_mean(20), "sma50": roll_mean(50)}


def run_backtest_cpu(
    candles: np.ndarray,       # (nsym, T, 4) f32 [close, high, low, volume]
    population: np.ndarray,    # (P, NPARAM) f32
    *,
    initial_equity: float = 1.0,
    record_equity: bool = False,
    record_net: bool = False,
):
    """Run every param-set against every symbol.

    Returns metrics (P, nsym, NMETRIC) f32; optionally the full equity
    curves (P, nsym, T) f32 for plotting/validation.
    """
    f32 = np.float32
    candles = np.asarray(candles, dtype=f32)
    population = np.asarray(population, dtype=f32)
    assert candles.ndim == 3 and candles.shape[2] == 4
    assert population.ndim == 2 and population.shape[1] == NPARAM
    nsym, T, _ = candles.shape
    P = population.shape[0]
    L = P * nsym

    # Per-lane parameter broadcast: lane l = (param p, symbol s), s fastest.
    par = np.repeat(population, nsym, axis=0)          # (L, NPARAM)
    # reciprocal multiply (not divide) — matches the HIP kernel exactly
    inv_rsi_p = (f32(1.0) / np.maximum(np.floor(par[:, 0]), f32(1.0))).astype(f32)
    rsi_os, rsi_ob = par[:, 1], par[:, 2]
    a_f = (f32(2.0) / (par[:, 3] + f32(1.0))).astype(f32)
    a_s = (f32(2.0) / (par[:, 4] + f32(1.0))).astype(f32)
    a_sig = (f32(2.0) / (par[:, 5] + f32(1.0))).astype(f32)
    bb_w = np.clip(par[:, 6].astype(np.int32), 2, MAX_WIN)
    bb_k, bb_bth, bb_sth = par[:, 7], par[:, 8], par[:, 9]
    entry_v = par[:, 10].astype(np.int32)
    exit_v = par[:, 11].astype(np.int32)
    size_pct, sl_pct, tp_pct = par[:, 12], par[:, 13], par[:, 14]
    trail_pct, trail_act = par[:, 15], par[:, 16]
    stoch_os, stoch_ob = par[:, 17], par[:, 18]
    will_os = stoch_os - f32(100.0)
    will_ob = stoch_ob - f32(100.0)

    shared = shared_series(candles)                    # (nsym, T) each

    # Indicator state.
    ema_f = np.empty(L, f32)
    ema_s = np.empty(L, f32)
    sig = np.zeros(L, f32)
    avg_gain = np.zeros(L, f32)
    avg_loss = np.zeros(L, f32)
    ring = np.zeros((L, MAX_WIN), f32)
    # f64 rolling sums: f32 incremental sums random-walk drift over ~1M
    # candles and sum2/n - mean^2 cancels catastrophically; f32*f32 products
    # are exact in f64, so these stay exact. The HIP kernel does the same.
    bb_sum = np.zeros(L, np.float64)
    bb_sum2 = np.zeros(L, np.float64)
    inv_w = 1.0 / bb_w.astype(np.float64)
    prev_close = np.zeros(L, f32)

    # Trading state.
    cash = np.full(L, initial_equity, f32)
    units = np.zeros(L, f32)
    in_pos = np.zeros(L, bool)
    entry_cost = np.zeros(L, f32)
    entry_price = np.zeros(L, f32)
    stop = np.zeros(L, f32)
    tp = np.zeros(L, f32)
    peak = np.zeros(L, f32)
    equity = np.full(L, initial_equity, f32)
    max_eq = np.full(L, initial_equity, f32)
    max_dd = np.zeros(L, f32)
    n_trades = np.zeros(L, f32)
    wins = np.zeros(L, f32)
    gross_p = np.zeros(L, f32)
    gross_l = np.zeros(L, f32)
    sum_ret = np.zeros(L, f32)
    sum_ret2 = np.zeros(L, f32)

    curves = np.empty((L, T), f32) if record_equity else None
    nets = np.empty((L, T), np.int32) if record_net else None

    close_all = np.ascontiguousarray(candles[:, :, 0])   # (nsym, T)
    high_all = np.ascontiguousarray(candles[:, :, 1])
    low_all = np.ascontiguousarray(candles[:, :, 2])
    sym_idx = np.tile(np.arange(nsym), P)                # lane -> symbol
    lanes = np.arange(L)

    for t in range(T):
        close = close_all[sym_idx, t]
        high = high_all[sym_idx, t]
        low = low_all[sym_idx, t]

        # --- 1. indicators -------------------------------------------------
        if t == 0:
            ema_f[:] = close
            ema_s[:] = close
            change = np.zeros(L, f32)
        else:
            ema_f += a_f * (close - ema_f)
            ema_s += a_s * (close - ema_s)
            change = close - prev_close
        macd = ema_f - ema_s
        sig += a_sig * (macd - sig)
        macd_hist = macd - sig

        gain = np.maximum(change, f32(0.0))
        loss = np.maximum(-change, f32(0.0))
        avg_gain += (gain - avg_gain) * inv_rsi_p
        avg_loss += (loss - avg_loss) * inv_rsi_p
        # division-free RSI votes: rsi = 100*ag/(ag+al'), so
        # rsi < thr  <=>  100*ag < thr*(ag+al')   (al' = max(al, eps) > 0)
        rsi_num = f32(100.0) * avg_gain
        rsi_den = avg_gain + np.maximum(avg_loss, EPS)

        ridx = t % bb_w                                   # per-lane ring slot
        if t > 0 and t % RESNAP == 0:
            # drift-free Bollinger resnap (strategy.py RESNAP): recompute
            # the window sums directly, oldest->newest, instead of the
            # incremental +new-old update. Keeps the BB state exactly
            # restartable at aligned boundaries (time-parallel kernel).
            acc = np.zeros(L, np.float64)
            acc2 = np.zeros(L, np.float64)
            for j in range(MAX_WIN - 1, -1, -1):
                cj = close_all[sym_idx, t - j].astype(np.float64)
                m = j < bb_w
                acc = acc + np.where(m, cj, 0.0)
                acc2 = acc2 + np.where(m, cj * cj, 0.0)
            bb_sum = acc
            bb_sum2 = acc2
        else:
            old = ring[lanes, ridx].astype(np.float64)
            c64 = close.astype(np.float64)
            bb_sum += c64 - old
            bb_sum2 += c64 * c64 - old * old
        ring[lanes, ridx] = close
        inv_cnt = np.where(t + 1 < bb_w, 1.0 / (t + 1.0), inv_w)
        mean64 = bb_sum * inv_cnt
        var64 = np.maximum(bb_sum2 * inv_cnt - mean64 * mean64, 0.0)
        mean = mean64.astype(f32)
        std = np.sqrt(var64.astype(f32))
        band = bb_k * std
        # division-free BB votes: bb_pos = num/den with den > 0, so
        # bb_pos < thr  <=>  num < thr*den
        bb_num = close - (mean - band)
        bb_den = np.maximum(f32(2.0) * band, EPS)

        prev_close = close

        # --- 2. votes (6-indicator TradingSignal voting) -------------------
        if t >= WARMUP:
            hmax = shared["hmax"][sym_idx, t]
            lmin = shared["lmin"][sym_idx, t]
            sma20 = shared["sma20"][sym_idx, t]
            sma50 = shared["sma50"][sym_idx, t]
            # stoch/will division-free: num/den vs thr <=> num vs thr*den
            srange = np.maximum(hmax - lmin, EPS)
            st_num = f32(100.0) * (close - lmin)
            wl_num = f32(-100.0) * (hmax - close)
            buy = (
                (rsi_num < rsi_os * rsi_den).astype(np.int32)
                + (macd_hist > 0).astype(np.int32)
                + (bb_num < bb_bth * bb_den).astype(np.int32)
                + (st_num < stoch_os * srange).astype(np.int32)
                + (wl_num < will_os * srange).astype(np.int32)
                + ((close > sma20) & (sma20 > sma50)).astype(np.int32)
            )
            sell = (
                (rsi_num > rsi_ob * rsi_den).astype(np.int32)
                + (macd_hist < 0).astype(np.int32)
                + (bb_num > bb_sth * bb_den).astype(np.int32)
                + (st_num > stoch_ob * srange).astype(np.int32)
                + (wl_num > will_ob * srange).astype(np.int32)
                + ((close < sma20) & (sma20 < sma50)).astype(np.int32)
            )
            net = buy - sell
        else:
            net = np.zeros(L, np.int32)
        if record_net:
            nets[:, t] = net

        # --- 3. position management ---------------------------------------
        pos = in_pos
        peak = np.where(pos, np.maximum(peak, high), peak)
        trail_on = pos & (trail_pct > 0) & (
            peak >= entry_price * (f32(1.0) + trail_act)
        )
        stop = np.where(
            trail_on, np.maximum(stop, peak * (f32(1.0) - trail_pct)), stop
        )

        hit_sl = pos & (low <= stop)
        hit_tp = pos & ~hit_sl & (high >= tp)
        hit_sig = pos & ~hit_sl & ~hit_tp & (net <= -exit_v)
        exiting = hit_sl | hit_tp | hit_sig
        exit_price = np.where(hit_sl, stop, np.where(hit_tp, tp, close))

        proceeds = units * exit_price * (f32(1.0) - f32(FEE))
        pnl = proceeds - entry_cost
        cash = np.where(exiting, cash + proceeds, cash)
        n_trades += exiting
        wins += exiting & (pnl > 0)
        gross_p += np.where(exiting, np.maximum(pnl, f32(0.0)), f32(0.0))
        gross_l += np.where(exiting, np.maximum(-pnl, f32(0.0)), f32(0.0))
        units = np.where(exiting, f32(0.0), units)
        in_pos = pos & ~exiting

        entering = (~pos) & (t >= WARMUP) & (net >= entry_v)
        cost = np.minimum(size_pct * equity, cash)
        new_units = cost * (f32(1.0) - f32(FEE)) / close
        cash = np.where(entering, cash - cost, cash)
        units = np.where(entering, new_units, units)
        entry_cost = np.where(entering, cost, entry_cost)
        entry_price = np.where(entering, close, entry_price)
        stop = np.where(entering, close * (f32(1.0) - sl_pct), stop)
        tp = np.where(entering, close * (f32(1.0) + tp_pct), tp)
        peak = np.where(entering, close, peak)
        in_pos = in_pos | entering

        # --- 4. mark to market --------------------------------------------
        new_eq = cash + units * close
        r = new_eq / equity - f32(1.0)
        sum_ret += r
        sum_ret2 += r * r
        equity = new_eq
        max_eq = np.maximum(max_eq, equity)
        max_dd = np.maximum(max_dd, (max_eq - equity) / max_eq)
        if record_equity:
            curves[:, t] = equity

    metrics = finalize_metrics(
        T, equity, n_trades, wins, gross_p, gross_l, max_dd, sum_ret, sum_ret2
    ).reshape(P, nsym, NMETRIC)
    extras = []
    if record_equity:
        extras.append(curves.reshape(P, nsym, T))
    if record_net:
        extras.append(nets.reshape(P, nsym, T))
    if extras:
        return (metrics, *extras)
    return metrics


def run_trades_from_flags_cpu(
    candles: np.ndarray,       # (nsym, T, 4) f32
    population: np.ndarray,    # (P, NPARAM) f32
    entry_flags: np.ndarray,   # (P, nsym, T) bool  (net >= entry_votes, t>=WARMUP)
    exit_flags: np.ndarray,    # (P, nsym, T) bool  (net <= -exit_votes)
    *,
    initial_equity: float = 1.0,
) -> np.ndarray:
    """Numpy twin of the bt_trades phase (ops/hip/backtest_tp.hip):
    the exact position state machine + equity accounting driven by
    precomputed vote flags. Identical op order to run_backtest_cpu
    sections 3-4, so given flags extracted from run_backtest_cpu
    (record_net=True) the metrics are bitwise equal — the decomposition
    property the time-parallel GPU path rests on
    (tests/test_backtest_cpu.py::test_flag_decomposition_bitwise)."""
    f32 = np.float32
    candles = np.asarray(candles, dtype=f32)
    population = np.asarray(population, dtype=f32)
    nsym, T, _ = candles.shape
    P = population.shape[0]
    L = P * nsym

    par = np.repeat(population, nsym, axis=0)
    size_pct, sl_pct, tp_pct = par[:, 12], par[:, 13], par[:, 14]
    trail_pct, trail_act = par[:, 15], par[:, 16]

    eflags = entry_flags.reshape(L, T)
    xflags = exit_flags.reshape(L, T)

    cash = np.full(L, initial_equity, f32)
    units = np.zeros(L, f32)
    in_pos = np.zeros(L, bool)
    entry_cost = np.zeros(L, f32)
    entry_price = np.zeros(L, f32)
    stop = np.zeros(L, f32)
    tp = np.zeros(L, f32)
    peak = np.zeros(L, f32)
    equity = np.full(L, initial_equity, f32)
    max_eq = np.full(L, initial_equity, f32)
    max_dd = np.zeros(L, f32)
    n_trades = np.zeros(L, f32)
    wins = np.zeros(L, f32)
    gross_p = np.zeros(L, f32)
    gross_l = np.zeros(L, f32)
    sum_ret = np.zeros(L, f32)
    sum_ret2 = np.zeros(L, f32)

    close_all = np.ascontiguousarray(candles[:, :, 0])
    high_all = np.ascontiguousarray(candles[:, :, 1])
    low_all = np.ascontiguousarray(candles[:, :, 2])
    sym_idx = np.tile(np.arange(nsym), P)

    for t in range(T):
        close = close_all[sym_idx, t]
        high = high_all[sym_idx, t]
        low = low_all[sym_idx, t]

        pos = in_pos
        peak = np.where(pos, np.maximum(peak, high), peak)
        trail_on = pos & (trail_pct > 0) & (
            peak >= entry_price * (f32(1.0) + trail_act)
        )
        stop = np.where(
            trail_on, np.maximum(stop, peak * (f32(1.0) - trail_pct)), stop
        )

        hit_sl = pos & (low <= stop)
        hit_tp = pos & ~hit_sl & (high >= tp)
        hit_sig = pos & ~hit_sl & ~hit_tp & xflags[:, t]
        exiting = hit_sl | hit_tp | hit_sig
        exit_price = np.where(hit_sl, stop, np.where(hit_tp, tp, close))

        proceeds = units * exit_price * (f32(1.0) - f32(FEE))
        pnl = proceeds - entry_cost
        cash = np.where(exiting, cash + proceeds, cash)
        n_trades += exiting
        wins += exiting & (pnl > 0)
        gross_p += np.where(exiting, np.maximum(pnl, f32(0.0)), f32(0.0))
        gross_l += np.where(exiting, np.maximum(-pnl, f32(0.0)), f32(0.0))
        units = np.where(exiting, f32(0.0), units)
        in_pos = pos & ~exiting

        entering = (~pos) & eflags[:, t]
        cost = np.minimum(size_pct * equity, cash)
        new_units = cost * (f32(1.0) - f32(FEE)) / close
        cash = np.where(entering, cash - cost, cash)
        units = np.where(entering, new_units, units)
        entry_cost = np.where(entering, cost, entry_cost)
        entry_price = np.where(entering, close, entry_price)
        stop = np.where(entering, close * (f32(1.0) - sl_pct), stop)
        tp = np.where(entering, close * (f32(1.0) + tp_pct), tp)
        peak = np.where(entering, close, peak)
        in_pos = in_pos | entering

        new_eq = cash + units * close
        r = new_eq / equity - f32(1.0)
        sum_ret += r
        sum_ret2 += r * r
        equity = new_eq
        max_eq = np.maximum(max_eq, equity)
        max_dd = np.maximum(max_dd, (max_eq - equity) / max_eq)

    return finalize_metrics(
        T, equity, n_trades, wins, gross_p, gross_l, max_dd, sum_ret,
        sum_ret2
    ).reshape(P, nsym, NMETRIC)


def finalize_metrics(
    T, equity, n_trades, wins, gross_p, gross_l, max_dd, sum_ret, sum_ret2
) -> np.ndarray:
    """Shared final-stats formula (strategy_evaluation.py:32-228 semantics:
    annualized Sharpe from per-candle returns, win-rate, profit factor via
    gross sums, running-peak drawdown). Fitness mirrors the GA objective of
    strategy_evolution_service.py:525-694 (sharpe + win-rate - drawdown
    penalty; no-trade strategies are penalized)."""
    f32 = np.float32
    n = f32(max(T, 1))
    mean = sum_ret / n
    var = np.maximum(sum_ret2 / n - mean * mean, f32(0.0))
    sharpe = mean / np.maximum(np.sqrt(var), EPS) * ANNUALIZE
    sharpe = np.where(n_trades > 0, sharpe, f32(0.0))
    win_rate = wins / np.maximum(n_trades, f32(1.0))
    fitness = np.where(
        n_trades > 0,
        sharpe + win_rate - f32(2.0) * max_dd,
        f32(-1.0),
    )
    return np.stack(
        [equity, n_trades, wins, gross_p, gross_l, max_dd, sum_ret,
         sum_ret2, sharpe.astype(f32), fitness.astype(f32)], axis=-1
    ).astype(f32)
