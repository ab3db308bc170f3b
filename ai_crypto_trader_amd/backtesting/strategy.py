"""Parameterized trading strategies — the shared CPU/GPU contract.

The reference drives its backtest with per-candle indicator votes
(binance_ml_strategy.py:489-543 TradingSignal), position sizing
(binance_ml_strategy.py:251-291), SL/TP/trailing-stop management
(trade_executor_service.py:55-399 TrailingStopManager) and a per-candle
loop (backtesting/strategy_tester.py:190-300). Here all of that is folded
into one deterministic parameterized state machine so that a whole GA
population of strategies can march through candles as independent GPU
lanes.

A strategy is a flat float32 vector with the layout below. The same layout
is consumed by the numpy reference engine (backtesting/engine_cpu.py) and
the HIP backtest kernel (ops/hip/backtest.hip) — keep the three in sync.

Param vector layout (NPARAM float32 slots):
  0  rsi_period        Wilder RSI period            [2, 64]   (rounded to int)
  1  rsi_oversold      buy-vote threshold           [5, 50]
  2  rsi_overbought    sell-vote threshold          [50, 95]
  3  ema_fast          MACD fast EMA period         [2, 32]   (int)
  4  ema_slow          MACD slow EMA period         [4, 64]   (int, > fast)
  5  macd_signal       MACD signal EMA period       [2, 32]   (int)
  6  bb_window         Bollinger window             [4, 32]   (int, <= MAX_WIN)
  7  bb_k              Bollinger band width (sigmas)[0.5, 4.0]
  8  bb_buy_th         bb position buy threshold    [0.0, 0.5]
  9  bb_sell_th        bb position sell threshold   [0.5, 1.0]
  10 entry_votes       net buy votes to enter       [1, 3]    (int)
  11 exit_votes        net sell votes to exit       [1, 3]    (int)
  12 position_size_pct fraction of equity per trade [0.05, 1.0]
  13 stop_loss_pct     stop-loss distance           [0.002, 0.2]
  14 take_profit_pct   take-profit distance         [0.004, 0.4]
  15 trailing_stop_pct trailing distance (0 = off)  [0.0, 0.1]
  16 trailing_act_pct  gain before trailing arms    [0.0, 0.1]
  17 stoch_os          stochastic oversold          [5, 45]
  18 stoch_ob          stochastic overbought        [55, 95]
                       (Williams %R thresholds are tied: stoch_os-100 /
                        stoch_ob-100, the reference's -80/-20 at 20/80)

Per-candle semantics (both engines implement EXACTLY this order):
  1. update indicators with candle t (EMA fast/slow, MACD + signal EMA,
     Wilder RSI with zero-init recurrence, Bollinger rolling sum/sumsq
     ring). Param-INDEPENDENT series are shared by every lane of a symbol
     and precomputed once (GPU: cooperatively during tile staging):
       hmax14/lmin14 = rolling max(high)/min(low) over min(t+1, 14)
       sma20/sma50   = rolling close means over min(t+1, 20/50)
                       (f64 window sums, f64 divide, cast f32)
       stoch = 100*(close-lmin14)/max(hmax14-lmin14, eps)
       will  = -100*(hmax14-close)/max(hmax14-lmin14, eps)
  2. votes (only when t >= WARMUP) — the reference's 6-indicator
     TradingSignal voting (binance_ml_strategy.py:489-543):
       buy  += rsi < rsi_oversold;     sell += rsi > rsi_overbought
       buy  += macd_hist > 0;          sell += macd_hist < 0
       buy  += bb_pos < bb_buy_th;     sell += bb_pos > bb_sell_th
       buy  += stoch < stoch_os;       sell += stoch > stoch_ob
       buy  += will < stoch_os - 100;  sell += will > stoch_ob - 100
       buy  += close > sma20 > sma50;  sell += close < sma20 < sma50
       net = buy - sell
  3. if in position:
       peak = max(peak, high)
       if trailing_stop_pct > 0 and peak >= entry*(1+trailing_act_pct):
           stop = max(stop, peak*(1-trailing_stop_pct))
       if low <= stop:           exit at stop  (stop fill, pessimistic first)
       elif high >= tp_price:    exit at tp
       elif net <= -exit_votes:  exit at close
     elif t >= WARMUP and net >= entry_votes:  enter long at close:
       cost  = position_size_pct * equity  (cash-capped)
       units = cost*(1-FEE)/close ; stop = close*(1-stop_loss_pct)
       tp    = close*(1+take_profit_pct) ; peak = close
  4. equity = cash + units*close ; log-return stats, running max drawdown

Prices are normalized (close[0] == 1.0) before either engine runs so fp32
rolling sums stay well-conditioned on GPU (see data/synthetic.py).
"""

from __future__ import annotations

import numpy as np

NPARAM = 19
MAX_WIN = 32           # max Bollinger window — sized to the kernel's LDS ring
SHARED_HALO = 64       # kernel halo: covers bb<=32 and sma50 windows
WARMUP = 128           # candles before the first vote (covers 3x max period)
FEE = 0.001            # taker fee per side (reference: strategy_tester.py 0.1%)
# Every RESNAP candles (t % RESNAP == 0, t > 0) the f64 Bollinger rolling
# sums are recomputed directly from the window (oldest->newest add order)
# instead of incrementally. This makes the BB state *exactly restartable*
# at any RESNAP-aligned boundary — the property the time-parallel GPU
# backtest (ops/hip/backtest_tp.hip) relies on to split the time axis
# across waves while staying bit-identical to the sequential engines.
# Multiple of the kernel tile (256) and LONGER than a seg-64 fitness
# segment (1M/64 = 15625), so the segmented headline path never
# resnaps and dispatches the resnap-free kernel instantiation (same
# codegen/occupancy as if the feature didn't exist). Cost on the
# continuous path: ~2x32 f64 adds per 16384 candles — noise.
RESNAP = 16_384

PARAM_NAMES = [
    "rsi_period", "rsi_oversold", "rsi_overbought",
    "ema_fast", "ema_slow", "macd_signal",
    "bb_window", "bb_k", "bb_buy_th", "bb_sell_th",
    "entry_votes", "exit_votes",
    "position_size_pct", "stop_loss_pct", "take_profit_pct",
    "trailing_stop_pct", "trailing_act_pct",
    "stoch_os", "stoch_ob",
]

# (low, high, is_int) bounds per slot — used by the GA and random init.
PARAM_BOUNDS = np.array([
    (2, 64, 1), (5, 50, 0), (50, 95, 0),
    (2, 32, 1), (4, 64, 1), (2, 32, 1),
    (4, 32, 1), (0.5, 4.0, 0), (0.0, 0.5, 0), (0.5, 1.0, 0),
    (1, 4, 1), (1, 4, 1),
    (0.05, 1.0, 0), (0.002, 0.2, 0), (0.004, 0.4, 0),
    (0.0, 0.1, 0), (0.0, 0.1, 0),
    (5, 45, 0), (55, 95, 0),
], dtype=np.float32)

# Named defaults ≈ the reference's dca/threshold strategy parameter block
# (config.json trading defaults: RSI 14/30/70, MACD 12/26/9, BB 20/2,
# Stoch 20/80).
DEFAULT_PARAMS = np.array([
    14, 30, 70,
    12, 26, 9,
    20, 2.0, 0.05, 0.95,
    2, 2,
    0.5, 0.02, 0.04,
    0.0, 0.01,
    20, 80,
], dtype=np.float32)


def params_to_dict(vec: np.ndarray) -> dict:
    return {name: float(v) for name, v in zip(PARAM_NAMES, vec)}


def dict_to_params(d: dict) -> np.ndarray:
    vec = DEFAULT_PARAMS.copy()
    for i, name in enumerate(PARAM_NAMES):
        if name in d:
            vec[i] = d[name]
    return vec


def clip_params(pop: np.ndarray) -> np.ndarray:
    """Clip a (pop, NPARAM) array into bounds; round integer slots."""
    lo = PARAM_BOUNDS[:, 0]
    hi = PARAM_BOUNDS[:, 1]
    is_int = PARAM_BOUNDS[:, 2] > 0
    out = np.clip(pop, lo, hi)
    out[:, is_int] = np.rint(out[:, is_int])
    # keep ema_slow > ema_fast
    bad = out[:, 4] <= out[:, 3]
    out[bad, 4] = out[bad, 3] + 1
    return out.astype(np.float32)


def random_population(pop_size: int, seed: int = 0) -> np.ndarray:
    """Seeded uniform random population within bounds, (pop, NPARAM) f32."""
    rng = np.random.default_rng(seed)
    lo = PARAM_BOUNDS[:, 0]
    hi = PARAM_BOUNDS[:, 1]
    pop = rng.uniform(lo, hi, size=(pop_size, NPARAM)).astype(np.float32)
    return clip_params(pop)
