"""Per-candle tester for ARBITRARY strategy callables (reference parity:
backtesting/strategy_tester.py:156-300 + the strategy-CODE simulation of
strategy_evaluation_system.py:358-432 and ai_strategy_evaluator).

The parameterized strategy family runs as the HIP kernel (engine.py);
this tester is the slow, fully-general path: a python callable decides
per candle from a context dict (indicators + position + social), with the
same SL/TP/fee mechanics, for strategies that cannot be expressed as
parameter vectors (e.g. LLM-generated logic behind the
ai_strategy_evaluator seam). Indicators are computed for every candle
position (the reference computed them once per window and replicated —
a known flaw, strategy_tester.py:63-125 — fixed here via the indicator
engine)."""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Callable

import numpy as np

from ..ops.indicators import IND_NAMES, indicators_fast
from .engine import ANNUAL_CANDLES

DecisionFn = Callable[[dict], str]     # context -> "BUY" | "SELL" | "HOLD"


@dataclass
class TesterConfig:
    fee: float = 0.001
    position_size_pct: float = 0.5
    stop_loss_pct: float = 0.02
    take_profit_pct: float = 0.04
    warmup: int = 64


@dataclass
class Trade:
    entry_t: int
    entry_price: float
    qty: float
    exit_t: int = -1
    exit_price: float = 0.0
    pnl: float = 0.0
    reason: str = ""


@dataclass
class TesterResult:
    stats: dict
    trades: list[Trade] = field(default_factory=list)
    equity_curve: np.ndarray | None = None


class StrategyTester:
    def __init__(self, config: TesterConfig | None = None,
                 social_provider=None):
        self.cfg = config or TesterConfig()
        self.social_provider = social_provider

    def backtest_strategy(self, candles: np.ndarray,
                          decision_fn: DecisionFn,
                          symbol: str = "SYM") -> TesterResult:
        """candles: (T, 4) [close, high, low, volume] f32."""
        cfg = self.cfg
        candles = np.asarray(candles, np.float32)
        T = len(candles)
        ind = indicators_fast(candles[None])[0]         # (T, 13)

        cash, units = 1.0, 0.0
        entry_cost = 0.0
        stop = tp = 0.0
        equity = np.empty(T, np.float64)
        prev_eq = 1.0
        trades: list[Trade] = []
        open_trade: Trade | None = None

        for t in range(T):
            close, high, low, vol = (float(x) for x in candles[t])
            if open_trade is not None:
                exit_price = None
                reason = ""
                if low <= stop:
                    exit_price, reason = stop, "stop_loss"
                elif high >= tp:
                    exit_price, reason = tp, "take_profit"
                if exit_price is None and t >= cfg.warmup:
                    ctx = self._context(symbol, t, close, candles, ind,
                                        in_position=True)
                    if decision_fn(ctx) == "SELL":
                        exit_price, reason = close, "signal"
                if exit_price is not None:
                    proceeds = units * exit_price * (1 - cfg.fee)
                    cash += proceeds
                    open_trade.exit_t = t
                    open_trade.exit_price = exit_price
                    open_trade.pnl = proceeds - entry_cost
                    open_trade.reason = reason
                    trades.append(open_trade)
                    open_trade = None
                    units = 0.0
            elif t >= cfg.warmup:
                ctx = self._context(symbol, t, close, candles, ind,
                                    in_position=False)
                if decision_fn(ctx) == "BUY":
                    cost = min(cfg.position_size_pct * prev_eq, cash)
                    if cost > 1e-9:
                        units = cost * (1 - cfg.fee) / close
                        cash -= cost
                        entry_cost = cost
                        stop = close * (1 - cfg.stop_loss_pct)
                        tp = close * (1 + cfg.take_profit_pct)
                        open_trade = Trade(t, close, units)
            prev_eq = cash + units * close
            equity[t] = prev_eq

        stats = self._finalize(equity, trades, T)
        stats["symbol"] = symbol
        return TesterResult(stats, trades, equity)

    def _context(self, symbol: str, t: int, close: float,
                 candles: np.ndarray, ind: np.ndarray,
                 in_position: bool) -> dict:
        ctx = {name: float(ind[t, i]) for i, name in enumerate(IND_NAMES)}
        ctx.update({
            "symbol": symbol, "t": t, "close": close,
            "volume": float(candles[t, 3]),
            "in_position": in_position,
            "price_change_5": float(close / candles[max(t - 5, 0), 0] - 1),
        })
        if self.social_provider is not None:
            ctx["social"] = self.social_provider.at(symbol, t * 60_000)
        return ctx

    @staticmethod
    def _finalize(equity: np.ndarray, trades: list[Trade], T: int) -> dict:
        rets = np.diff(equity) / equity[:-1]
        peak = np.maximum.accumulate(equity)
        dd = (peak - equity) / peak
        pnls = [tr.pnl for tr in trades]
        wins = [p for p in pnls if p > 0]
        gl = -sum(p for p in pnls if p <= 0)
        return {
            "final_equity": float(equity[-1]),
            "total_return_pct": float((equity[-1] - 1) * 100),
            "n_trades": len(trades),
            "win_rate": len(wins) / len(trades) if trades else 0.0,
            "profit_factor":
                (sum(wins) / gl) if gl > 0 else float("inf"),
            "max_drawdown_pct": float(dd.max() * 100),
            "sharpe": float(rets.mean() / (rets.std() + 1e-12)
                            * np.sqrt(ANNUAL_CANDLES)),
        }


def rsi_threshold_strategy(oversold: float = 30.0,
                           overbought: float = 70.0) -> DecisionFn:
    """The reference's toy RSI strategy (strategy_evaluation.py:746-878)."""

    def fn(ctx: dict) -> str:
        if not ctx["in_position"] and ctx["rsi14"] < oversold:
            return "BUY"
        if ctx["in_position"] and ctx["rsi14"] > overbought:
            return "SELL"
        return "HOLD"

    return fn
