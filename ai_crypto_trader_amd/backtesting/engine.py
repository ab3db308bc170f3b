"""Backtest orchestrator (reference parity: backtesting/backtest_engine.py
:14-325 + strategy_tester.py:17-487).

run_backtest() fetches-if-missing then runs the per-candle engine —
the CPU reference engine for small runs and the HIP backtest kernel on
GPU (one lane per param-set x symbol) — and produces the final-stats dict
(strategy_tester.py:403-430 formulas: total return, win rate, profit
factor, annualized Sharpe, max drawdown). Multi-symbol x param sweeps and
GA-based parameter optimization ride the same kernels (SURVEY.md §3.2:
the reference's per-candle OpenAI call is replaced by deterministic
parameterized strategies, which is what makes the loop a pure kernel)."""

from __future__ import annotations

import json
import time
from pathlib import Path

import numpy as np

from ..backtesting.engine_cpu import (
    METRIC_NAMES, run_backtest_cpu,
)
from ..backtesting.strategy import (
    clip_params, dict_to_params, params_to_dict,
)
from ..ops import gpu_available
from .data_manager import HistoricalDataManager

ANNUAL_CANDLES = 525_600.0      # 1m bars

# bars per year by interval (interval-aware Sharpe annualization)
INTERVAL_BARS_PER_YEAR = {
    "1m": 525_600.0, "3m": 175_200.0, "5m": 105_120.0,
    "15m": 35_040.0, "30m": 17_520.0, "1h": 8_760.0,
    "4h": 2_190.0, "1d": 365.0,
}


# Named strategy presets (the reference ships dca/grid/threshold bots as
# separate services; in backtests they are parameter presets of the same
# state machine — dca_strategy.py:347-741 / grid semantics approximated by
# vote cadence + sizing).
STRATEGY_PRESETS = {
    "default": {},
    "dca_strategy": {
        "entry_votes": 1, "exit_votes": 3, "position_size_pct": 0.1,
        "stop_loss_pct": 0.2, "take_profit_pct": 0.4,
        "trailing_stop_pct": 0.0,
    },
    "momentum": {
        "entry_votes": 2, "exit_votes": 2, "position_size_pct": 0.5,
        "stop_loss_pct": 0.02, "take_profit_pct": 0.05,
        "trailing_stop_pct": 0.01, "trailing_act_pct": 0.02,
    },
    "mean_reversion": {
        "rsi_oversold": 25.0, "rsi_overbought": 75.0, "bb_buy_th": 0.02,
        "bb_sell_th": 0.98, "entry_votes": 2, "exit_votes": 1,
        "take_profit_pct": 0.02, "stop_loss_pct": 0.03,
    },
    "conservative": {
        "entry_votes": 3, "exit_votes": 1, "position_size_pct": 0.2,
        "stop_loss_pct": 0.01, "take_profit_pct": 0.03,
    },
}


def metrics_to_stats(m: np.ndarray, T: int, interval: str = "1m") -> dict:
    """One lane's metric vector -> result dict
    (strategy_tester.py:403-430 / strategy_evaluation.py:32-228).
    Sharpe is recomputed HOST-SIDE from the kernel's raw return moments
    (sum_ret/sum_ret2) with the interval's bars-per-year, so non-1m
    backtests annualize correctly (the in-kernel sharpe/fitness are
    1m-annualized — the GA fitness convention)."""
    d = dict(zip(METRIC_NAMES, (float(x) for x in m)))
    gp, gl = d["gross_profit"], d["gross_loss"]
    bars_py = INTERVAL_BARS_PER_YEAR.get(interval, ANNUAL_CANDLES)
    years = T / bars_py
    n = max(T, 1)
    mean_r = d["sum_ret"] / n
    var_r = max(d["sum_ret2"] / n - mean_r * mean_r, 0.0)
    sharpe = (mean_r / max(np.sqrt(var_r), 1e-9) * np.sqrt(bars_py)
              if d["n_trades"] > 0 else 0.0)
    d["sharpe"] = float(sharpe)
    return {
        "final_equity": d["final_equity"],
        "total_return_pct": (d["final_equity"] - 1.0) * 100.0,
        "annualized_return_pct":
            ((d["final_equity"] ** (1 / max(years, 1e-9))) - 1.0) * 100.0
            if d["final_equity"] > 0 else -100.0,
        "n_trades": int(d["n_trades"]),
        "win_rate": d["wins"] / max(d["n_trades"], 1.0),
        # capped: float('inf') serializes as the non-standard 'Infinity'
        # token that strict JSON parsers (and dashboards) reject
        "profit_factor": min(gp / gl, 1e9) if gl > 0 else (
            1e9 if gp > 0 else 0.0),
        "max_drawdown_pct": d["max_drawdown"] * 100.0,
        "sharpe": d["sharpe"],
        "fitness": d["fitness"],
    }


class BacktestEngine:
    def __init__(self, data_dir: str = "backtesting_data",
                 device: str | None = None):
        self.dm = HistoricalDataManager(data_dir)
        if device is None:
            device = "cuda" if gpu_available() else "cpu"
        self.device = device
        self.results_dir = Path(data_dir) / "results"
        self.results_dir.mkdir(parents=True, exist_ok=True)

    # --- single run ------------------------------------------------------
    def run_backtest(self, symbol: str, strategy: str = "default",
                     interval: str = "1m", n_candles: int = 10_000,
                     params: dict | None = None,
                     record_equity: bool = False) -> dict:
        df = self.dm.load_market_data(symbol, interval,
                                      n_candles=n_candles)
        candles = self.dm.to_chlv(df.iloc[:n_candles])
        preset = dict(STRATEGY_PRESETS.get(strategy, {}))
        if params:
            preset.update(params)
        vec = clip_params(dict_to_params(preset)[None])
        t0 = time.perf_counter()
        out = self._run(candles, vec, record_equity)
        elapsed = time.perf_counter() - t0
        metrics = out[0] if record_equity else out
        T = candles.shape[1]
        stats = metrics_to_stats(metrics[0, 0], T, interval)
        stats.update({
            "symbol": symbol, "strategy": strategy, "interval": interval,
            "n_candles": T, "engine": self.device,
            "candles_per_sec": T / max(elapsed, 1e-9),
            "params": params_to_dict(vec[0]),
        })
        if record_equity:
            stats["equity_curve"] = out[1][0, 0].tolist()
        self._save(stats)
        return stats

    def _run(self, candles: np.ndarray, pop: np.ndarray,
             record_equity: bool = False):
        if self.device.startswith("cuda") and not record_equity:
            import torch

            from ..ops.backtest import run_backtest_gpu
            c = torch.from_numpy(candles).to(self.device)
            p = torch.from_numpy(pop).to(self.device)
            m = run_backtest_gpu(c, p)
            torch.cuda.synchronize()
            return m.cpu().numpy()
        return run_backtest_cpu(candles, pop,
                                record_equity=record_equity)

    # --- sweeps (reference :127-178) -------------------------------------
    def run_multiple_backtests(self, symbols: list[str],
                               strategies: list[str],
                               n_candles: int = 10_000) -> dict:
        results = []
        for sym in symbols:
            for strat in strategies:
                results.append(self.run_backtest(sym, strat,
                                                 n_candles=n_candles))
        best = max(results, key=lambda r: r["sharpe"])
        return {
            "results": results,
            "best": {"symbol": best["symbol"], "strategy": best["strategy"],
                     "sharpe": best["sharpe"]},
        }

    # --- GA parameter optimization over the GPU kernel -------------------
    def optimize(self, symbol: str, pop_size: int = 256,
                 generations: int = 10, n_candles: int = 20_000,
                 seed: int = 0) -> dict:
        from .ga_engine import GAEngine

        df = self.dm.load_market_data(symbol, n_candles=n_candles)
        candles = self.dm.to_chlv(df.iloc[:n_candles])
        eng = GAEngine(candles, pop_per_rank=pop_size, device=self.device,
                       seed=seed, segments="auto")
        t0 = time.perf_counter()
        history = []
        for _ in range(generations):
            eng.step()
            history.append(float(eng.last_fitness_global.max()))
        eng.eval_fitness()
        elapsed = time.perf_counter() - t0
        fit, best = eng.best()
        stats = self.run_backtest(symbol, "optimized",
                                  params=params_to_dict(best),
                                  n_candles=n_candles)
        stats["optimize"] = {
            "pop_size": pop_size, "generations": generations,
            "best_fitness": fit, "fitness_history": history,
            "candle_evals_per_sec":
                eng.candle_evals_per_step * generations / elapsed,
        }
        return stats

    def _save(self, stats: dict):
        name = (f"{stats['symbol']}_{stats['strategy']}_"
                f"{int(time.time())}.json")
        slim = {k: v for k, v in stats.items() if k != "equity_curve"}
        with open(self.results_dir / name, "w") as f:
            json.dump(slim, f, indent=2)

    def list_results(self) -> list[dict]:
        out = []
        for p in sorted(self.results_dir.glob("*.json")):
            with open(p) as f:
                out.append(json.load(f))
        return out
