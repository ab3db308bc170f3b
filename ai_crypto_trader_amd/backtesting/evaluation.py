"""Strategy evaluation engine (reference parity:
services/strategy_evaluation.py:32-2277 + strategy_evaluation_system.py).

calculate_metrics (:32-228): win rate, profit factor, equity curve with
running-peak drawdown, daily buckets, annualized Sharpe.
calculate_advanced_metrics (:231-319): Calmar, Sortino, streaks, recovery
factor, expectancy.
Time-series k-fold cross-validation (:635) over the GPU/CPU backtest
engines, per-regime buckets (strategy_evaluation_system.py:587),
synthetic market-condition generators (:1197-1297), multi-strategy
comparison + radar/bar viz (:1439-2276, matplotlib-Agg)."""

from __future__ import annotations

import numpy as np

from ..backtesting.engine_cpu import run_backtest_cpu
from ..backtesting.strategy import clip_params, dict_to_params
from ..data.synthetic import candles_chl_v, generate_ohlcv
from .engine import metrics_to_stats

ANNUAL = 525_600.0


def calculate_metrics(equity_curve: np.ndarray,
                      trades: list[dict] | None = None) -> dict:
    """Metrics from an equity curve (+ optional trade list)
    (strategy_evaluation.py:32-228)."""
    eq = np.asarray(equity_curve, np.float64)
    rets = np.diff(eq) / eq[:-1]
    peak = np.maximum.accumulate(eq)
    dd = (peak - eq) / peak
    sharpe = float(rets.mean() / (rets.std() + 1e-12) * np.sqrt(ANNUAL))
    out = {
        "total_return_pct": float((eq[-1] / eq[0] - 1) * 100),
        "sharpe": sharpe,
        "max_drawdown_pct": float(dd.max() * 100),
        "volatility_ann": float(rets.std() * np.sqrt(ANNUAL)),
        "n_candles": len(eq),
        "daily": period_returns(eq, 1440),
        "monthly": period_returns(eq, 43_200),
    }
    if trades:
        pnls = [t["pnl"] for t in trades]
        wins = [p for p in pnls if p > 0]
        losses = [p for p in pnls if p <= 0]
        gp, gl = sum(wins), -sum(losses)
        out.update({
            "n_trades": len(trades),
            "win_rate": len(wins) / len(trades),
            "profit_factor": gp / gl if gl > 0 else float("inf"),
            "avg_win": float(np.mean(wins)) if wins else 0.0,
            "avg_loss": float(np.mean(losses)) if losses else 0.0,
        })
    return out


def period_returns(equity_curve: np.ndarray,
                   candles_per_period: int = 1440) -> dict:
    """Per-period (daily at 1440 1m-candles, monthly at 43200) return
    buckets (strategy_evaluation.py:180-188): the return of each complete
    period plus best/worst/positive-share summaries."""
    eq = np.asarray(equity_curve, np.float64)
    k = max(int(candles_per_period), 1)
    n = (len(eq) - 1) // k
    if n < 1:
        return {"returns_pct": [], "best_pct": 0.0, "worst_pct": 0.0,
                "positive_share": 0.0, "n_periods": 0}
    marks = eq[:n * k + 1:k]
    rets = (marks[1:] / marks[:-1] - 1.0) * 100.0
    return {
        "returns_pct": [round(float(r), 6) for r in rets],
        "best_pct": float(rets.max()),
        "worst_pct": float(rets.min()),
        "positive_share": float((rets > 0).mean()),
        "n_periods": int(n),
    }


def calculate_advanced_metrics(equity_curve: np.ndarray,
                               trades: list[dict] | None = None) -> dict:
    """Calmar, Sortino, streaks, recovery factor, expectancy (:231-319)."""
    eq = np.asarray(equity_curve, np.float64)
    rets = np.diff(eq) / eq[:-1]
    peak = np.maximum.accumulate(eq)
    mdd = float(((peak - eq) / peak).max())
    years = len(eq) / ANNUAL
    # log-space exponentiation: the naive ratio**(1/years) overflows to
    # inf for short curves (years ~ 1e-6); clamp the annualized log-return
    # so calmar stays finite and JSON-serializable
    ratio = max(eq[-1] / eq[0], 1e-12)
    ann_ret = float(np.exp(np.clip(np.log(ratio) / max(years, 1e-9),
                                   -50.0, 50.0))) - 1
    downside = rets[rets < 0]
    # finite caps instead of inf: keeps every metric JSON-serializable
    sortino = min(float(rets.mean() / (downside.std() + 1e-12)
                        * np.sqrt(ANNUAL)), 1e9) if len(downside) else 1e9
    out = {
        "calmar": min(float(ann_ret / mdd), 1e9) if mdd > 0 else 1e9,
        "sortino": sortino,
        "recovery_factor":
            min(float((eq[-1] - eq[0]) / (mdd * eq[0])), 1e9) if mdd > 0
            else 1e9,
    }
    if trades:
        pnls = np.asarray([t["pnl"] for t in trades])
        wins = pnls > 0
        # streaks
        best = worst = cur = 0
        for w in wins:
            cur = cur + 1 if w else 0
            best = max(best, cur)
        cur = 0
        for w in wins:
            cur = cur + 1 if not w else 0
            worst = max(worst, cur)
        win_rate = wins.mean() if len(pnls) else 0.0
        avg_w = pnls[wins].mean() if wins.any() else 0.0
        avg_l = pnls[~wins].mean() if (~wins).any() else 0.0
        out.update({
            "max_win_streak": int(best),
            "max_loss_streak": int(worst),
            "expectancy": float(win_rate * avg_w + (1 - win_rate) * avg_l),
        })
    return out


def generate_condition_market(condition: str, n_candles: int = 5000,
                              seed: int = 0) -> np.ndarray:
    """Synthetic bull/bear/ranging/volatile/crash market
    (strategy_evaluation.py:1197-1297)."""
    params = {
        "bull": dict(mu=3.0, sigma=0.5),
        "bear": dict(mu=-3.0, sigma=0.6),
        "ranging": dict(mu=0.0, sigma=0.3),
        "volatile": dict(mu=0.0, sigma=1.8),
        "crash": dict(mu=-12.0, sigma=2.2),
    }[condition]
    return candles_chl_v(generate_ohlcv(n_candles, 1, seed=seed, **params))


class StrategyEvaluationSystem:
    """K-fold time-series CV + per-condition evaluation + comparison."""

    def __init__(self, device: str = "cpu"):
        self.device = device

    def _run(self, candles: np.ndarray, params: np.ndarray) -> np.ndarray:
        if self.device.startswith("cuda"):
            import torch

            from ..ops.backtest import run_backtest_gpu
            m = run_backtest_gpu(torch.from_numpy(candles).cuda(),
                                 torch.from_numpy(params).cuda())
            torch.cuda.synchronize()
            return m.cpu().numpy()
        return run_backtest_cpu(candles, params)

    def cross_validate(self, candles: np.ndarray, strategy_params: dict,
                       k: int = 5) -> dict:
        """Time-series k-fold: evaluate on k sequential segments
        (strategy_evaluation.py:635). Folds must stay ordered (no
        shuffling of time)."""
        vec = clip_params(dict_to_params(strategy_params)[None])
        T = candles.shape[1]
        fold_len = T // k
        folds = []
        for i in range(k):
            seg = np.ascontiguousarray(
                candles[:, i * fold_len:(i + 1) * fold_len])
            m = self._run(seg, vec)
            folds.append(metrics_to_stats(m[0, 0], seg.shape[1]))
        sharpes = [f["sharpe"] for f in folds]
        return {
            "folds": folds,
            "mean_sharpe": float(np.mean(sharpes)),
            "std_sharpe": float(np.std(sharpes)),
            "consistency": float((np.asarray(sharpes) > 0).mean()),
        }

    def evaluate_by_condition(self, strategy_params: dict,
                              n_candles: int = 5000,
                              seed: int = 0) -> dict:
        """Per-regime evaluation buckets
        (strategy_evaluation_system.py:587)."""
        vec = clip_params(dict_to_params(strategy_params)[None])
        out = {}
        for cond in ("bull", "bear", "ranging", "volatile", "crash"):
            mkt = generate_condition_market(cond, n_candles, seed)
            m = self._run(mkt, vec)
            out[cond] = metrics_to_stats(m[0, 0], mkt.shape[1])
        return out

    def compare_strategies(self, strategies: dict[str, dict],
                           candles: np.ndarray) -> dict:
        """Multi-strategy comparison table (:1439-2276)."""
        rows = {}
        for name, params in strategies.items():
            vec = clip_params(dict_to_params(params)[None])
            m = self._run(candles, vec)
            rows[name] = metrics_to_stats(
                m.mean(axis=1)[0], candles.shape[1])
        ranking = sorted(rows, key=lambda n: -rows[n]["sharpe"])
        return {"strategies": rows, "ranking": ranking}

    def radar_chart(self, comparison: dict, path: str):
        """Radar viz over sharpe/winrate/PF/drawdown/return (:1439)."""
        import matplotlib
        matplotlib.use("Agg")
        import matplotlib.pyplot as plt

        axes_names = ["sharpe", "win_rate", "profit_factor",
                      "total_return_pct", "max_drawdown_pct"]
        fig = plt.figure(figsize=(6, 6))
        ax = fig.add_subplot(111, polar=True)
        ang = np.linspace(0, 2 * np.pi, len(axes_names), endpoint=False)
        for name, st in comparison["strategies"].items():
            vals = []
            for a in axes_names:
                v = st.get(a, 0.0)
                if a == "max_drawdown_pct":
                    v = -v
                if not np.isfinite(v):
                    v = 0.0
                vals.append(v)
            vals = np.asarray(vals)
            rng = max(np.abs(vals).max(), 1e-9)
            ax.plot(np.r_[ang, ang[0]], np.r_[vals, vals[0]] / rng,
                    label=name)
        ax.set_xticks(ang)
        ax.set_xticklabels(axes_names, fontsize=7)
        ax.legend(fontsize=7)
        fig.savefig(path, dpi=80)
        plt.close(fig)
        return path

    def meets_requirements(self, stats: dict, cfg) -> tuple[bool, list]:
        """Minimum-requirements gate (config evolution block parity:
        min sharpe/win-rate/profit-factor, max drawdown)."""
        fails = []
        if stats.get("sharpe", 0) < cfg.min_sharpe_ratio:
            fails.append("sharpe")
        if stats.get("win_rate", 0) < cfg.min_win_rate:
            fails.append("win_rate")
        if stats.get("profit_factor", 0) < cfg.min_profit_factor:
            fails.append("profit_factor")
        if stats.get("max_drawdown_pct", 100) > cfg.max_drawdown * 100:
            fails.append("max_drawdown")
        return not fails, fails
