"""Backtest result analysis (reference parity:
backtesting/result_analyzer.py:12-427 — equity/drawdown plots, trade
analysis, summary report + best-by-metric, comparison chart). Plots are
matplotlib-Agg PNG files; summaries are plain dict/JSON."""

from __future__ import annotations

from pathlib import Path

import numpy as np


class ResultAnalyzer:
    def __init__(self, out_dir: str = "backtesting_data/analysis"):
        self.out_dir = Path(out_dir)
        self.out_dir.mkdir(parents=True, exist_ok=True)

    def plot_equity_curve(self, stats: dict, fname: str | None = None):
        """(:73-149) equity + running-peak drawdown panels."""
        eq = np.asarray(stats.get("equity_curve", []))
        if eq.size == 0:
            return None
        import matplotlib
        matplotlib.use("Agg")
        import matplotlib.pyplot as plt

        peak = np.maximum.accumulate(eq)
        dd = (peak - eq) / peak * 100
        fig, (a1, a2) = plt.subplots(2, 1, figsize=(10, 6), sharex=True)
        a1.plot(eq)
        a1.set_ylabel("equity")
        a1.set_title(f"{stats.get('symbol')} {stats.get('strategy')} — "
                     f"return {stats.get('total_return_pct', 0):.2f}%  "
                     f"sharpe {stats.get('sharpe', 0):.2f}")
        a2.fill_between(np.arange(len(dd)), -dd, 0, alpha=0.5, color="r")
        a2.set_ylabel("drawdown %")
        p = self.out_dir / (fname or
                            f"{stats.get('symbol')}_equity.png")
        fig.savefig(p, dpi=80)
        plt.close(fig)
        return p

    def summary_report(self, results: list[dict]) -> dict:
        """(:226-329) aggregate summary + best-by-metric."""
        if not results:
            return {"n": 0}
        by = {}
        for metric in ("sharpe", "total_return_pct", "win_rate",
                       "profit_factor"):
            vals = [(r.get(metric, float("-inf")), i)
                    for i, r in enumerate(results)]
            vals = [(v if np.isfinite(v) else -1e18, i) for v, i in vals]
            _, best_i = max(vals)
            b = results[best_i]
            by[metric] = {"symbol": b.get("symbol"),
                          "strategy": b.get("strategy"),
                          "value": b.get(metric)}
        return {
            "n": len(results),
            "mean_return_pct": float(np.mean(
                [r.get("total_return_pct", 0) for r in results])),
            "mean_sharpe": float(np.mean(
                [r.get("sharpe", 0) for r in results])),
            "mean_win_rate": float(np.mean(
                [r.get("win_rate", 0) for r in results])),
            "best_by": by,
        }

    def comparison_chart(self, results: list[dict],
                         fname: str = "comparison.png"):
        """(:330-416) bar chart of returns per (symbol, strategy)."""
        if not results:
            return None
        import matplotlib
        matplotlib.use("Agg")
        import matplotlib.pyplot as plt

        labels = [f"{r.get('symbol')}\n{r.get('strategy')}"
                  for r in results]
        vals = [r.get("total_return_pct", 0) for r in results]
        fig, ax = plt.subplots(figsize=(max(6, len(labels)), 4))
        colors = ["g" if v >= 0 else "r" for v in vals]
        ax.bar(range(len(vals)), vals, color=colors)
        ax.set_xticks(range(len(labels)))
        ax.set_xticklabels(labels, fontsize=7)
        ax.set_ylabel("total return %")
        p = self.out_dir / fname
        fig.savefig(p, dpi=80)
        plt.close(fig)
        return p

    def trade_analysis(self, stats: dict) -> dict:
        """(:150-225) trade-level aggregates from the stats dict."""
        n = stats.get("n_trades", 0)
        return {
            "n_trades": n,
            "win_rate": stats.get("win_rate", 0.0),
            "profit_factor": stats.get("profit_factor", 0.0),
            "expectancy_pct":
                (stats.get("total_return_pct", 0.0) / n) if n else 0.0,
            "max_drawdown_pct": stats.get("max_drawdown_pct", 0.0),
        }
