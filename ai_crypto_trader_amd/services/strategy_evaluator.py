"""Systematic strategy evaluate-improve cycle + regime dataset collection
(reference parity):
  AIStrategyEvaluator        services/ai_strategy_evaluator.py:31-1492 —
      generate -> evaluate (CV) -> quality score -> per-regime scores ->
      min-requirements gate -> improvement suggestions -> apply -> repeat.
      The reference drives generation/critique through GPT + deploys JS to
      Cloudflare Workers (:1402-1510); here candidates are parameter sets
      of the native strategy machine, improvement suggestions are derived
      from measured weaknesses, and an LLM can be plugged into the
      `suggest_fn` seam. HTML report (:910-1359) -> JSON/markdown report.
  MarketRegimeDataCollector  services/utils/market_regime_data_collector.py
      :11-512 — assembles regime-training datasets from bus history.
"""

from __future__ import annotations

import json
import time
from pathlib import Path

import numpy as np

from ..backtesting.evaluation import StrategyEvaluationSystem
from ..bus.schema import Keys
from ..config import get_config


class AIStrategyEvaluator:
    def __init__(self, device: str = "cpu", suggest_fn=None,
                 report_dir: str = "evaluation_reports"):
        self.ses = StrategyEvaluationSystem(device)
        self.suggest_fn = suggest_fn or self.heuristic_suggestions
        self.report_dir = Path(report_dir)
        self.cfg = get_config().evolution

    # --- scoring (quality score :345; regime scores :423) ----------------
    def quality_score(self, cv: dict, conditions: dict) -> float:
        """0-100 composite: CV sharpe + consistency + regime robustness."""
        sharpe_part = np.clip(cv["mean_sharpe"] / 3.0, -1, 1) * 35 + 35
        consistency_part = cv["consistency"] * 15
        cond_sharpes = [c["sharpe"] for c in conditions.values()]
        robust_part = np.clip(min(cond_sharpes) / 2.0, -1, 1) * 7.5 + 7.5
        return float(np.clip(sharpe_part + consistency_part + robust_part,
                             0, 100))

    def evaluate(self, params: dict, candles: np.ndarray,
                 k: int = 4) -> dict:
        cv = self.ses.cross_validate(candles, params, k=k)
        conditions = self.ses.evaluate_by_condition(params,
                                                    n_candles=3000)
        mean_stats = {
            "sharpe": cv["mean_sharpe"],
            "win_rate": float(np.mean(
                [f["win_rate"] for f in cv["folds"]])),
            "profit_factor": float(np.median(
                [min(f["profit_factor"], 10.0) for f in cv["folds"]])),
            "max_drawdown_pct": float(np.max(
                [f["max_drawdown_pct"] for f in cv["folds"]])),
        }
        ok, fails = self.ses.meets_requirements(mean_stats, self.cfg)
        return {
            "params": dict(params),
            "cv": cv,
            "conditions": conditions,
            "stats": mean_stats,
            "quality": self.quality_score(cv, conditions),
            "meets_requirements": ok,
            "failed_requirements": fails,
        }

    # --- improvement loop (:535-900) -------------------------------------
    @staticmethod
    def heuristic_suggestions(evaluation: dict) -> list[dict]:
        """Derive parameter nudges from measured weaknesses (the GPT seam:
        pass an LLM-backed suggest_fn with the same signature)."""
        s = evaluation["stats"]
        p = evaluation["params"]
        out = []
        if s["max_drawdown_pct"] > 10.0:
            out.append({"reason": "drawdown too deep: tighten stops, "
                                  "smaller size",
                        "changes": {
                            "stop_loss_pct":
                                p.get("stop_loss_pct", 0.02) * 0.7,
                            "position_size_pct":
                                p.get("position_size_pct", 0.5) * 0.7}})
        if s["win_rate"] < 0.5:
            out.append({"reason": "win rate low: demand more confirmation",
                        "changes": {
                            "entry_votes":
                                min(p.get("entry_votes", 2) + 1, 3)}})
        if s["profit_factor"] < 1.2:
            out.append({"reason": "profit factor low: let winners run",
                        "changes": {
                            "take_profit_pct":
                                p.get("take_profit_pct", 0.04) * 1.4,
                            "trailing_stop_pct": max(
                                p.get("trailing_stop_pct", 0.0), 0.01)}})
        if s["sharpe"] < 0:
            out.append({"reason": "negative sharpe: trade less",
                        "changes": {
                            "entry_votes": 3,
                            "exit_votes":
                                max(p.get("exit_votes", 2) - 1, 1)}})
        return out

    def improve_cycle(self, params: dict, candles: np.ndarray,
                      rounds: int = 3) -> dict:
        """Systematic evaluate-improve cycle (:732)."""
        history = []
        best = self.evaluate(params, candles)
        history.append(best)
        cur = best
        for _ in range(rounds):
            suggestions = self.suggest_fn(cur)
            if not suggestions:
                break
            improved = False
            for sug in suggestions:
                cand = dict(cur["params"])
                cand.update(sug["changes"])
                ev = self.evaluate(cand, candles)
                ev["applied"] = sug["reason"]
                history.append(ev)
                if ev["quality"] > cur["quality"]:
                    cur = ev
                    improved = True
            if cur["quality"] > best["quality"]:
                best = cur
            if not improved:
                break
        return {"best": best, "history_len": len(history),
                "improved": best["quality"] > history[0]["quality"],
                "initial_quality": history[0]["quality"],
                "final_quality": best["quality"]}

    def report(self, result: dict, name: str = "evaluation") -> Path:
        """JSON + markdown report (reference emits HTML :910-1359)."""
        self.report_dir.mkdir(parents=True, exist_ok=True)
        ts = int(time.time())
        jp = self.report_dir / f"{name}-{ts}.json"
        slim = json.loads(json.dumps(result, default=float))
        jp.write_text(json.dumps(slim, indent=2))
        best = result["best"]
        md = [
            f"# Strategy evaluation — {name}",
            f"- quality: {best['quality']:.1f}/100 "
            f"(initial {result['initial_quality']:.1f})",
            f"- meets requirements: {best['meets_requirements']} "
            f"(failed: {best['failed_requirements']})",
            f"- CV mean sharpe: {best['cv']['mean_sharpe']:.2f} "
            f"(consistency {best['cv']['consistency']:.0%})",
            "", "## Per-condition sharpe",
        ]
        for cond, st in best["conditions"].items():
            md.append(f"- {cond}: {st['sharpe']:.2f}")
        (self.report_dir / f"{name}-{ts}.md").write_text("\n".join(md))
        return jp


PARAM_ALIASES = {
    # common spellings in generated strategy code -> canonical param name
    "rsi_period": "rsi_period", "rsiperiod": "rsi_period",
    "rsi_oversold": "rsi_oversold", "oversold": "rsi_oversold",
    "rsi_overbought": "rsi_overbought", "overbought": "rsi_overbought",
    "ema_fast": "ema_fast", "fast_ema": "ema_fast",
    "ema_slow": "ema_slow", "slow_ema": "ema_slow",
    "macd_signal": "macd_signal", "signal_period": "macd_signal",
    "bb_window": "bb_window", "bb_period": "bb_window",
    "bollinger_period": "bb_window",
    "bb_k": "bb_k", "bb_std": "bb_k", "bollinger_std": "bb_k",
    "entry_votes": "entry_votes", "exit_votes": "exit_votes",
    "position_size": "position_size_pct",
    "position_size_pct": "position_size_pct",
    "stop_loss": "stop_loss_pct", "stop_loss_pct": "stop_loss_pct",
    "stoploss": "stop_loss_pct",
    "take_profit": "take_profit_pct", "take_profit_pct": "take_profit_pct",
    "takeprofit": "take_profit_pct",
    "trailing_stop": "trailing_stop_pct",
    "trailing_stop_pct": "trailing_stop_pct",
    "trailing_activation": "trailing_act_pct",
    "trailing_act_pct": "trailing_act_pct",
    "stoch_oversold": "stoch_os", "stoch_os": "stoch_os",
    "stoch_overbought": "stoch_ob", "stoch_ob": "stoch_ob",
}


def params_from_code(code: str) -> dict:
    """Extract strategy parameters from strategy CODE text (reference
    strategy_evolution_service.py:1512-1569: regex pulls params out of the
    GPT-generated Cloudflare-worker JS). Accepts python/JS-ish assignments
    and dict/object literals ('stop_loss_pct = 0.03', 'stopLoss: 0.03',
    '"take_profit": 0.05'); returns canonical param names -> floats, which
    clip_params/params_to_vec turn into a native kernel strategy."""
    import re

    found: dict = {}
    pat = re.compile(
        r'["\']?([A-Za-z_][A-Za-z0-9_]*)["\']?\s*[:=]\s*'
        r'(-?\d+(?:\.\d+)?(?:e-?\d+)?)')
    for name, raw in pat.findall(code):
        # camelCase -> snake_case, then alias lookup
        snake = re.sub(r'(?<=[a-z0-9])([A-Z])', r'_\1', name).lower()
        canon = PARAM_ALIASES.get(snake)
        if canon is not None:
            v = float(raw)
            # percent-style values ("stop_loss = 3" meaning 3%)
            if canon.endswith("_pct") and v >= 1.0:
                v /= 100.0
            found[canon] = v
    return found


def validate_strategy(params: dict) -> tuple[bool, list[str]]:
    """Strategy validator (reference STRATEGY_EVOLUTION.md 'Strategy
    Validator Service' + ai_strategy_evaluator's min-requirements gate):
    sanity-checks a proposed parameter set — risk management present,
    values inside the native bounds, sizes/ratios coherent — before it
    is allowed to trade or evolve."""
    from ..backtesting.strategy import PARAM_BOUNDS, PARAM_NAMES

    issues = []
    bounds = {n: (float(b[0]), float(b[1]))
              for n, b in zip(PARAM_NAMES, PARAM_BOUNDS)}
    for name, v in params.items():
        if name in bounds:
            lo, hi = bounds[name]
            if not (lo <= float(v) <= hi):
                issues.append(f"{name}={v} outside [{lo}, {hi}]")
    if float(params.get("stop_loss_pct", 0.0)) <= 0:
        issues.append("no stop loss (risk management required)")
    tp = float(params.get("take_profit_pct", 0.0))
    sl = float(params.get("stop_loss_pct", 1.0))
    if tp > 0 and sl > 0 and tp / sl < 0.5:
        issues.append("take profit under half the stop "
                      "(reward:risk < 0.5)")
    if float(params.get("position_size_pct", 0.0)) > 1.0:
        issues.append("position size above 100%")
    return (not issues), issues


class MarketRegimeDataCollector:
    """Assembles regime-training datasets from bus history
    (market_regime_data_collector.py:44-395: price/signal/outcome history
    -> engineered features -> labeled windows)."""

    def __init__(self, bus):
        self.bus = bus
        self.rows: list[dict] = []

    async def collect(self) -> dict | None:
        prices = await self.bus.hgetall(Keys.CURRENT_PRICES)
        regime = await self.bus.get_json(Keys.CURRENT_MARKET_REGIME)
        risk = await self.bus.get_json(Keys.PORTFOLIO_RISK)
        if not prices or not regime:
            return None
        row = {
            "at": time.time(),
            "prices": {k: float(v) for k, v in prices.items()},
            "regime": regime.get("regime"),
            "volatility": regime.get("volatility", 0.0),
            "portfolio_var": (risk or {}).get("portfolio_var", 0.0),
        }
        self.rows.append(row)
        del self.rows[:-20_000]
        return row

    def dataset(self, min_rows: int = 32):
        """-> (X (n, 4), y labels) engineered from collected rows."""
        if len(self.rows) < min_rows:
            return None, None
        X, y = [], []
        for prev, cur in zip(self.rows[:-1], self.rows[1:]):
            syms = set(prev["prices"]) & set(cur["prices"])
            if not syms:
                continue
            rets = [cur["prices"][s] / prev["prices"][s] - 1.0
                    for s in syms]
            X.append([float(np.mean(rets)), float(np.std(rets)),
                      cur["volatility"], cur["portfolio_var"]])
            y.append(cur["regime"])
        return np.asarray(X, np.float32), np.asarray(y)
