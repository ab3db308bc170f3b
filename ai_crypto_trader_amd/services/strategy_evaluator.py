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


STRATEGY_CODE_TEMPLATE = '''\
# auto-generated strategy (proposer: {proposer})
rsi_oversold = {rsi_oversold}
rsi_overbought = {rsi_overbought}
stop_loss_pct = {stop_loss_pct}
take_profit_pct = {take_profit_pct}
position_size_pct = {position_size_pct}

def decide(ctx):
    """Return BUY / SELL / HOLD from a per-candle context."""
    rsi = ctx["rsi14"]
    macd_hist = ctx["macd"] - ctx["macd_signal"]
    if not ctx["in_position"]:
        if rsi < rsi_oversold:
            return "BUY"
    else:
        if rsi > rsi_overbought or macd_hist < 0:
            return "SELL"
    return "HOLD"
'''


class CodeStrategyCycle:
    """Evaluate-improve cycle over strategy *code* (reference
    ai_strategy_evaluator.py:732-1359 + strategy_evaluation_system.py
    :358-432: GPT generates strategy code, statically evaluates it, runs
    it through cross-validated simulation, applies improvement
    suggestions to the code, iterates, and emits a comparison report).

    Here the code artifact is a python `decide(ctx)` snippet; the
    offline proposer is a deterministic template (the LocalAnalyst
    stand-in — an LLM plugs into `propose_fn`/`suggest_fn`), the
    simulator is backtesting/strategy_tester.py (per-candle context ->
    BUY/SELL/HOLD callable), scoring is the evaluation system's metrics
    over time folds and synthetic market conditions, and improvements
    are applied by rewriting parameter assignments in the code text."""

    def __init__(self, propose_fn=None, suggest_fn=None,
                 report_dir: str = "evaluation_reports", seed: int = 0):
        from ..backtesting.strategy_tester import StrategyTester

        self.propose_fn = propose_fn or self._default_proposal
        self.suggest_fn = suggest_fn or self._default_suggestions
        self.tester = StrategyTester()
        self.report_dir = Path(report_dir)
        self.seed = seed

    # --- generation (reference :100) ------------------------------------
    @staticmethod
    def _default_proposal(spec: dict) -> str:
        base = {"rsi_oversold": 30.0, "rsi_overbought": 70.0,
                "stop_loss_pct": 0.03, "take_profit_pct": 0.06,
                "position_size_pct": 0.5, "proposer": "local"}
        base.update(spec or {})
        return STRATEGY_CODE_TEMPLATE.format(**base)

    # --- static code evaluation (reference :148) ------------------------
    @staticmethod
    def static_eval(code: str) -> tuple[bool, list[str]]:
        issues = []
        if "def decide" not in code:
            issues.append("no decide(ctx) entrypoint")
        if "BUY" not in code or "SELL" not in code:
            issues.append("code never trades")
        params = params_from_code(code)
        ok, val_issues = validate_strategy(params)
        issues += val_issues
        try:
            compile(code, "<strategy>", "exec")
        except SyntaxError as e:
            issues.append(f"syntax error: {e}")
        return (not issues), issues

    @staticmethod
    def compile_code(code: str):
        """Compile the code artifact into the tester's decision
        callable in a minimal namespace (no builtins beyond math-safe
        ones — the reference 'deploys' to an isolated worker,
        strategy_evolution_service.py:1465-1510)."""
        ns: dict = {"__builtins__": {"abs": abs, "min": min, "max": max,
                                     "float": float}}
        exec(code, ns)                       # noqa: S102 (sandboxed ns)
        fn = ns.get("decide")
        if not callable(fn):
            raise ValueError("strategy code defines no decide(ctx)")
        return fn

    # --- simulation scoring (reference :261 run CV) ---------------------
    def run_code(self, code: str, candles: np.ndarray,
                 k: int = 3) -> dict:
        fn = self.compile_code(code)
        T = len(candles)
        fold_stats = []
        for i in range(k):
            fold = candles[i * T // k:(i + 1) * T // k]
            res = self.tester.backtest_strategy(fold, fn)
            fold_stats.append(res.stats)
        sharpes = [f["sharpe"] for f in fold_stats]
        mean_sharpe = float(np.mean(sharpes))
        consistency = float(np.mean([s > 0 for s in sharpes]))
        n_trades = int(sum(f["n_trades"] for f in fold_stats))
        score = float(np.clip(mean_sharpe / 3.0, -1, 1) * 40 + 40
                      + consistency * 20)
        if n_trades == 0:
            score = 10.0        # a strategy that never trades is inert
        return {
            "folds": fold_stats,
            "mean_sharpe": mean_sharpe,
            "consistency": consistency,
            "n_trades": int(sum(f["n_trades"] for f in fold_stats)),
            "win_rate": float(np.mean(
                [f["win_rate"] for f in fold_stats])),
            "quality": score,
        }

    # --- improvement (reference :535-731) -------------------------------
    @staticmethod
    def _default_suggestions(code: str, ev: dict) -> list[dict]:
        params = params_from_code(code)
        sugs = []
        if ev["n_trades"] < 5:
            sugs.append({
                "reason": "too few trades: loosen RSI entry",
                "changes": {"rsi_oversold":
                            min(params.get("rsi_oversold", 30) + 8, 48)},
            })
        if ev["win_rate"] < 0.5 and ev["n_trades"] >= 5:
            sugs.append({
                "reason": "low win rate: tighten entry, wider TP",
                "changes": {
                    "rsi_oversold":
                        max(params.get("rsi_oversold", 30) - 5, 10),
                    "take_profit_pct":
                        min(params.get("take_profit_pct", 0.06) * 1.5,
                            0.3),
                },
            })
        if ev["mean_sharpe"] < 0:
            sugs.append({
                "reason": "negative sharpe: cut risk",
                "changes": {
                    "position_size_pct":
                        max(params.get("position_size_pct", 0.5) / 2,
                            0.1),
                    "stop_loss_pct":
                        max(params.get("stop_loss_pct", 0.03) * 0.75,
                            0.01),
                },
            })
        if not sugs:
            sugs.append({
                "reason": "explore: slightly wider take profit",
                "changes": {"take_profit_pct":
                            min(params.get("take_profit_pct", 0.06)
                                * 1.25, 0.3)},
            })
        return sugs

    @staticmethod
    def apply_to_code(code: str, changes: dict) -> str:
        """Rewrite parameter assignments in the code text (the
        improvement lands in the ARTIFACT, as the reference edits its
        generated JS)."""
        import re

        for name, val in changes.items():
            pat = re.compile(rf"^{name}\s*=\s*[-0-9.e]+",
                             flags=re.MULTILINE)
            if pat.search(code):
                code = pat.sub(f"{name} = {val}", code)
            else:
                code = f"{name} = {val}\n" + code
        return code

    # --- the cycle (reference :732) -------------------------------------
    def cycle(self, candles: np.ndarray, spec: dict | None = None,
              rounds: int = 3) -> dict:
        code = self.propose_fn(spec or {})
        ok, issues = self.static_eval(code)
        iterations = []
        ev = self.run_code(code, candles)
        iterations.append({"code": code, "eval": ev,
                           "static_ok": ok, "static_issues": issues,
                           "applied": "initial proposal"})
        best = iterations[0]
        frontier = iterations[0]
        seen = {code}
        for _ in range(rounds):
            # suggestions explore from the FRONTIER (latest evaluated
            # candidate) so non-improving rounds still move — the
            # reference's cycle likewise iterates on the latest code,
            # keeping the best separately (:732-900)
            sugs = self.suggest_fn(frontier["code"], frontier["eval"])
            any_new = False
            for sug in sugs:
                cand_code = self.apply_to_code(frontier["code"],
                                               sug["changes"])
                if cand_code in seen:
                    continue        # converged on this edit already
                seen.add(cand_code)
                any_new = True
                s_ok, s_issues = self.static_eval(cand_code)
                if not s_ok:
                    iterations.append({
                        "code": cand_code, "eval": None,
                        "static_ok": False, "static_issues": s_issues,
                        "applied": sug["reason"]})
                    continue
                cand_ev = self.run_code(cand_code, candles)
                iterations.append({"code": cand_code, "eval": cand_ev,
                                   "static_ok": True,
                                   "static_issues": [],
                                   "applied": sug["reason"]})
                frontier = iterations[-1]
                if cand_ev["quality"] > best["eval"]["quality"]:
                    best = iterations[-1]
            if not any_new:
                break               # every suggested edit already tried
        return {
            "best_code": best["code"],
            "best_eval": best["eval"],
            "iterations": len(iterations),
            "initial_quality": iterations[0]["eval"]["quality"],
            "final_quality": best["eval"]["quality"],
            "improved":
                best["eval"]["quality"]
                > iterations[0]["eval"]["quality"],
            "trail": [{"applied": it["applied"],
                       "quality": (it["eval"] or {}).get("quality"),
                       "static_ok": it["static_ok"]}
                      for it in iterations],
        }

    # --- comparison report (reference :910-1359) ------------------------
    def report(self, result: dict, name: str = "code-cycle") -> Path:
        self.report_dir.mkdir(parents=True, exist_ok=True)
        ts = int(time.time())
        jp = self.report_dir / f"{name}-{ts}.json"
        jp.write_text(json.dumps(
            json.loads(json.dumps(result, default=float)), indent=2))
        md = [f"# Strategy-code evaluate-improve cycle — {name}",
              f"- iterations: {result['iterations']}",
              f"- quality: {result['initial_quality']:.1f} -> "
              f"{result['final_quality']:.1f} "
              f"({'improved' if result['improved'] else 'no gain'})",
              "", "## Trail"]
        for it in result["trail"]:
            q = "static-reject" if it["quality"] is None \
                else f"{it['quality']:.1f}"
            md.append(f"- {it['applied']}: {q}")
        md += ["", "## Best strategy code", "```python",
               result["best_code"], "```"]
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
