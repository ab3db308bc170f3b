"""Strategy selection service (reference parity:
services/strategy_selection_service.py:30-1117).

Scores candidate strategies on 6 weighted factors (:71-83 weights:
market_regime 0.30, historical_performance 0.25, risk_profile 0.15,
social_sentiment 0.15, volatility 0.10, feature_importance 0.05), with
time-of-day adjustments (:689), switch hysteresis (min improvement 0.15,
min confidence 0.7, :86-87) and conservative/moderate/aggressive risk
profiles (:96-112)."""

from __future__ import annotations

import time

from ..backtesting.engine import STRATEGY_PRESETS
from ..bus.schema import Channels, Keys, StrategySwitch
from .base import Service

FACTOR_WEIGHTS = {
    "market_regime": 0.30,
    "historical_performance": 0.25,
    "risk_profile": 0.15,
    "social_sentiment": 0.15,
    "volatility": 0.10,
    "feature_importance": 0.05,
}

# which preset suits which regime (regime->strategy mapping,
# strategy_evolution_service.py:977-1090 analog)
REGIME_FIT = {
    "bull": {"momentum": 1.0, "default": 0.6, "dca_strategy": 0.7,
             "mean_reversion": 0.2, "conservative": 0.4},
    "bear": {"conservative": 1.0, "dca_strategy": 0.8, "default": 0.4,
             "mean_reversion": 0.5, "momentum": 0.1},
    "ranging": {"mean_reversion": 1.0, "default": 0.6, "conservative": 0.5,
                "momentum": 0.2, "dca_strategy": 0.5},
    "volatile": {"conservative": 1.0, "mean_reversion": 0.5,
                 "default": 0.3, "momentum": 0.2, "dca_strategy": 0.6},
}

RISK_PROFILES = {
    "conservative": {"conservative": 1.0, "dca_strategy": 0.8,
                     "mean_reversion": 0.5, "default": 0.4, "momentum": 0.1},
    "moderate": {"default": 0.8, "mean_reversion": 0.7, "momentum": 0.6,
                 "dca_strategy": 0.7, "conservative": 0.6},
    "aggressive": {"momentum": 1.0, "default": 0.7, "mean_reversion": 0.6,
                   "dca_strategy": 0.3, "conservative": 0.1},
}


class StrategySelectionService(Service):
    name = "strategy_selection"

    def __init__(self, bus, config=None, risk_profile: str = "moderate",
                 min_improvement: float = 0.15,
                 min_confidence: float = 0.7):
        super().__init__(bus, config)
        self.risk_profile = risk_profile
        self.min_improvement = min_improvement
        self.min_confidence = min_confidence
        self.current = "default"
        self.performance: dict[str, dict] = {}
        # regime -> strategy -> {"pnl_sum", "n"} — the reference's
        # per-regime strategy performance table that drives switching
        # (market_regime_service.py:637-1113)
        self.regime_performance: dict[str, dict[str, dict]] = {}
        self._regime_now = "ranging"
        self.switches = 0

    def record_trade(self, strategy: str, regime: str, pnl_pct: float):
        """Feed a closed trade into the per-regime performance table."""
        row = self.regime_performance.setdefault(regime, {}) \
            .setdefault(strategy, {"pnl_sum": 0.0, "n": 0})
        row["pnl_sum"] += pnl_pct
        row["n"] += 1

    def record_performance(self, strategy: str, perf: dict):
        """Feed an evolution/backtest performance report (reference's
        strategy_performance_{id} keys)."""
        self.performance[strategy] = dict(perf)

    # --- factor scorers (:299-688) ---------------------------------------
    def score_regime(self, strat: str, regime: str) -> float:
        """Static regime fit blended with the MEASURED per-regime trade
        performance once enough outcomes exist (the reference switches
        on its regime-bucketed performance table)."""
        fit = REGIME_FIT.get(regime, REGIME_FIT["ranging"]).get(strat, 0.3)
        row = self.regime_performance.get(regime, {}).get(strat)
        if not row or row["n"] < 3:
            return fit
        avg_pnl = row["pnl_sum"] / row["n"]        # pct per trade
        measured = max(0.0, min(1.0, 0.5 + avg_pnl / 2.0))
        w = min(row["n"] / 10.0, 0.6)              # up to 60% measured
        return (1.0 - w) * fit + w * measured

    def score_history(self, strat: str) -> float:
        p = self.performance.get(strat)
        if not p:
            return 0.5
        sharpe = p.get("sharpe", 0.0)
        return max(0.0, min(1.0, 0.5 + sharpe / 4.0))

    def score_risk(self, strat: str) -> float:
        return RISK_PROFILES[self.risk_profile].get(strat, 0.5)

    def score_social(self, strat: str, sentiment: float) -> float:
        if strat == "momentum":
            return sentiment
        if strat in ("mean_reversion", "conservative"):
            return 1.0 - abs(sentiment - 0.5)
        return 0.5

    def score_volatility(self, strat: str, vol: float) -> float:
        hi = min(vol, 1.0)
        if strat in ("conservative", "dca_strategy"):
            return hi
        if strat == "momentum":
            return 1.0 - hi
        return 0.5

    def score_feature_importance(self, strat: str, fi: dict | None) -> float:
        if not fi:
            return 0.5
        top = fi.get("top_features", [])
        if strat == "momentum" and any("price_change" in f for f in top[:2]):
            return 1.0
        if strat == "mean_reversion" and any(
                f in ("rsi", "bb_position") for f in top[:2]):
            return 1.0
        return 0.5

    def time_of_day_factor(self, hour: int | None = None) -> float:
        """(:689) liquidity-hours adjustment."""
        h = hour if hour is not None else time.gmtime().tm_hour
        return 1.0 if 12 <= h <= 20 else 0.9

    def score(self, strat: str, regime: str, sentiment: float, vol: float,
              fi: dict | None, hour: int | None = None) -> float:
        f = {
            "market_regime": self.score_regime(strat, regime),
            "historical_performance": self.score_history(strat),
            "risk_profile": self.score_risk(strat),
            "social_sentiment": self.score_social(strat, sentiment),
            "volatility": self.score_volatility(strat, vol),
            "feature_importance":
                self.score_feature_importance(strat, fi),
        }
        s = sum(FACTOR_WEIGHTS[k] * v for k, v in f.items())
        return s * self.time_of_day_factor(hour)

    def select_optimal(self, regime: str, sentiment: float, vol: float,
                       fi: dict | None = None,
                       hour: int | None = None) -> tuple[str, float, dict]:
        """(:772) returns (strategy, score, all_scores)."""
        scores = {
            s: self.score(s, regime, sentiment, vol, fi, hour)
            for s in STRATEGY_PRESETS
        }
        best = max(scores, key=scores.get)
        return best, scores[best], scores

    def should_switch(self, best: str, scores: dict) -> bool:
        """Hysteresis (:884-935)."""
        if best == self.current:
            return False
        cur = scores.get(self.current, 0.0)
        imp = (scores[best] - cur) / max(cur, 1e-9)
        return imp >= self.min_improvement and \
            scores[best] >= self.min_confidence * max(scores.values())

    def run_tasks(self):
        return [self._consume_outcomes(), self.run()]

    async def _consume_outcomes(self):
        """Feed the per-regime performance table from closed trades and
        evolution performance reports."""
        sub = self.bus.subscribe(Channels.TRADE_EXECUTIONS,
                                 Channels.STRATEGY_EVOLUTION_UPDATES)

        def on_msg(chan, m):
            if not isinstance(m, dict):
                return
            if chan == Channels.TRADE_EXECUTIONS:
                if m.get("side") == "SELL" and "pnl_pct" in m:
                    self.record_trade(self.current, self._regime_now,
                                      float(m["pnl_pct"]))
            else:
                perf = m.get("performance")
                if perf:
                    self.record_performance(
                        m.get("strategy_id", self.current), perf)

        await self.consume(sub, on_msg)

    async def run(self):
        while self.running:
            regime_d = await self.bus.get_json(
                Keys.CURRENT_MARKET_REGIME) or {}
            regime = regime_d.get("regime", "ranging")
            self._regime_now = regime
            vol = regime_d.get("volatility", 0.5)
            sent = 0.5
            metrics = await self.bus.hgetall(Keys.SOCIAL_METRICS)
            if metrics:
                import json
                vals = [json.loads(v).get("sentiment", 0.5)
                        for v in metrics.values()]
                sent = sum(vals) / len(vals)
            fi = await self.bus.get_json(Keys.FEATURE_IMPORTANCE)
            best, sc, scores = self.select_optimal(regime, sent, vol, fi)
            # the selected_strategy block the reference embeds in signals
            # (README.md:529-543)
            await self.bus.set(Keys.SELECTED_STRATEGY, {
                "name": best,
                "market_regime": regime,
                "performance_score": round(sc, 4),
                "risk_profile": self.risk_profiles.get(
                    best, "moderate") if hasattr(self, "risk_profiles")
                    else "moderate",
                "selection_factors": {k: round(v, 4)
                                      for k, v in scores.items()},
            })
            if self.should_switch(best, scores):
                old = self.current
                self.current = best
                self.switches += 1
                await self.bus.publish(
                    Channels.STRATEGY_SWITCH,
                    StrategySwitch(regime, old, best,
                                   f"selection score {sc:.3f}").to_dict())
                await self.bus.set(Keys.STRATEGY_PARAMS,
                                   STRATEGY_PRESETS[best])
                await self.bus.publish(Channels.STRATEGY_UPDATE, "reload")
            await self.sleep(10.0)
