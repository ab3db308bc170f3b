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
        self.switches = 0

    # --- factor scorers (:299-688) ---------------------------------------
    def score_regime(self, strat: str, regime: str) -> float:
        return REGIME_FIT.get(regime, REGIME_FIT["ranging"]).get(strat, 0.3)

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

    async def run(self):
        while self.running:
            regime_d = await self.bus.get_json(
                Keys.CURRENT_MARKET_REGIME) or {}
            regime = regime_d.get("regime", "ranging")
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
