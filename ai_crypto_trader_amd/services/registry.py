"""Model registry, explainability, feature importance (reference parity):
  ModelRegistryService        services/model_registry_service.py:23-539
  AIExplainabilityService     services/ai_explainability_service.py:22-474
  FeatureImportanceAnalyzer   services/feature_importance_analyzer.py:41-634
  FeatureImportanceIntegrator services/model_integration.py:21-387
"""

from __future__ import annotations

import json
import time
import uuid
from pathlib import Path

import numpy as np

from ..bus.schema import Channels, Keys
from .base import Service


class ModelRegistryService(Service):
    """JSON-file + bus-key model registry: register / perf update /
    best-model query / status mgmt / comparison, with pubsub events."""

    name = "model_registry"

    def __init__(self, bus, config=None,
                 store_path: str = "models_store/registry.json"):
        super().__init__(bus, config)
        self.store_path = Path(store_path)
        self.models: dict[str, dict] = {}
        if self.store_path.exists():
            self.models = json.loads(self.store_path.read_text())

    def _persist(self):
        self.store_path.parent.mkdir(parents=True, exist_ok=True)
        self.store_path.write_text(json.dumps(self.models, indent=2))

    async def register(self, name: str, model_type: str,
                       params: dict | None = None) -> str:
        mid = f"{model_type}-{uuid.uuid4().hex[:8]}"
        self.models[mid] = {
            "id": mid, "name": name, "type": model_type,
            "params": params or {}, "status": "active",
            "registered_at": time.time(), "performance": {},
        }
        self._persist()
        await self.bus.set(Keys.MODEL_REGISTRY, self.models)
        await self.bus.publish(Channels.MODEL_REGISTRY_EVENTS, {
            "event": "registered", "id": mid, "type": model_type,
        })
        return mid

    async def update_performance(self, mid: str, perf: dict):
        if mid not in self.models:
            return
        self.models[mid]["performance"] = perf
        self.models[mid]["updated_at"] = time.time()
        self._persist()
        await self.bus.set(Keys.MODEL_REGISTRY, self.models)
        await self.bus.publish(Channels.MODEL_PERFORMANCE_UPDATES, {
            "id": mid, "performance": perf,
        })

    def best_model(self, model_type: str | None = None,
                   metric: str = "sharpe") -> dict | None:
        cands = [m for m in self.models.values()
                 if m["status"] == "active"
                 and (model_type is None or m["type"] == model_type)
                 and metric in m.get("performance", {})]
        return max(cands, key=lambda m: m["performance"][metric],
                   default=None)

    async def set_status(self, mid: str, status: str):
        if mid in self.models:
            self.models[mid]["status"] = status
            self._persist()

    def compare(self, ids: list[str], metric: str = "sharpe") -> list[dict]:
        rows = [
            {"id": i, metric: self.models[i]["performance"].get(metric)}
            for i in ids if i in self.models
        ]
        return sorted(rows, key=lambda r: (r[metric] is None,
                                           -(r[metric] or 0)))

    async def run(self):
        while self.running:
            await self.sleep(5.0)


class AIExplainabilityService(Service):
    """Formats explanation/factor-weight breakdowns of trading signals,
    persists them, publishes `explained_trading_signals`."""

    name = "ai_explainability"

    def __init__(self, bus, config=None,
                 out_dir: str = "explanations"):
        super().__init__(bus, config)
        self.out_dir = Path(out_dir)
        self.explained = 0

    @staticmethod
    def format_explanation(sig: dict) -> dict:
        exp = sig.get("explanation", {})
        weights = sig.get("factor_weights", {})
        rows = sorted(
            ({"factor": k,
              "score": exp.get(k, 0.0),
              "weight": weights.get(k, 0.0),
              "contribution": exp.get(k, 0.0) * weights.get(k, 0.0)}
             for k in set(exp) | set(weights)),
            key=lambda r: -abs(r["contribution"]))
        return {
            "symbol": sig.get("symbol"),
            "decision": sig.get("decision"),
            "confidence": sig.get("confidence"),
            "reasoning": sig.get("reasoning", ""),
            "factors": rows,
            "model_version": sig.get("model_version"),
        }

    async def run(self):
        sub = self.bus.subscribe(Channels.TRADING_SIGNALS)
        self.out_dir.mkdir(parents=True, exist_ok=True)

        async def on_msg(_, sig):
            if not isinstance(sig, dict) or "decision" not in sig:
                return
            ex = self.format_explanation(sig)
            await self.bus.publish(Channels.EXPLAINED_TRADING_SIGNALS, ex)
            self.explained += 1
            if self.explained % 50 == 1:      # sampled persistence
                p = self.out_dir / f"{sig.get('symbol')}_{int(time.time())}.json"
                p.write_text(json.dumps(ex, indent=2))

        await self.consume(sub, on_msg)


class FeatureImportanceAnalyzer(Service):
    """RandomForest over historical signal -> outcome pairs with
    permutation + impurity importance, category grouping, pruned model;
    publishes `feature_importance` (feature_importance_analyzer.py)."""

    name = "feature_importance"

    FEATURES = ["rsi", "stoch_k", "williams_r", "macd", "bb_position",
                "trend_strength", "price_change_1m", "price_change_5m",
                "confidence", "sentiment"]
    CATEGORIES = {
        "oscillators": ["rsi", "stoch_k", "williams_r"],
        "trend": ["macd", "trend_strength"],
        "momentum": ["price_change_1m", "price_change_5m"],
        "bands": ["bb_position"],
        "model": ["confidence"],
        "social": ["sentiment"],
    }

    def __init__(self, bus, config=None):
        super().__init__(bus, config)
        self.samples: list[tuple[list[float], float]] = []
        self.pending: dict[str, tuple[list[float], float]] = {}
        self.report: dict | None = None

    def _featurize(self, sig: dict) -> list[float]:
        md = sig.get("market_data", {})
        return [
            md.get("rsi", 50.0), md.get("stoch_k", 50.0),
            md.get("williams_r", -50.0), md.get("macd", 0.0),
            md.get("bb_position", 0.5), md.get("trend_strength", 0.0),
            md.get("price_change_1m", 0.0), md.get("price_change_5m", 0.0),
            sig.get("confidence", 0.5),
            sig.get("explanation", {}).get("social_sentiment", 0.0),
        ]

    def record_outcome(self, sig: dict, future_return: float):
        self.samples.append((self._featurize(sig), future_return))
        del self.samples[:-5000]

    def analyze(self, n_repeats: int = 10) -> dict | None:
        """Impurity + permutation importances (:333-395)."""
        if len(self.samples) < 100:
            return None
        from sklearn.ensemble import RandomForestRegressor
        from sklearn.inspection import permutation_importance

        X = np.asarray([s[0] for s in self.samples])
        y = np.asarray([s[1] for s in self.samples])
        rf = RandomForestRegressor(n_estimators=40, random_state=0,
                                   max_depth=6).fit(X, y)
        perm = permutation_importance(rf, X, y, n_repeats=n_repeats,
                                      random_state=0)
        imp = {f: {"impurity": float(i), "permutation": float(p)}
               for f, i, p in zip(self.FEATURES, rf.feature_importances_,
                                  perm.importances_mean)}
        cats = {
            c: float(sum(imp[f]["permutation"] for f in fs))
            for c, fs in self.CATEGORIES.items()
        }
        top = sorted(imp, key=lambda f: -imp[f]["permutation"])
        cat_sorted = sorted(cats, key=lambda c: -cats[c])
        # shape per reference README.md:420-468 (+ the richer per-feature
        # importances block the integrator consumes)
        self.report = {
            "at": time.time(),
            "analysis_type": "feature_importance",
            "model_type": "RandomForest",
            "feature_count": len(self.FEATURES),
            "n_samples": len(self.samples),
            "importances": imp, "categories": cats,
            "top_features": top[:5],
            "top_features_permutation": {
                ff: round(imp[ff]["permutation"], 5) for ff in top[:10]},
            "top_categories": {c: round(cats[c], 5) for c in cat_sorted},
            "recommendations": {
                "features_to_prioritize": top[:5],
                "features_to_reconsider": top[-3:],
                "categories_to_prioritize": cat_sorted[:2],
                "categories_to_reconsider": cat_sorted[-1:],
            },
            # pruned "optimized model" on the top features only
            # (reference feature_importance_analyzer.py:550): shows how
            # much predictive power the top-5 retain vs the full set
            "optimized_model": self._pruned_model(X, y, rf, top[:5]),
        }
        return self.report

    def _pruned_model(self, X, y, full_rf, keep: list[str]) -> dict:
        from sklearn.ensemble import RandomForestRegressor

        idx = [self.FEATURES.index(f) for f in keep]
        pruned = RandomForestRegressor(
            n_estimators=40, random_state=0, max_depth=6,
        ).fit(X[:, idx], y)
        return {
            "features": keep,
            "score_full": round(float(full_rf.score(X, y)), 4),
            "score_pruned": round(float(pruned.score(X[:, idx], y)), 4),
        }

    async def run(self):
        sub = self.bus.subscribe(Channels.TRADING_SIGNALS,
                                 Channels.MARKET_UPDATES)

        def on_msg(chan, msg):
            if chan == Channels.TRADING_SIGNALS and "decision" in msg:
                sym = msg.get("symbol")
                price = msg.get("market_data", {}).get("current_price", 0.0)
                if sym:
                    self.pending[sym] = (self._featurize(msg), price)
            elif chan == Channels.MARKET_UPDATES:
                sym = msg.get("symbol")
                if sym in self.pending:
                    feats, p0 = self.pending.pop(sym)
                    if p0 > 0:
                        self.samples.append(
                            (feats, msg["current_price"] / p0 - 1.0))

        task = self.consume(sub, on_msg)

        async def periodic():
            while self.running:
                await self.sleep(10.0)
                rep = self.analyze(n_repeats=5)
                if rep:
                    await self.bus.set(Keys.FEATURE_IMPORTANCE, rep)
                    await self.bus.publish(Channels.FEATURE_IMPORTANCE, rep)

        import asyncio
        await asyncio.gather(task, periodic())


class FeatureImportanceIntegrator:
    """Client-side consumer of the feature-importance output
    (model_integration.py:21-387): cached weights, outcome predictor,
    strategy weight adjustment."""

    def __init__(self):
        self.weights: dict[str, float] = {}
        self.updated_at = 0.0

    def update(self, report: dict):
        imp = report.get("importances", {})
        tot = sum(v["permutation"] for v in imp.values()) or 1.0
        self.weights = {f: v["permutation"] / tot for f, v in imp.items()}
        self.updated_at = report.get("at", time.time())

    def predict_outcome(self, features: dict) -> float:
        """Logistic-ish weighted score of a prospective trade (:220-287)."""
        z = sum(self.weights.get(f, 0.0) * np.tanh(v / 50.0)
                for f, v in features.items())
        return float(1.0 / (1.0 + np.exp(-4.0 * z)))

    def adjust_factor_weights(self, factor_weights: dict) -> dict:
        """Shift analyst factor weights toward measured importance
        (:288-350)."""
        cat_map = {"oscillators": "oscillators", "trend": "trend",
                   "momentum": "momentum", "social": "social_sentiment"}
        out = dict(factor_weights)
        for cat, factor in cat_map.items():
            w = sum(self.weights.get(f, 0.0)
                    for f in FeatureImportanceAnalyzer.CATEGORIES.get(
                        cat, []))
            if factor in out and w > 0:
                out[factor] = 0.7 * out[factor] + 0.3 * w
        s = sum(out.values()) or 1.0
        return {k: v / s for k, v in out.items()}
