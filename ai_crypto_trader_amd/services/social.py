"""Social stack (reference parity):
  SocialMonitorService           services/social_monitor_service.py:34-338
  EnhancedSocialMonitorService   enhanced_social_monitor_service.py:38-580
  SocialMetricsAnalyzer          services/utils/social_metrics_analyzer.py
  SocialRiskAdjuster             services/social_risk_adjuster.py:36-673
  SocialStrategyIntegrator       services/social_strategy_integrator.py

Offline-first: a deterministic synthetic social source (seeded, weakly
coupled to price so lead/lag analysis has signal) replaces the LunarCrush
fetcher; a live fetcher can be swapped in behind the same provider seam.
The lead/lag cross-correlation (lags -24..+24, Pearson + Spearman,
argmax |rho| — reference :321-455) runs vectorized; IsolationForest
anomaly detection (:175-290) via sklearn."""

from __future__ import annotations

import math
import time

import numpy as np

from ..bus.schema import Channels, Keys, SocialMetricsBlock, SocialUpdate
from .base import Service


class SyntheticSocialSource:
    """Deterministic social metrics that noisily LEAD price returns by a
    configurable number of steps, so the lead/lag analyzer has ground
    truth to recover (tests assert it)."""

    def __init__(self, seed: int = 0, lead_steps: int = 3,
                 coupling: float = 0.6):
        self.seed = seed
        self.lead = lead_steps
        self.coupling = coupling
        self._rng = np.random.default_rng(seed)
        self._future_ret: dict[str, list[float]] = {}

    def observe_price(self, symbol: str, ret: float):
        self._future_ret.setdefault(symbol, []).append(ret)

    def metrics(self, symbol: str, t: int) -> SocialMetricsBlock:
        rets = self._future_ret.get(symbol, [])
        # sentiment correlates with the return `lead` steps ahead
        future = rets[t + self.lead] if t + self.lead < len(rets) else 0.0
        noise = self._rng.standard_normal() * 0.08
        sent = 0.5 + np.clip(self.coupling * future * 50 + noise, -0.45,
                             0.45)
        base = abs(self._rng.standard_normal())
        return SocialMetricsBlock(
            social_volume=1000 * (1 + base),
            social_engagement=10_000 * (1 + base),
            social_contributors=100 * (1 + base),
            social_sentiment=float(sent),
            twitter_volume=600 * (1 + base),
            reddit_volume=300 * (1 + base),
            news_volume=100 * (1 + base),
        )


class SocialMetricsAnalyzer:
    """Enhanced social math (services/utils/social_metrics_analyzer.py)."""

    def __init__(self, half_life_h: float = 6.0, max_lag: int = 24):
        self.half_life_h = half_life_h
        self.max_lag = max_lag
        self.history: dict[str, list[tuple[float, float]]] = {}
        self.price_history: dict[str, list[float]] = {}
        self.source_weights = {"twitter": 0.35, "reddit": 0.25, "news": 0.4}

    def record(self, symbol: str, sentiment: float, price: float,
               ts: float | None = None):
        self.history.setdefault(symbol, []).append(
            (ts if ts is not None else time.time(), sentiment))
        self.price_history.setdefault(symbol, []).append(price)
        if len(self.history[symbol]) > 4096:
            del self.history[symbol][:2048]
            del self.price_history[symbol][:2048]

    def decayed_sentiment(self, symbol: str,
                          now: float | None = None) -> float:
        """Exponential time-decay sentiment (reference :119)."""
        h = self.history.get(symbol, [])
        if not h:
            return 0.5
        now = now if now is not None else h[-1][0]
        lam = math.log(2) / (self.half_life_h * 3600.0)
        num = den = 0.0
        for ts, s in h[-512:]:
            w = math.exp(-lam * max(now - ts, 0.0))
            num += w * s
            den += w
        return num / den if den > 0 else 0.5

    def lead_lag(self, symbol: str) -> dict:
        """Cross-correlation over lags -max..+max: Pearson + Spearman per
        lag, pick argmax |rho| (reference :321-455). Positive lag means
        sentiment LEADS price."""
        s = np.asarray([x[1] for x in self.history.get(symbol, [])])
        p = np.asarray(self.price_history.get(symbol, []))
        n = min(len(s), len(p))
        if n < 3 * self.max_lag:
            return {"lag": 0, "pearson": 0.0, "spearman": 0.0,
                    "n": int(n)}
        s = s[-n:]
        ret = np.diff(np.log(np.maximum(p[-n:], 1e-12)))
        sd = s[:-1] - s[:-1].mean()

        def rank(x):
            r = np.empty_like(x)
            r[np.argsort(x)] = np.arange(len(x))
            return r

        best = (0.0, 0, 0.0)
        for lag in range(-self.max_lag, self.max_lag + 1):
            if lag >= 0:            # sentiment[t] vs ret[t+lag]
                a, b = sd[: len(sd) - lag or None], ret[lag:]
            else:
                a, b = sd[-lag:], ret[: len(ret) + lag]
            m = min(len(a), len(b))
            if m < 8:
                continue
            a, b = a[:m], b[:m]
            den = a.std() * b.std()
            pe = float((a * b).mean() - a.mean() * b.mean()) / den \
                if den > 0 else 0.0
            ra, rb = rank(a), rank(b)
            dsp = ra.std() * rb.std()
            sp = float(((ra * rb).mean() - ra.mean() * rb.mean()) / dsp) \
                if dsp > 0 else 0.0
            if abs(pe) > abs(best[0]):
                best = (pe, lag, sp)
        return {"lag": best[1], "pearson": best[0], "spearman": best[2],
                "n": int(n)}

    def detect_anomalies(self, symbol: str) -> dict:
        """IsolationForest over recent sentiment+volume rows (:175-290)."""
        h = self.history.get(symbol, [])
        if len(h) < 64:
            return {"anomaly": False, "score": 0.0}
        from sklearn.ensemble import IsolationForest

        X = np.asarray([[s] for _, s in h[-256:]])
        forest = IsolationForest(n_estimators=32, random_state=0,
                                 contamination=0.05)
        labels = forest.fit_predict(X)
        score = float(forest.score_samples(X[-1:])[0])
        return {"anomaly": bool(labels[-1] == -1), "score": score}

    def direction_accuracy(self, symbol: str, horizon: int = 3) -> dict:
        """Sentiment direction-accuracy / information-coefficient
        evaluation (reference :457-634): how often sign(sentiment - 0.5)
        predicted the sign of the forward `horizon`-step return, plus the
        Pearson IC between sentiment and forward return."""
        s = np.asarray([x[1] for x in self.history.get(symbol, [])])
        p = np.asarray(self.price_history.get(symbol, []))
        n = min(len(s), len(p))
        if n < horizon + 16:
            return {"accuracy": 0.5, "ic": 0.0, "n": 0}
        s = s[-n:]
        p = p[-n:]
        fwd = np.log(np.maximum(p[horizon:], 1e-12)) - \
            np.log(np.maximum(p[:-horizon], 1e-12))
        sig = s[:-horizon] - 0.5
        mask = (np.abs(sig) > 1e-6) & (np.abs(fwd) > 1e-12)
        if mask.sum() < 8:
            return {"accuracy": 0.5, "ic": 0.0, "n": int(mask.sum())}
        acc = float((np.sign(sig[mask]) == np.sign(fwd[mask])).mean())
        a = sig[mask] - sig[mask].mean()
        b = fwd[mask] - fwd[mask].mean()
        den = a.std() * b.std()
        ic = float((a * b).mean() / den) if den > 0 else 0.0
        return {"accuracy": acc, "ic": ic, "n": int(mask.sum())}

    def update_source_weights(self, per_source_accuracy: dict) -> dict:
        """Adaptive source-weight updates (reference :635-750): shift
        weight toward sources whose sentiment has been predictive,
        blended 80/20 with the current weights and floored so no source
        dies permanently."""
        names = list(self.source_weights)
        acc = np.asarray([per_source_accuracy.get(k, 0.5) for k in names])
        raw = np.exp((acc - 0.5) * 4.0)
        new = raw / raw.sum()
        cur = np.asarray([self.source_weights[k] for k in names])
        w = 0.8 * cur + 0.2 * new
        w = np.maximum(w, 0.05)
        w = w / w.sum()
        self.source_weights = {k: float(v) for k, v in zip(names, w)}
        return self.source_weights

    def enhanced_sentiment(self, symbol: str) -> dict:
        """Main API (:751): decayed sentiment corrected by lead/lag."""
        base = self.decayed_sentiment(symbol)
        ll = self.lead_lag(symbol)
        an = self.detect_anomalies(symbol)
        conf = min(1.0, abs(ll["pearson"]) * 2.0) * \
            (0.5 if an["anomaly"] else 1.0)
        return {
            "sentiment": base, "confidence": conf, "lead_lag": ll,
            "anomaly": an,
        }


class SocialMonitorService(Service):
    name = "social_monitor"

    def __init__(self, bus, config=None, source=None):
        super().__init__(bus, config)
        self.source = source or SyntheticSocialSource(self.config.seed)
        self.tracked: dict[str, int] = {}
        self.published = 0

    def run_tasks(self):
        return [self._track_symbols(), self._publish_loop()]

    async def _track_symbols(self):
        """Discover symbols from market_updates (reference :188-215)."""
        sub = self.bus.subscribe(Channels.MARKET_UPDATES)
        last_price: dict[str, float] = {}

        def on_msg(_, m):
            sym = m.get("symbol")
            if not sym:
                return
            p = m["current_price"]
            prev = last_price.get(sym)
            last_price[sym] = p
            if prev:
                self.source.observe_price(sym, p / prev - 1.0)
            self.tracked[sym] = self.tracked.get(sym, 0) + 1

        await self.consume(sub, on_msg)

    async def _publish_loop(self):
        t = 0
        while self.running:
            for sym in list(self.tracked):
                mb = self.source.metrics(sym, t)
                upd = SocialUpdate(symbol=sym, metrics=mb,
                                   weighted_sentiment=mb.social_sentiment)
                await self.bus.publish(Channels.SOCIAL_UPDATES,
                                       upd.to_dict())
                await self.bus.hset(Keys.SOCIAL_METRICS, sym, {
                    "sentiment": mb.social_sentiment,
                    "social_volume": mb.social_volume,
                })
                self.published += 1
                self.metrics.social_sentiment.labels(sym).set(mb.social_sentiment)
            t += 1
            await self.sleep(
                min(self.config.social.update_interval_s, 1.0))

    async def run(self):
        pass


class EnhancedSocialMonitorService(Service):
    """Runs raw social metrics through the analyzer and publishes
    `enhanced_social_updates` + lead/lag keys (reference :365-516)."""

    name = "enhanced_social_monitor"

    def __init__(self, bus, config=None):
        super().__init__(bus, config)
        self.analyzer = SocialMetricsAnalyzer(
            half_life_h=self.config.social.sentiment_half_life_h,
            max_lag=self.config.social.lead_lag_max_h)
        self.prices: dict[str, float] = {}
        self.enhanced = 0

    def run_tasks(self):
        return [self._consume(), self._report_loop()]

    async def _consume(self):
        sub = self.bus.subscribe(Channels.SOCIAL_UPDATES,
                                 Channels.MARKET_UPDATES)

        def on_msg(chan, m):
            if chan == Channels.MARKET_UPDATES:
                if m.get("symbol"):
                    self.prices[m["symbol"]] = m["current_price"]
                return
            sym = m.get("symbol")
            if not sym:
                return
            sent = m.get("data", {}).get("weighted_sentiment", 0.5)
            self.analyzer.record(sym, sent, self.prices.get(sym, 1.0))

        await self.consume(sub, on_msg)

    async def _report_loop(self):
        while self.running:
            for sym in list(self.analyzer.history):
                enh = self.analyzer.enhanced_sentiment(sym)
                await self.bus.publish(Channels.ENHANCED_SOCIAL_UPDATES, {
                    "symbol": sym, **enh,
                })
                await self.bus.set(Keys.social_lead_lag(sym),
                                   enh["lead_lag"])
                self.enhanced += 1
            await self.sleep(2.0)

    async def run(self):
        pass


class SocialRiskAdjuster(Service):
    """Sentiment-weighted risk deltas (services/social_risk_adjuster.py:
    source-weighted score, exponential decay half-life 6h, BULLISH/BEARISH
    -> position/SL/TP multipliers, data-quality gate). Writes the
    `social_risk_adjustments` hash consumed at trade time
    (trade_executor_service.py:799-814)."""

    name = "social_risk_adjuster"

    def __init__(self, bus, config=None):
        super().__init__(bus, config)
        self.analyzer = SocialMetricsAnalyzer(
            half_life_h=self.config.social.sentiment_half_life_h)
        self.samples: dict[str, int] = {}

    def run_tasks(self):
        return [self._consume(), self._adjust_loop()]

    async def _consume(self):
        sub = self.bus.subscribe(Channels.SOCIAL_UPDATES)

        def on_msg(_, m):
            sym = m.get("symbol")
            if sym:
                self.analyzer.record(
                    sym, m.get("data", {}).get("weighted_sentiment", 0.5),
                    1.0)
                self.samples[sym] = self.samples.get(sym, 0) + 1

        await self.consume(sub, on_msg)

    def adjustment(self, sym: str) -> dict:
        if self.samples.get(sym, 0) < 10:       # data-quality gate (:323)
            return {"position_multiplier": 1.0, "stop_multiplier": 1.0,
                    "tp_multiplier": 1.0, "stance": "NEUTRAL"}
        s = self.analyzer.decayed_sentiment(sym)
        if s > 0.65:
            return {"position_multiplier": 1.2, "stop_multiplier": 1.1,
                    "tp_multiplier": 1.2, "stance": "BULLISH",
                    "sentiment": s}
        if s < 0.35:
            return {"position_multiplier": 0.6, "stop_multiplier": 0.8,
                    "tp_multiplier": 0.8, "stance": "BEARISH",
                    "sentiment": s}
        return {"position_multiplier": 1.0, "stop_multiplier": 1.0,
                "tp_multiplier": 1.0, "stance": "NEUTRAL", "sentiment": s}

    async def _adjust_loop(self):
        while self.running:
            report = {}
            for sym in list(self.analyzer.history):
                adj = self.adjustment(sym)
                await self.bus.hset(Keys.SOCIAL_RISK_ADJUSTMENTS, sym, adj)
                report[sym] = adj
            if report:
                await self.bus.set(Keys.SOCIAL_RISK_REPORT, {
                    "at": time.time(), "adjustments": report,
                })
                await self.bus.publish(Channels.RISK_ADJUSTMENT_UPDATES,
                                       {"adjustments": report})
            await self.sleep(2.0)

    async def run(self):
        pass


class SocialStrategyIntegrator:
    """Social <-> price correlation, impact analysis, strategy variants
    (services/social_strategy_integrator.py:25-741). Used by the evolution
    service to bias parameters with social context."""

    def __init__(self, analyzer: SocialMetricsAnalyzer):
        self.analyzer = analyzer

    def impact(self, symbol: str) -> dict:
        ll = self.analyzer.lead_lag(symbol)
        strength = abs(ll["pearson"])
        return {
            "symbol": symbol, "lead_lag": ll,
            "impact": "high" if strength > 0.3 else
                      ("medium" if strength > 0.1 else "low"),
        }

    def strategy_variants(self, symbol: str, base_params: dict) -> list[dict]:
        """trend-following vs contrarian social variants (:566-664)."""
        imp = self.impact(symbol)
        follow = dict(base_params)
        follow.update({"entry_votes": max(
            1, int(base_params.get("entry_votes", 2)) - 1)})
        contra = dict(base_params)
        contra.update({"rsi_oversold":
                       base_params.get("rsi_oversold", 30) - 5,
                       "rsi_overbought":
                       base_params.get("rsi_overbought", 70) + 5})
        return [
            {"name": "social_trend_following", "params": follow,
             "impact": imp},
            {"name": "social_contrarian", "params": contra, "impact": imp},
        ]

    def adjust_parameters(self, symbol: str, params: dict) -> dict:
        """Parameter adjustment API used by evolution (:316-391)."""
        s = self.analyzer.decayed_sentiment(symbol)
        out = dict(params)
        if s > 0.65:
            out["position_size_pct"] = min(
                params.get("position_size_pct", 0.5) * 1.2, 1.0)
        elif s < 0.35:
            out["position_size_pct"] = max(
                params.get("position_size_pct", 0.5) * 0.7, 0.05)
        return out
