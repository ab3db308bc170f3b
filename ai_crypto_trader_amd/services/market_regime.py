"""Market regime service (reference parity:
services/market_regime_service.py + services/utils/market_regime_detector.py).

Features per window: return, volatility, trend strength (rolling slope),
RSI, MACD, BB width (market_regime_detector.py:64-110). Detectors: KMeans
/ GMM / Gaussian HMM (torch, GPU-capable) and the reference's heuristic
cluster->label
mapping by mean return & volatility (:226-296), plus the rule-based
fallback (market_regime_service.py:503-604). Publishes `strategy_switch`
on regime changes; maintains `current_market_regime` /
`market_regime_history` keys."""

from __future__ import annotations

import time

import numpy as np
import torch

from ..bus.schema import Channels, Keys, StrategySwitch
from .base import Service

REGIMES = ("bull", "bear", "ranging", "volatile")


def extract_features(closes: np.ndarray, win: int = 32) -> np.ndarray:
    """(T,) closes -> (n_win, 6) features [ret, vol, slope, rsi, macd, bbw]
    over trailing windows (market_regime_detector.py:64-110 semantics)."""
    T = len(closes)
    n = T - win
    if n <= 0:
        return np.zeros((0, 6), np.float32)
    out = np.zeros((n, 6), np.float32)
    lr = np.diff(np.log(closes))
    for i in range(n):
        w = closes[i:i + win]
        r = lr[i:i + win - 1]
        out[i, 0] = w[-1] / w[0] - 1.0
        out[i, 1] = r.std()
        x = np.arange(win)
        out[i, 2] = np.polyfit(x, w / w.mean(), 1)[0] * win
        g = np.maximum(np.diff(w), 0).mean()
        l = np.maximum(-np.diff(w), 0).mean()
        out[i, 3] = 100 * g / max(g + l, 1e-12)
        ema_f = w[-12:].mean()
        ema_s = w[-26:].mean() if win >= 26 else w.mean()
        out[i, 4] = (ema_f - ema_s) / w[-1]
        out[i, 5] = w.std() / max(w.mean(), 1e-9)
    return out


class KMeansTorch:
    """Plain torch KMeans (runs on GPU when tensors are cuda) — the
    detector's default model (market_regime_detector.py:138)."""

    def __init__(self, k: int = 4, iters: int = 50, seed: int = 0):
        self.k = k
        self.iters = iters
        self.seed = seed
        self.centers: torch.Tensor | None = None

    def fit(self, X: torch.Tensor):
        g = torch.Generator(device="cpu").manual_seed(self.seed)
        idx = torch.randperm(X.shape[0], generator=g)[: self.k]
        self.centers = X[idx.to(X.device)].clone()
        for _ in range(self.iters):
            d = torch.cdist(X, self.centers)
            assign = d.argmin(dim=1)
            for c in range(self.k):
                m = assign == c
                if m.any():
                    self.centers[c] = X[m].mean(dim=0)
        return self

    def predict(self, X: torch.Tensor) -> torch.Tensor:
        return torch.cdist(X, self.centers).argmin(dim=1)


class GMMTorch:
    """Diagonal-covariance GMM via EM (market_regime_detector.py:144)."""

    def __init__(self, k: int = 4, iters: int = 60, seed: int = 0):
        self.k = k
        self.iters = iters
        self.seed = seed

    def fit(self, X: torch.Tensor):
        km = KMeansTorch(self.k, 20, self.seed).fit(X)
        self.mu = km.centers.clone()
        self.var = torch.ones_like(self.mu) * X.var(dim=0, keepdim=True)
        self.pi = torch.full((self.k,), 1.0 / self.k, device=X.device)
        for _ in range(self.iters):
            logp = self._log_prob(X)                       # (N, k)
            logr = logp - torch.logsumexp(logp, dim=1, keepdim=True)
            r = logr.exp()
            nk = r.sum(dim=0) + 1e-9
            self.mu = (r.T @ X) / nk[:, None]
            self.var = (r.T @ (X ** 2)) / nk[:, None] - self.mu ** 2
            self.var = torch.clamp(self.var, min=1e-8)
            self.pi = nk / nk.sum()
        return self

    def _log_prob(self, X):
        d = X[:, None, :] - self.mu[None]
        return (torch.log(self.pi)[None]
                - 0.5 * ((d ** 2) / self.var[None]).sum(-1)
                - 0.5 * torch.log(self.var[None]).sum(-1))

    def predict(self, X):
        return self._log_prob(X).argmax(dim=1)

    def predict_proba(self, X):
        lp = self._log_prob(X)
        return (lp - torch.logsumexp(lp, 1, keepdim=True)).exp()


class GaussianHMMTorch:
    """Diagonal-covariance Gaussian HMM via Baum-Welch EM, log-space
    forward-backward (market_regime_detector.py:150 — hmmlearn GaussianHMM
    equivalent; data is tiny so a per-timestep torch loop is fine)."""

    def __init__(self, k: int = 4, iters: int = 30, seed: int = 0):
        self.k = k
        self.iters = iters
        self.seed = seed

    def _emission_logp(self, X):
        d = X[:, None, :] - self.mu[None]                    # (T, k, F)
        return (-0.5 * ((d ** 2) / self.var[None]).sum(-1)
                - 0.5 * torch.log(self.var[None]).sum(-1))

    def fit(self, X: torch.Tensor):
        T = X.shape[0]
        k = self.k
        km = KMeansTorch(k, 20, self.seed).fit(X)
        self.mu = km.centers.clone()
        self.var = torch.ones_like(self.mu) * X.var(dim=0, keepdim=True)
        self.var = torch.clamp(self.var, min=1e-6)
        dev = X.device
        # sticky prior: regimes persist (diag-heavy transition init)
        self.logA = torch.log(torch.full((k, k), 0.05 / (k - 1), device=dev)
                              .fill_diagonal_(0.95))
        self.logpi = torch.full((k,), -np.log(k), device=dev)
        for _ in range(self.iters):
            logb = self._emission_logp(X)                    # (T, k)
            # forward / backward in log space
            la = torch.empty(T, k, device=dev)
            lb = torch.zeros(T, k, device=dev)
            la[0] = self.logpi + logb[0]
            for t in range(1, T):
                la[t] = logb[t] + torch.logsumexp(
                    la[t - 1][:, None] + self.logA, dim=0)
            for t in range(T - 2, -1, -1):
                lb[t] = torch.logsumexp(
                    self.logA + (logb[t + 1] + lb[t + 1])[None], dim=1)
            lg = la + lb
            lg = lg - torch.logsumexp(lg, dim=1, keepdim=True)
            g = lg.exp()                                     # (T, k) gammas
            # xi sums for the transition update
            xi = torch.zeros(k, k, device=dev)
            ll = torch.logsumexp(la[-1], dim=0)
            for t in range(T - 1):
                lx = (la[t][:, None] + self.logA
                      + (logb[t + 1] + lb[t + 1])[None] - ll)
                xi += lx.exp()
            self.logA = torch.log(xi / (g[:-1].sum(0)[:, None] + 1e-9)
                                  + 1e-12)
            self.logpi = torch.log(g[0] + 1e-12)
            nk = g.sum(0) + 1e-9
            self.mu = (g.T @ X) / nk[:, None]
            self.var = torch.clamp(
                (g.T @ (X ** 2)) / nk[:, None] - self.mu ** 2, min=1e-6)
        return self

    def predict(self, X: torch.Tensor) -> torch.Tensor:
        """Viterbi path."""
        T = X.shape[0]
        logb = self._emission_logp(X)
        delta = self.logpi + logb[0]
        back = torch.zeros(T, self.k, dtype=torch.long, device=X.device)
        for t in range(1, T):
            s = delta[:, None] + self.logA
            best = s.max(dim=0)
            delta = best.values + logb[t]
            back[t] = best.indices
        path = torch.empty(T, dtype=torch.long, device=X.device)
        path[-1] = delta.argmax()
        for t in range(T - 2, -1, -1):
            path[t] = back[t + 1, path[t + 1]]
        return path


def label_clusters(X: np.ndarray, assign: np.ndarray, k: int) -> dict:
    """Heuristic cluster -> regime mapping by mean return & volatility
    (market_regime_detector.py:226-296)."""
    stats = []
    for c in range(k):
        m = assign == c
        if not m.any():
            stats.append((0.0, 0.0))
            continue
        stats.append((float(X[m, 0].mean()), float(X[m, 1].mean())))
    vols = [s[1] for s in stats]
    vol_hi = np.percentile(vols, 75)
    labels = {}
    for c, (ret, vol) in enumerate(stats):
        if vol >= vol_hi and vol > 0:
            labels[c] = "volatile"
        elif ret > 0.002:
            labels[c] = "bull"
        elif ret < -0.002:
            labels[c] = "bear"
        else:
            labels[c] = "ranging"
    return labels


class RandomForestRegime:
    """Supervised RandomForest regime classifier — the reference's 4th
    detector mode (market_regime_detector.py:156): trained on labeled
    historical windows, predicts the regime of the latest one.

    Labels come from `label_windows` (rule-based labeling of each
    historical window — the offline stand-in for the reference's
    collected outcome labels, market_regime_data_collector.py:44-241);
    a caller with real labeled data passes them straight to fit()."""

    def __init__(self, n_estimators: int = 100, seed: int = 0):
        from sklearn.ensemble import RandomForestClassifier

        self.clf = RandomForestClassifier(
            n_estimators=n_estimators, random_state=seed, n_jobs=1)
        self.classes_: list[str] = []

    @staticmethod
    def label_windows(closes: np.ndarray, win: int = 32) -> np.ndarray:
        """Rule-label each feature window (bull/bear/ranging/volatile)."""
        n = len(closes) - win + 1
        labels = np.empty(n, dtype=object)
        for i in range(n):
            labels[i] = rule_based_regime(closes[i:i + win])[0]
        return labels

    def fit(self, X: np.ndarray, y: np.ndarray):
        self.clf.fit(X, y)
        self.classes_ = list(self.clf.classes_)
        return self

    def predict(self, X: np.ndarray) -> np.ndarray:
        return self.clf.predict(X)

    def predict_proba(self, X: np.ndarray) -> np.ndarray:
        return self.clf.predict_proba(X)

    def feature_importances(self) -> np.ndarray:
        return self.clf.feature_importances_


def rule_based_regime(closes: np.ndarray) -> tuple[str, float]:
    """Rule-based thresholds (market_regime_service.py:503-604)."""
    if len(closes) < 30:
        return "ranging", 0.0
    r = np.diff(np.log(closes[-100:]))
    ret = closes[-1] / closes[-min(len(closes), 100)] - 1.0
    vol = float(r.std() * np.sqrt(525_600))
    if vol > 1.2:
        return "volatile", min(vol / 2.0, 1.0)
    if ret > 0.01:
        return "bull", min(abs(ret) * 20, 1.0)
    if ret < -0.01:
        return "bear", min(abs(ret) * 20, 1.0)
    return "ranging", 0.3


class MarketRegimeService(Service):
    name = "market_regime"

    def __init__(self, bus, config=None, device="cpu"):
        super().__init__(bus, config)
        self.device = device
        self.prices: dict[str, list[float]] = {}
        self.model = None
        self.labels: dict[int, str] = {}
        self.current = "ranging"
        self.history: list[dict] = []

    def run_tasks(self):
        return [self._consume_market(), self._regime_loop()]

    async def _consume_market(self):
        sub = self.bus.subscribe(Channels.MARKET_UPDATES)

        def on_msg(_, m):
            if m.get("symbol"):
                h = self.prices.setdefault(m["symbol"], [])
                h.append(m["current_price"])
                if len(h) > 4096:
                    del h[:2048]

        await self.consume(sub, on_msg)

    def _primary_closes(self) -> np.ndarray | None:
        if not self.prices:
            return None
        sym = self.config.trading.symbols[0] \
            if self.config.trading.symbols[0] in self.prices \
            else next(iter(self.prices))
        h = self.prices[sym]
        return np.asarray(h, np.float64) if len(h) >= 64 else None

    def detect(self, closes: np.ndarray) -> tuple[str, float]:
        method = self.config.regime.method
        if method == "rule" or len(closes) < 200:
            return rule_based_regime(closes)
        X_np = extract_features(closes)
        if len(X_np) < 4 * self.config.regime.n_regimes:
            return rule_based_regime(closes)
        if method == "rf":
            # supervised mode (market_regime_detector.py:156): rule
            # labels over historical windows -> RandomForest -> predict
            # the latest window with a class-probability confidence
            y = RandomForestRegime.label_windows(closes)
            n = min(len(X_np), len(y))
            rf = RandomForestRegime(seed=self.config.seed)
            rf.fit(X_np[:n - 1], y[:n - 1])
            proba = rf.predict_proba(X_np[n - 1:n])[0]
            i = int(np.argmax(proba))
            self.model = rf
            return str(rf.classes_[i]), float(proba[i])
        X = torch.from_numpy(X_np)
        if self.device != "cpu":
            X = X.to(self.device)
        Xn = (X - X.mean(0)) / (X.std(0) + 1e-9)
        cls = {"gmm": GMMTorch, "hmm": GaussianHMMTorch}.get(
            method, KMeansTorch)
        self.model = cls(self.config.regime.n_regimes,
                         seed=self.config.seed).fit(Xn)
        assign = self.model.predict(Xn).cpu().numpy()
        self.labels = label_clusters(X_np, assign, self.config.regime.n_regimes)
        regime = self.labels.get(int(assign[-1]), "ranging")
        conf = float((assign == assign[-1]).mean())
        return regime, conf

    async def _regime_loop(self):
        while self.running:
            closes = self._primary_closes()
            if closes is not None:
                closes = closes[-self.config.regime.lookback:]
                try:
                    regime, conf = self.detect(closes)
                except Exception as e:
                    self.log.warning("detect failed: %s", e)
                    regime, conf = rule_based_regime(closes)
                vol = float(np.diff(np.log(closes[-100:])).std()
                            * np.sqrt(525_600)) if len(closes) > 2 else 0.0
                entry = {"regime": regime, "confidence": conf,
                         "volatility": min(vol, 2.0), "at": time.time()}
                await self.bus.set(Keys.CURRENT_MARKET_REGIME, entry)
                self.history.append(entry)
                await self.bus.set(Keys.MARKET_REGIME_HISTORY,
                                   self.history[-100:])
                if regime != self.current:
                    await self.bus.publish(
                        Channels.STRATEGY_SWITCH,
                        StrategySwitch(regime, f"for-{self.current}",
                                       f"for-{regime}",
                                       "regime change").to_dict())
                    self.current = regime
            await self.sleep(5.0)

    async def run(self):
        pass
