"""Hyperparameter search (reference parity: neural_network_service.py's
Optuna study :588-767 — 20 trials over lr/hidden/batch/seq). Offline
self-contained (no optuna in the image): seeded random search and a TPE
sampler (Bergstra et al. 2011 — the algorithm Optuna defaults to), both
with successive-halving pruning."""

from __future__ import annotations

import time
from dataclasses import dataclass, field

import numpy as np


@dataclass
class Trial:
    number: int
    params: dict
    value: float = float("inf")
    state: str = "running"        # running | complete | pruned
    history: list = field(default_factory=list)


SPACE = {
    "lr": ("log", 1e-4, 1e-2),
    "hidden1": ("choice", [32, 64]),
    "hidden2": ("choice", [32, 64]),
    "batch_size": ("choice", [32, 64, 128, 256]),
    "model_type": ("choice", ["lstm", "gru"]),
}


class RandomSearchStudy:
    def __init__(self, space: dict | None = None, seed: int = 0,
                 prune_after: int = 2, prune_quantile: float = 0.6):
        self.space = space or SPACE
        self.rng = np.random.default_rng(seed)
        self.trials: list[Trial] = []
        self.prune_after = prune_after
        self.prune_quantile = prune_quantile

    def sample(self) -> dict:
        out = {}
        for name, spec in self.space.items():
            kind = spec[0]
            if kind == "log":
                lo, hi = spec[1], spec[2]
                out[name] = float(np.exp(self.rng.uniform(
                    np.log(lo), np.log(hi))))
            elif kind == "uniform":
                out[name] = float(self.rng.uniform(spec[1], spec[2]))
            else:
                out[name] = spec[1][int(self.rng.integers(len(spec[1])))]
        return out

    def should_prune(self, trial: Trial, step: int, value: float) -> bool:
        """Successive-halving style: prune a trial whose intermediate value
        is worse than the quantile of completed trials at the same step."""
        trial.history.append(value)
        if step < self.prune_after:
            return False
        peers = [t.history[step] for t in self.trials
                 if t is not trial and len(t.history) > step]
        if len(peers) < 3:
            return False
        return value > float(np.quantile(peers, self.prune_quantile))

    def optimize(self, objective, n_trials: int = 20) -> Trial:
        """objective(trial) -> final value; may call
        study.should_prune(trial, step, v) and raise Pruned."""
        for i in range(n_trials):
            t = Trial(i, self.sample())
            self.trials.append(t)
            t0 = time.time()
            try:
                t.value = float(objective(t))
                t.state = "complete"
            except Pruned:
                t.state = "pruned"
                t.value = t.history[-1] if t.history else float("inf")
            t.params["_seconds"] = time.time() - t0
        return self.best_trial

    @property
    def best_trial(self) -> Trial:
        done = [t for t in self.trials if t.state == "complete"]
        return min(done or self.trials, key=lambda t: t.value)


class TPEStudy(RandomSearchStudy):
    """Tree-structured Parzen Estimator sampler: after `n_startup` random
    trials, split completed trials at the gamma quantile into good/bad
    sets, fit a Parzen (KDE) density to each per parameter, and pick the
    candidate maximizing the good/bad density ratio among `n_ei` draws
    from the good density — Optuna's default sampler, self-contained."""

    def __init__(self, space: dict | None = None, seed: int = 0,
                 n_startup: int = 6, gamma: float = 0.3, n_ei: int = 24,
                 **kw):
        super().__init__(space, seed, **kw)
        self.n_startup = n_startup
        self.gamma = gamma
        self.n_ei = n_ei

    def _to_unit(self, name, v):
        spec = self.space[name]
        if spec[0] == "log":
            lo, hi = np.log(spec[1]), np.log(spec[2])
            return (np.log(v) - lo) / (hi - lo)
        if spec[0] == "uniform":
            return (v - spec[1]) / (spec[2] - spec[1])
        return spec[1].index(v) / max(len(spec[1]) - 1, 1)

    def _from_unit(self, name, u):
        spec = self.space[name]
        u = float(np.clip(u, 0.0, 1.0))
        if spec[0] == "log":
            lo, hi = np.log(spec[1]), np.log(spec[2])
            return float(np.exp(lo + u * (hi - lo)))
        if spec[0] == "uniform":
            return float(spec[1] + u * (spec[2] - spec[1]))
        i = int(round(u * (len(spec[1]) - 1)))
        return spec[1][i]

    @staticmethod
    def _parzen_logpdf(x, centers, bw):
        if len(centers) == 0:
            return np.zeros_like(x)
        d = (x[:, None] - centers[None, :]) / bw
        return np.log(np.exp(-0.5 * d * d).mean(axis=1) / bw + 1e-12)

    def sample(self) -> dict:
        done = [t for t in self.trials if t.state == "complete"]
        if len(done) < self.n_startup:
            return super().sample()
        done = sorted(done, key=lambda t: t.value)
        n_good = max(1, int(np.ceil(self.gamma * len(done))))
        good, bad = done[:n_good], done[n_good:]
        out = {}
        for name in self.space:
            g = np.array([self._to_unit(name, t.params[name])
                          for t in good])
            b = np.array([self._to_unit(name, t.params[name])
                          for t in bad])
            bw = max(1.0 / (1 + len(g)), 0.08)
            cand = np.clip(
                g[self.rng.integers(len(g), size=self.n_ei)]
                + self.rng.normal(0, bw, self.n_ei), 0, 1)
            score = (self._parzen_logpdf(cand, g, bw)
                     - self._parzen_logpdf(cand, b, bw))
            out[name] = self._from_unit(name, cand[int(np.argmax(score))])
        return out


class Pruned(Exception):
    pass
