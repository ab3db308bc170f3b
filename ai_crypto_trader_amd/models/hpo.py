"""Hyperparameter search (reference parity: neural_network_service.py's
Optuna study :588-767 — 20 trials over lr/hidden/batch/seq). Offline
self-contained: seeded random search + successive-halving pruning (no
optuna in the image)."""

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


class Pruned(Exception):
    pass
