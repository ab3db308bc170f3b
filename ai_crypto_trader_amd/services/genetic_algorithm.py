"""Generic GA API (reference parity: services/genetic_algorithm.py:27-392
— `GeneticAlgorithm` over param dicts with tournament selection + elitism
:135-161, uniform crossover :163-189, int/float mutation :191-223, a
`run()` generation loop :254-291 and diversity :322-348).

Unlike the reference's serial fitness loop (:119-133 — pop=20 evaluated
one-by-one), fitness here is evaluated as ONE vectorized call (and, for
backtest fitness, one GPU kernel launch over the whole population via
backtesting/ga_engine.GAEngine — that class is the production path; this
one is the dict-API seam for custom fitness functions)."""

from __future__ import annotations

import numpy as np


class GeneticAlgorithm:
    def __init__(self, param_ranges: dict[str, tuple],
                 fitness_function, population_size: int = 64,
                 generations: int = 10, elite_k: int = 4,
                 tournament: int = 4, cx_rate: float = 0.5,
                 mut_rate: float = 0.15, mut_scale: float = 0.1,
                 seed: int = 0):
        """param_ranges: name -> (lo, hi) or (lo, hi, 'int').
        fitness_function: EITHER f(individual_dict) -> float (reference
        signature) OR f_batch(list[dict]) -> array (preferred: vectorized;
        auto-detected via the `batch` attribute)."""
        self.ranges = param_ranges
        self.fitness_function = fitness_function
        self.batched = getattr(fitness_function, "batch", False)
        self.pop_size = population_size
        self.generations = generations
        self.elite_k = min(elite_k, max(population_size // 4, 1))
        self.tournament = tournament
        self.cx_rate = cx_rate
        self.mut_rate = mut_rate
        self.mut_scale = mut_scale
        self.rng = np.random.default_rng(seed)
        self.names = list(param_ranges)
        self.population = [self._random_individual()
                           for _ in range(population_size)]
        self.best: tuple[float, dict] | None = None
        self.history: list[dict] = []

    def _is_int(self, name) -> bool:
        r = self.ranges[name]
        return len(r) > 2 and r[2] == "int"

    def _random_individual(self) -> dict:
        ind = {}
        for n in self.names:
            lo, hi = self.ranges[n][:2]
            v = self.rng.uniform(lo, hi)
            ind[n] = int(round(v)) if self._is_int(n) else float(v)
        return ind

    def _clip(self, ind: dict) -> dict:
        out = {}
        for n in self.names:
            lo, hi = self.ranges[n][:2]
            v = min(max(ind[n], lo), hi)
            out[n] = int(round(v)) if self._is_int(n) else float(v)
        return out

    def evaluate(self) -> np.ndarray:
        if self.batched:
            return np.asarray(self.fitness_function(self.population),
                              dtype=np.float64)
        return np.asarray([self.fitness_function(ind)
                           for ind in self.population], dtype=np.float64)

    def _evolve(self, fitness: np.ndarray):
        order = np.argsort(-fitness)
        new = [dict(self.population[i]) for i in order[: self.elite_k]]
        while len(new) < self.pop_size:
            pa = self._tournament_pick(fitness)
            pb = self._tournament_pick(fitness)
            child = {}
            for n in self.names:
                src = pa if self.rng.random() < self.cx_rate else pb
                v = src[n]
                if self.rng.random() < self.mut_rate:
                    lo, hi = self.ranges[n][:2]
                    v = v + self.rng.standard_normal() * \
                        self.mut_scale * (hi - lo)
                child[n] = v
            new.append(self._clip(child))
        self.population = new

    def _tournament_pick(self, fitness: np.ndarray) -> dict:
        idx = self.rng.integers(0, self.pop_size, self.tournament)
        return self.population[int(idx[np.argmax(fitness[idx])])]

    def run(self) -> tuple[dict, float]:
        """(:254-291) returns (best_individual, best_fitness)."""
        for gen in range(self.generations):
            fitness = self.evaluate()
            i = int(np.argmax(fitness))
            if self.best is None or fitness[i] > self.best[0]:
                self.best = (float(fitness[i]), dict(self.population[i]))
            self.history.append({
                "gen": gen, "best": float(fitness[i]),
                "mean": float(fitness.mean()),
                "diversity": self.diversity(),
            })
            self._evolve(fitness)
        return self.best[1], self.best[0]

    def diversity(self) -> float:
        """Mean normalized per-param variance (:322-348)."""
        vals = np.asarray([[ind[n] for n in self.names]
                           for ind in self.population], dtype=np.float64)
        lo = np.asarray([self.ranges[n][0] for n in self.names])
        hi = np.asarray([self.ranges[n][1] for n in self.names])
        norm = (vals - lo) / np.maximum(hi - lo, 1e-12)
        return float(norm.var(axis=0).mean())
