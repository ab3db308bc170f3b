"""Distributed GA strategy-evolution engine (BASELINE config #3).

Fitness evaluation (the expensive part: the per-candle backtest over every
(individual x symbol) pair) is sharded one population slice per GPU; the
tiny (pop x NPARAM) population and fitness vectors are all-gathered over
RCCL/xGMI each generation, and every rank then runs the SAME deterministic
evolution step over the global population (same Philox seed -> identical
children on every rank) and keeps its own slice. This replaces the
reference's serial fitness loop (genetic_algorithm.py:119-133, pop=20)
with pop=1024/GPU marching millions of candles per lane.

Works on GPU (HIP kernels + RCCL) and CPU (numpy reference engines + gloo)
with the same code path, so multi-process correctness is testable without
a GPU (tests/test_distributed.py).
"""

from __future__ import annotations

import numpy as np
import torch

from ..ops.ga import ga_evolve_cpu
from ..parallel import dist as pdist
from .engine_cpu import run_backtest_cpu
from .strategy import PARAM_BOUNDS, random_population


class GAEngine:
    def __init__(
        self,
        candles: np.ndarray,        # (nsym, T, 4) f32
        *,
        pop_per_rank: int = 1024,
        rank: int = 0,
        world: int = 1,
        device: torch.device | str = "cpu",
        seed: int = 0,
        elite_k: int = 16,
        tournament: int = 4,
        cx_rate: float = 0.5,
        mut_rate: float = 0.15,
        mut_scale: float = 0.1,
        segments: int | str = 1,
        continuous: bool = False,
    ):
        """segments > 1 splits each symbol's history into `segments`
        independent backtest segments (a pure reshape of the candle
        tensor): fitness becomes a time-segmented cross-validation — the
        standard GA anti-overfit practice — AND multiplies backtest-lane
        parallelism by `segments` (at pop=1024 x 64 symbols the launch is
        exactly 1 wave/SIMD on MI355X and latency-bound; segments=8 gives
        the SIMDs dependent-latency cover). Total candle-evals per
        generation are unchanged."""
        self.device = torch.device(device)
        self.rank, self.world = rank, world
        self.pop_per_rank = pop_per_rank
        self.seed = seed
        self.elite_k = min(elite_k, max(pop_per_rank * world // 4, 1))
        self.tournament = tournament
        self.cx_rate = cx_rate
        self.mut_rate = mut_rate
        self.mut_scale = mut_scale
        self.gen = 0
        self.nsym, self.T, _ = candles.shape
        if segments == "auto":
            # largest divisor of T (<= 64) keeping segments >= 4096
            # candles: measured optimum on MI355X (profiles/
            # backtest_seg64_pmc.json); 1 on CPU (the numpy reference
            # gains nothing from extra lanes)
            if str(device) != "cpu":
                segments = next(
                    (s for s in (64, 32, 16, 8, 4, 2)
                     if self.T % s == 0 and self.T // s >= 4096), 1)
            else:
                segments = 1
        self.segments = max(int(segments), 1)
        # continuous=True routes GPU fitness through the time-parallel
        # kernel pair (ops/hip/backtest_tp.hip): unsegmented multi-year
        # fitness at full occupancy. Requires segments == 1.
        self.continuous = continuous and self.segments == 1
        if self.segments > 1:
            assert self.T % self.segments == 0, \
                "T must divide evenly into segments"
            candles = np.ascontiguousarray(candles).reshape(
                self.nsym * self.segments, self.T // self.segments, 4)
        self.use_gpu = self.device.type == "cuda"

        # full global population replicated (tiny); rank evaluates its slice
        global_pop = random_population(pop_per_rank * world, seed=seed)
        self.pop_t = torch.from_numpy(global_pop).to(self.device)
        if self.use_gpu:
            self.candles_t = torch.from_numpy(
                np.ascontiguousarray(candles)
            ).to(self.device)
            self.bounds_t = torch.from_numpy(
                np.ascontiguousarray(PARAM_BOUNDS)
            ).to(self.device)
        else:
            self.candles_np = np.ascontiguousarray(candles)
        self.last_metrics = None          # local shard (P, nsym, NMETRIC)
        self.last_fitness_global = None   # (world*P,) torch tensor

    def _my_slice(self):
        p = self.pop_per_rank
        return slice(self.rank * p, (self.rank + 1) * p)

    def eval_fitness(self) -> torch.Tensor:
        """Backtest the local shard; all-gather to global fitness."""
        shard = self.pop_t[self._my_slice()]
        if self.use_gpu:
            if self.continuous:
                from ..ops.backtest import run_backtest_continuous_gpu

                # (activity-sorted lanes were tried here — permuting
                # the shard by trade count to make the trades kernel's
                # wave-word skip fire — and measured a wash, 204 vs 208
                # G/s: total trade count doesn't align the TIMING of
                # activity, so whole-wave-flat 64-candle words barely
                # increase. See profiles/backtest_continuous_pmc.json.)
                metrics = run_backtest_continuous_gpu(
                    self.candles_t, shard)
            else:
                from ..ops.backtest import run_backtest_gpu
                metrics = run_backtest_gpu(self.candles_t, shard)
            fitness_local = metrics[..., 9].mean(dim=1)
        else:
            metrics_np = run_backtest_cpu(
                self.candles_np, shard.cpu().numpy()
            )
            metrics = torch.from_numpy(metrics_np)
            fitness_local = metrics[..., 9].mean(dim=1).to(self.device)
        self.last_metrics = metrics
        self.last_fitness_global = pdist.all_gather_rows(
            fitness_local.contiguous()
        )
        return self.last_fitness_global

    def evolve(self):
        """Deterministic global evolution; all ranks produce identical
        children (same seed/gen), each keeps the full population so no
        parameter broadcast is needed."""
        fitness = self.last_fitness_global
        if self.use_gpu:
            from ..ops.ga import ga_evolve_gpu
            self.pop_t = ga_evolve_gpu(
                self.pop_t, fitness,
                elite_k=self.elite_k, tournament=self.tournament,
                cx_rate=self.cx_rate, mut_rate=self.mut_rate,
                mut_scale=self.mut_scale, seed=self.seed, gen=self.gen,
                bounds_t=self.bounds_t,
            )
        else:
            child = ga_evolve_cpu(
                self.pop_t.cpu().numpy(), fitness.cpu().numpy(),
                elite_k=self.elite_k, tournament=self.tournament,
                cx_rate=self.cx_rate, mut_rate=self.mut_rate,
                mut_scale=self.mut_scale, seed=self.seed, gen=self.gen,
            )
            self.pop_t = torch.from_numpy(child).to(self.device)
        self.gen += 1

    def step(self):
        """One GA generation = fitness eval (sharded) + evolve."""
        self.eval_fitness()
        self.evolve()

    @property
    def candle_evals_per_step(self) -> int:
        """Candle-evaluations per generation per rank: every lane
        (individual x symbol) marches every candle."""
        return self.pop_per_rank * self.nsym * self.T

    def best(self):
        f = self.last_fitness_global
        i = int(torch.argmax(f))
        return float(f[i]), self.pop_t[i].cpu().numpy()
