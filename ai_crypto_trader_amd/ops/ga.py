"""GA evolution op wrapper (HIP kernel: ops/hip/ga.hip) + host-side GA.

The device kernel evolves a device-resident population in place;
GeneticAlgorithm (services side) uses these ops when a GPU is present and
the numpy path otherwise (genetic_algorithm.py:27-291 parity).
"""

from __future__ import annotations

import numpy as np

from . import require_hip_ops
from ..backtesting.strategy import NPARAM, PARAM_BOUNDS, clip_params


def ga_evolve_gpu(
    pop,                 # (P, NPARAM) f32 cuda
    fitness,             # (P,) f32 cuda
    *,
    elite_k: int = 8,
    tournament: int = 4,
    cx_rate: float = 0.5,
    mut_rate: float = 0.15,
    mut_scale: float = 0.1,
    seed: int = 0,
    gen: int = 0,
    bounds_t=None,
):
    """One generation: returns the next population (new tensor)."""
    import torch

    ops = require_hip_ops()
    P = pop.shape[0]
    assert pop.is_cuda and fitness.is_cuda
    order = torch.argsort(fitness, descending=True).to(torch.int32)
    if bounds_t is None:
        bounds_t = torch.from_numpy(np.ascontiguousarray(PARAM_BOUNDS)).to(
            pop.device
        )
    out = torch.empty_like(pop)
    stream = torch.cuda.current_stream(pop.device).cuda_stream
    ops.ga_evolve(
        pop.contiguous().data_ptr(), fitness.contiguous().data_ptr(),
        order.contiguous().data_ptr(), bounds_t.contiguous().data_ptr(),
        out.data_ptr(), P, elite_k, tournament, cx_rate, mut_rate, mut_scale,
        seed, gen, stream,
    )
    return out


def ga_evolve_cpu(
    pop: np.ndarray,
    fitness: np.ndarray,
    *,
    elite_k: int = 8,
    tournament: int = 4,
    cx_rate: float = 0.5,
    mut_rate: float = 0.15,
    mut_scale: float = 0.1,
    seed: int = 0,
    gen: int = 0,
) -> np.ndarray:
    """numpy GA generation (tournament + elitism + uniform crossover +
    gaussian mutation — genetic_algorithm.py:135-223 semantics). Not
    bit-identical to the GPU kernel (different RNG streams); tests check
    invariants on both."""
    rng = np.random.default_rng((seed * 1_000_003 + gen) & 0xFFFFFFFF)
    P = pop.shape[0]
    elite_k = min(elite_k, max(P // 4, 1))
    order = np.argsort(-fitness)
    out = np.empty_like(pop)
    out[:elite_k] = pop[order[:elite_k]]
    n_child = P - elite_k
    cand_a = rng.integers(0, P, size=(n_child, tournament))
    cand_b = rng.integers(0, P, size=(n_child, tournament))
    pa = cand_a[np.arange(n_child), np.argmax(fitness[cand_a], axis=1)]
    pb = cand_b[np.arange(n_child), np.argmax(fitness[cand_b], axis=1)]
    mask = rng.random((n_child, NPARAM)) < cx_rate
    child = np.where(mask, pop[pa], pop[pb])
    lo, hi = PARAM_BOUNDS[:, 0], PARAM_BOUNDS[:, 1]
    mut = rng.random((n_child, NPARAM)) < mut_rate
    noise = rng.standard_normal((n_child, NPARAM)) * mut_scale * (hi - lo)
    child = child + np.where(mut, noise, 0.0)
    out[elite_k:] = child.astype(np.float32)
    return clip_params(out)


def population_diversity(pop: np.ndarray) -> float:
    """Mean normalized per-param variance (genetic_algorithm.py:322-348)."""
    lo, hi = PARAM_BOUNDS[:, 0], PARAM_BOUNDS[:, 1]
    norm = (np.asarray(pop) - lo) / (hi - lo + 1e-12)
    return float(norm.var(axis=0).mean())
