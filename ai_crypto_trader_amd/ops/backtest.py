"""GPU backtest op wrapper (HIP kernel: ops/hip/backtest.hip).

CPU golden reference: backtesting/engine_cpu.run_backtest_cpu — both
implement the per-candle state machine specified in
backtesting/strategy.py.
"""

from __future__ import annotations

import torch

from . import require_hip_ops
from ..backtesting.engine_cpu import NMETRIC
from ..backtesting.strategy import NPARAM


def run_backtest_gpu(
    candles: torch.Tensor,     # (nsym, T, 4) f32 cuda
    population: torch.Tensor,  # (P, NPARAM) f32 cuda
    *,
    initial_equity: float = 1.0,
) -> torch.Tensor:             # (P, nsym, NMETRIC) f32 cuda
    ops = require_hip_ops()
    assert candles.is_cuda and population.is_cuda
    assert candles.dtype == torch.float32
    assert population.dtype == torch.float32
    assert candles.dim() == 3 and candles.shape[2] == 4
    assert population.dim() == 2 and population.shape[1] == NPARAM
    candles = candles.contiguous()
    population = population.contiguous()
    nsym, T, _ = candles.shape
    P = population.shape[0]
    metrics = torch.empty(
        (P, nsym, NMETRIC), dtype=torch.float32, device=candles.device
    )
    stream = torch.cuda.current_stream(candles.device).cuda_stream
    ops.backtest(
        candles.data_ptr(), population.data_ptr(), metrics.data_ptr(),
        nsym, T, P, float(initial_equity), stream,
    )
    return metrics


def fitness_from_metrics(metrics: torch.Tensor) -> torch.Tensor:
    """Aggregate per-(param, symbol) fitness to per-param GA fitness:
    mean across symbols (the reference's GA evaluates one fitness per
    individual, strategy_evolution_service.py:525-694)."""
    return metrics[..., 9].mean(dim=1)
