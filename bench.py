#!/usr/bin/env python3
"""Flagship benchmark: GA strategy-evolution backtest throughput.

Headline metric (BASELINE.json): backtested 1m-candles/sec during GA
population fitness evaluation — pop=1024 strategies/GPU x 64 symbols x T
synthetic 1m candles, the per-candle HIP backtest kernel
(ops/hip/backtest.hip), fitness all-gathered over RCCL each generation.

One step = one GA generation (population fitness backtest + evolution).
Weak scaling: each of the N ranks evaluates its own pop=1024 shard of a
global population of 1024*N.

Run (driver contract):
  python bench.py --gpus 1 --steps 5 --warmup 2
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 bench.py --gpus 8 --steps 5 --warmup 2

Synthetic data (seeded GBM, data/synthetic.py), random-init population.
Outside the timed GA loop the same run also measures (and reports under
"config"): the CONTINUOUS unsegmented backtest via the time-parallel
kernel pair (all ranks), and under "secondary": the Monte-Carlo VaR path
rate (10M correlated-GBM paths x 64 assets), the LSTM-predictor train
rate (BASELINE #2) and the PPO train rate (BASELINE #4) — so the
driver's clock attests every BASELINE config.
"""

from __future__ import annotations

import argparse
import json
import sys
import time
from pathlib import Path

import numpy as np
import torch

from ai_crypto_trader_amd.backtesting.ga_engine import GAEngine
from ai_crypto_trader_amd.data.synthetic import candles_chl_v, generate_ohlcv
from ai_crypto_trader_amd.parallel import dist as pdist


def parse_args():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--pop", type=int, default=1024,
                    help="population per GPU")
    ap.add_argument("--symbols", type=int, default=64)
    ap.add_argument("--candles", type=int, default=1_000_000)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--skip-mc", action="store_true",
                    help="skip the secondary Monte-Carlo measurement")
    ap.add_argument("--skip-train", action="store_true",
                    help="skip the secondary LSTM/PPO train measurements")
    ap.add_argument("--skip-continuous", action="store_true",
                    help="skip the continuous (unsegmented) backtest loop")
    ap.add_argument("--segments", type=int, default=64,
                    help="fitness time-CV segments per symbol (64 measured "
                         "fastest; the continuous seg=1 case is measured "
                         "separately via the time-parallel kernel pair)")
    return ap.parse_args()


def measure_train(device):
    """Secondary metrics: BASELINE configs #2 (LSTM predictor train) and
    #4 (PPO train) on this rank's GPU — driver-clocked via this run
    (tools/bench_train.py holds the full-size standalone variants)."""
    sys.path.insert(0, str(Path(__file__).resolve().parent / "tools"))
    from bench_train import bench_lstm_train, bench_ppo

    lstm = bench_lstm_train(nsym=32, T=1_000_000, steps=10, warmup=2)
    ppo = bench_ppo(n_envs=256, horizon=128, steps=4, warmup=1,
                    use_graph=True)
    return {
        "lstm_windows_per_sec": lstm["windows_per_sec"],
        "lstm_ms_per_step": lstm["ms_per_step"],
        "lstm_batch": lstm["config"]["batch"],
        "ppo_env_steps_per_sec": ppo["env_steps_per_sec"],
        "ppo_ms_per_step": ppo["ms_per_step"],
        "ppo_n_envs": ppo["config"]["n_envs"],
    }


def measure_mc(device, n_paths=10_000_000, n_assets=64, n_steps=30):
    """Secondary metric: correlated-GBM VaR paths/sec on this rank's GPU."""
    from ai_crypto_trader_amd.ops.montecarlo import mc_paths_gpu, risk_stats

    rho = 0.4
    corr = np.full((n_assets, n_assets), rho) + (1 - rho) * np.eye(n_assets)
    chol = np.linalg.cholesky(corr)
    mu = np.full(n_assets, 0.1)
    sigma = np.full(n_assets, 0.5)
    w = np.full(n_assets, 1.0 / n_assets)
    # warmup
    fv, _ = mc_paths_gpu(chol, mu, sigma, w, n_steps=n_steps,
                         n_paths=1_000_000, dt=1 / 252, seed=1, device=device)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    fv, dd = mc_paths_gpu(chol, mu, sigma, w, n_steps=n_steps,
                          n_paths=n_paths, dt=1 / 252, seed=2, device=device)
    torch.cuda.synchronize()
    dt_s = time.perf_counter() - t0
    stats = risk_stats(fv, v0=1.0)
    return {
        "mc_paths_per_sec": n_paths / dt_s,
        "mc_asset_steps_per_sec": n_paths * n_assets * n_steps / dt_s,
        "mc_var_95": stats["var_95"],
        "mc_seconds": dt_s,
    }


def main():
    args = parse_args()
    rank, world, device = pdist.init_distributed()
    on_gpu = device.type == "cuda"

    pop = args.pop
    nsym = args.symbols
    T = args.candles
    if not on_gpu:
        # CPU smoke mode (no GPU in this container): shrink so the
        # numpy reference engine finishes quickly; same code path.
        pop, nsym, T = min(pop, 8), min(nsym, 2), min(T, 4000)

    # deterministic synthetic market, identical on all ranks
    ohlcv = generate_ohlcv(T, nsym, seed=args.seed)
    candles = candles_chl_v(ohlcv)

    # fitness = time-segmented CV (see GAEngine
    # docstring: standard GA anti-overfit practice; also fills the chip —
    # pop x symbols alone is exactly 1 wave/SIMD). Candle-eval totals are
    # unchanged: every candle of every symbol is backtested each step.
    segments = args.segments if (on_gpu and T % args.segments == 0) else 1
    engine = GAEngine(
        candles, pop_per_rank=pop, rank=rank, world=world, device=device,
        seed=args.seed + 1, segments=segments,
    )

    for _ in range(args.warmup):
        engine.step()

    pdist.barrier()
    if on_gpu:
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        engine.step()
    if on_gpu:
        torch.cuda.synchronize()
    pdist.barrier()
    elapsed = time.perf_counter() - t0
    elapsed = pdist.all_reduce_max_scalar(elapsed, device)

    ms_per_step = elapsed / args.steps * 1000.0
    total_candle_evals = engine.candle_evals_per_step * world * args.steps
    value = total_candle_evals / elapsed

    # Continuous (unsegmented) fitness: the same GA generation loop with
    # the time-parallel kernel pair (ops/hip/backtest_tp.hip) — the
    # multi-year-single-history case, all ranks participating.
    continuous = {}
    if on_gpu and not args.skip_continuous and segments > 1:
        engine_c = GAEngine(
            candles, pop_per_rank=pop, rank=rank, world=world,
            device=device, seed=args.seed + 1, segments=1,
            continuous=True,
        )
        for _ in range(max(args.warmup, 1)):
            engine_c.step()
        pdist.barrier()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.steps):
            engine_c.step()
        torch.cuda.synchronize()
        pdist.barrier()
        el_c = pdist.all_reduce_max_scalar(
            time.perf_counter() - t0, device)
        continuous = {
            "candles_per_sec":
                engine_c.candle_evals_per_step * world * args.steps / el_c,
            "ms_per_step": el_c / args.steps * 1000.0,
            "fitness_segments": 1,
        }

    secondary = {}
    if on_gpu and rank == 0 and not args.skip_mc:
        secondary = measure_mc(device)
    if on_gpu and rank == 0 and not args.skip_train:
        secondary.update(measure_train(device))
    pdist.barrier()      # all ranks leave together (rank 0 runs MC above)

    if rank == 0:
        best_fit, _ = engine.best()
        out = {
            "metric": "backtested_1m_candles_per_sec_ga_pop_eval",
            "value": value,
            "unit": "candles/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": "ga_backtest_evolution",
                "global_batch": pop * world,
                "seq_len": T,
                "parallelism": f"dp{world}",
                "symbols": nsym,
                "pop_per_gpu": pop,
                "candle_evals_per_step_per_gpu":
                    engine.candle_evals_per_step,
                "fitness_segments": segments,
                "best_fitness": best_fit,
                "continuous": continuous,
                "secondary": secondary,
            },
        }
        print(json.dumps(out), flush=True)

    pdist.destroy()


if __name__ == "__main__":
    main()
