#!/usr/bin/env python3
"""Per-kernel microbenchmarks (GPU): backtest, Monte-Carlo, covariance,
indicators, LSTM fwd/bwd, GA evolve, env step, GAE.

Used for rocprofv3 runs and optimization A/Bs:
  python tools/bench_kernels.py [--kernel backtest] [--reps 5]
Prints one JSON line per kernel with ms + domain throughput.
"""

from __future__ import annotations

import argparse
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import numpy as np
import torch


def timed(fn, reps, warmup=2):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps


def bench_backtest(reps):
    from ai_crypto_trader_amd.backtesting.strategy import random_population
    from ai_crypto_trader_amd.data.synthetic import (
        candles_chl_v, generate_ohlcv,
    )
    from ai_crypto_trader_amd.ops.backtest import run_backtest_gpu

    nsym, T, P = 64, 1_000_000, 1024
    candles = torch.from_numpy(
        candles_chl_v(generate_ohlcv(T, nsym, seed=0))).cuda()
    pop = torch.from_numpy(random_population(P, seed=1)).cuda()
    dt = timed(lambda: run_backtest_gpu(candles, pop), reps)
    for seg in (4, 8, 16):
        cseg = candles.reshape(nsym * seg, T // seg, 4).contiguous()
        dts = timed(lambda: run_backtest_gpu(cseg, pop), reps)
        print(json.dumps({"kernel": f"backtest_seg{seg}", "ms": dts * 1e3,
                          "candles_per_sec": P * nsym * T / dts}),
              flush=True)
    # continuous (unsegmented) time-parallel pair (backtest_tp.hip)
    from ai_crypto_trader_amd.ops.backtest import (
        run_backtest_continuous_gpu,
    )
    dtc = timed(lambda: run_backtest_continuous_gpu(candles, pop), reps)
    print(json.dumps({"kernel": "backtest_continuous(tp)",
                      "ms": dtc * 1e3,
                      "candles_per_sec": P * nsym * T / dtc}), flush=True)
    return {"kernel": "backtest", "ms": dt * 1e3,
            "candles_per_sec": P * nsym * T / dt,
            "config": {"nsym": nsym, "T": T, "P": P}}


def bench_mc(reps):
    from ai_crypto_trader_amd.ops.montecarlo import mc_paths_gpu

    A, steps, paths = 64, 30, 10_000_000
    rho = 0.4
    corr = np.full((A, A), rho) + (1 - rho) * np.eye(A)
    chol = np.linalg.cholesky(corr)
    mu = np.full(A, 0.1)
    sig = np.full(A, 0.5)
    w = np.full(A, 1.0 / A)
    dt = timed(lambda: mc_paths_gpu(chol, mu, sig, w, n_steps=steps,
                                    n_paths=paths, dt=1 / 252, seed=2,
                                    use_mfma=False),
               reps)
    dt_m = timed(lambda: mc_paths_gpu(chol, mu, sig, w, n_steps=steps,
                                      n_paths=paths, dt=1 / 252, seed=2,
                                      use_mfma=True),
                 reps)
    print(json.dumps({"kernel": "mc_paths_mfma", "ms": dt_m * 1e3,
                      "paths_per_sec": paths / dt_m,
                      "asset_steps_per_sec": paths * A * steps / dt_m,
                      "mfma_tflops_bf16":
                          paths * steps * A * A * 2 / dt_m / 1e12}),
          flush=True)
    return {"kernel": "mc_paths", "ms": dt * 1e3,
            "paths_per_sec": paths / dt,
            "asset_steps_per_sec": paths * A * steps / dt,
            "gflops_f32": paths * steps * A * A * 2 / dt / 1e9,
            "config": {"A": A, "steps": steps, "paths": paths}}


def bench_mc_bootstrap(reps):
    from ai_crypto_trader_amd.ops.montecarlo import mc_bootstrap_gpu

    rng = np.random.default_rng(3)
    T_hist, A = 365, 64
    lr = (rng.standard_normal((T_hist, A)) * 0.01).astype(np.float32)
    w = np.full(A, 1.0 / A)
    n_paths, steps = 10_000_000, 30
    dt = timed(lambda: mc_bootstrap_gpu(lr, w, n_steps=steps,
                                        n_paths=n_paths, seed=1), reps)
    return {"kernel": "mc_bootstrap", "ms": dt * 1e3,
            "paths_per_sec": n_paths / dt,
            "config": {"paths": n_paths, "assets": A, "steps": steps,
                       "t_hist": T_hist}}


def bench_cov(reps):
    from ai_crypto_trader_amd.ops.covar import cov_gpu

    T, N = 1_000_000, 64
    X = torch.randn(T, N, device="cuda") * 0.01
    dt = timed(lambda: cov_gpu(X), reps)
    return {"kernel": "cov", "ms": dt * 1e3,
            "gflops_f32": 2 * T * N * N / dt / 1e9,
            "config": {"T": T, "N": N}}


def bench_indicators(reps):
    from ai_crypto_trader_amd.data.synthetic import (
        candles_chl_v, generate_ohlcv,
    )
    from ai_crypto_trader_amd.ops.indicators import indicators_gpu

    nsym, T = 64, 1_000_000
    candles = torch.from_numpy(
        candles_chl_v(generate_ohlcv(T, nsym, seed=3))).cuda()
    dt = timed(lambda: indicators_gpu(candles), reps)
    return {"kernel": "indicators", "ms": dt * 1e3,
            "candles_per_sec": nsym * T / dt,
            "config": {"nsym": nsym, "T": T}}


def bench_lstm(reps):
    from ai_crypto_trader_amd.models.lstm import FusedLSTMLayer

    T, B, F, H = 60, 16_384, 9, 64
    layer = FusedLSTMLayer(F, H).cuda()
    x = torch.randn(T, B, F, device="cuda")

    def fwd():
        return layer(x)

    dt_f = timed(fwd, reps)

    def fwdbwd():
        out = layer(x)
        out.float().pow(2).mean().backward()

    dt_fb = timed(fwdbwd, reps)
    cells = T * B
    return {"kernel": "lstm_seq", "fwd_ms": dt_f * 1e3,
            "fwdbwd_ms": dt_fb * 1e3,
            "cells_per_sec_fwd": cells / dt_f,
            "cells_per_sec_fwdbwd": cells / dt_fb,
            "config": {"T": T, "B": B, "H": H}}


def bench_gru(reps):
    from ai_crypto_trader_amd.models.gru import FusedGRULayer

    T, B, H = 60, 16384, 64
    layer = FusedGRULayer(9, H).cuda()
    x = torch.randn(T, B, 9, device="cuda")
    dt_f = timed(lambda: layer(x), reps)

    def fwdbwd():
        layer.zero_grad()
        layer(x).float().pow(2).mean().backward()

    dt_b = timed(fwdbwd, reps)
    return {"kernel": "gru_seq", "fwd_ms": dt_f * 1e3,
            "fwdbwd_ms": dt_b * 1e3,
            "cells_per_sec_fwd": T * B / dt_f,
            "config": {"T": T, "B": B, "H": H}}


def bench_env(reps):
    from ai_crypto_trader_amd.data.synthetic import (
        candles_chl_v, generate_ohlcv,
    )
    from ai_crypto_trader_amd.models.rl import TradingVecEnv

    market = torch.from_numpy(
        candles_chl_v(generate_ohlcv(200_000, 8, seed=4))).cuda()
    env = TradingVecEnv(market, n_envs=4096, ep_len=1024, seed=5)
    env.reset()
    actions = torch.randint(0, 3, (4096,), device="cuda")

    def steps():
        for _ in range(64):
            env.step(actions)

    dt = timed(steps, reps)
    return {"kernel": "env_step", "ms_per_64steps": dt * 1e3,
            "env_steps_per_sec": 4096 * 64 / dt,
            "config": {"n_envs": 4096}}


def bench_attn(reps):
    from ai_crypto_trader_amd.models.attention import attn_fwd_hip

    BH, S, D = 16384, 60, 16        # transformer predictor shape (4096x4)
    q = torch.randn(BH, S, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    dt = timed(lambda: attn_fwd_hip(q, k, v, D ** -0.5), reps)
    return {"kernel": "attn_fwd", "ms": dt * 1e3,
            "seqs_per_sec": BH / dt,
            "config": {"BH": BH, "S": S, "D": D}}


def bench_gae(reps):
    from ai_crypto_trader_amd.models.rl import gae_gpu

    T, E = 512, 8192
    rew = torch.randn(T, E, device="cuda")
    val = torch.randn(T + 1, E, device="cuda")
    dn = (torch.rand(T, E, device="cuda") < 0.01).float()
    dt = timed(lambda: gae_gpu(rew, val, dn), reps)
    return {"kernel": "gae", "ms": dt * 1e3,
            "steps_per_sec": T * E / dt, "config": {"T": T, "E": E}}


BENCHES = {
    "backtest": bench_backtest, "mc": bench_mc, "mc_bootstrap": bench_mc_bootstrap, "cov": bench_cov,
    "indicators": bench_indicators, "lstm": bench_lstm, "gru": bench_gru, "env": bench_env,
    "gae": bench_gae, "attn": bench_attn,
}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--kernel", default="all",
                    choices=["all", *BENCHES])
    ap.add_argument("--reps", type=int, default=3)
    args = ap.parse_args()
    assert torch.cuda.is_available()
    names = list(BENCHES) if args.kernel == "all" else [args.kernel]
    for n in names:
        try:
            print(json.dumps(BENCHES[n](args.reps)), flush=True)
        except Exception as e:
            print(json.dumps({"kernel": n, "error": repr(e)}), flush=True)


if __name__ == "__main__":
    main()
