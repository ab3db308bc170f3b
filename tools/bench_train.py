#!/usr/bin/env python3
"""Training-workload benchmarks for the BASELINE.json configs beyond the
bench.py headline:

  #2  LSTM price predictor — 32 symbols x 1M 1m-candles resident in HBM,
      bf16 compute, windows (seq 60 x 9 features) sampled on-device,
      full train steps (fwd + bwd + Adam).
  #4  PPO agent — 256 synthetic-market envs stepped by the HIP env
      kernel, GAE on the HIP reverse-scan kernel, full train_step.

  python tools/bench_train.py [--workload lstm|ppo|all]
"""

from __future__ import annotations

import argparse
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch


def build_features_gpu(candles: torch.Tensor) -> torch.Tensor:
    """(nsym, T, 4) cuda -> (nsym, T, 9) features, all on device
    (the GPU twin of services/neural_network.build_features)."""
    from ai_crypto_trader_amd.ops.indicators import indicators_gpu

    ind = indicators_gpu(candles)                  # (nsym, T, 13)
    close = candles[..., 0]
    ret1 = torch.zeros_like(close)
    ret1[:, 1:] = close[:, 1:] / close[:, :-1] - 1.0
    bb_pos = (close - ind[..., 8]) / torch.clamp(
        ind[..., 7] - ind[..., 8], min=1e-9)
    feats = torch.stack([
        close, candles[..., 1], candles[..., 2], candles[..., 3],
        ind[..., 5] / 100.0, ind[..., 4], bb_pos, ret1 * 100.0,
        ind[..., 9],
    ], dim=-1)
    lo = feats.amin(dim=(0, 1), keepdim=True)
    hi = feats.amax(dim=(0, 1), keepdim=True)
    return (feats - lo) / torch.clamp(hi - lo, min=1e-9)


def bench_lstm_train(nsym=32, T=1_000_000, seq=60, batch=16_384,
                     steps=20, warmup=3, seed=0):
    from ai_crypto_trader_amd.data.synthetic import (
        candles_chl_v, generate_ohlcv,
    )
    from ai_crypto_trader_amd.models.zoo import create_model

    candles = torch.from_numpy(
        candles_chl_v(generate_ohlcv(T, nsym, seed=seed))).cuda()
    feats = build_features_gpu(candles)            # (nsym, T, 9) resident
    close = candles[..., 0]
    target = torch.zeros_like(close)
    target[:, :-1] = (close[:, 1:] / close[:, :-1] - 1.0) * 100.0

    model = create_model("lstm", n_features=9).cuda()
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    g = torch.Generator(device="cuda").manual_seed(seed)

    def sample_batch():
        s = torch.randint(0, nsym, (batch,), device="cuda", generator=g)
        t0 = torch.randint(0, T - seq - 1, (batch,), device="cuda",
                           generator=g)
        idx = t0[:, None] + torch.arange(seq, device="cuda")[None]
        x = feats[s[:, None], idx]                 # (B, seq, 9)
        y = target[s, t0 + seq]
        return x, y

    def step():
        x, y = sample_batch()
        opt.zero_grad(set_to_none=True)
        loss = ((model(x) - y) ** 2).mean()
        loss.backward()
        opt.step()
        return loss

    for _ in range(warmup):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        loss = step()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / steps
    return {
        "workload": "lstm_train (BASELINE config #2)",
        "ms_per_step": dt * 1e3,
        "windows_per_sec": batch / dt,
        "cells_per_sec": batch * seq / dt,
        "final_loss": float(loss.detach()),
        "config": {"nsym": nsym, "T": T, "seq": seq, "batch": batch,
                   "dtype": "bf16 (recurrent cells) + f32 master",
                   "resident_bytes": int(feats.numel() * 4 +
                                         candles.numel() * 4)},
    }


def bench_ppo(n_envs=256, horizon=128, steps=6, warmup=2, seed=0,
              use_graph=False):
    from ai_crypto_trader_amd.data.synthetic import (
        candles_chl_v, generate_ohlcv,
    )
    from ai_crypto_trader_amd.models.rl import PPOAgent, TradingVecEnv

    market = torch.from_numpy(
        candles_chl_v(generate_ohlcv(500_000, 16, seed=seed))).cuda()
    env = TradingVecEnv(market, n_envs=n_envs, ep_len=1024, seed=seed)
    env.reset()
    agent = PPOAgent("cuda", seed=seed, use_graph=use_graph)

    for _ in range(warmup):
        agent.train_step(env, horizon)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    stats = {}
    for _ in range(steps):
        stats = agent.train_step(env, horizon)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / steps
    return {
        "workload": "ppo_train (BASELINE config #4, 1-GPU shard)",
        "rollout_graph": agent.use_graph,
        "ms_per_step": dt * 1e3,
        "env_steps_per_sec": n_envs * horizon / dt,
        "stats": stats,
        "config": {"n_envs": n_envs, "horizon": horizon,
                   "note": "DP grad all-reduce engages under torchrun"},
    }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--workload", default="all",
                    choices=["all", "lstm", "ppo"])
    args = ap.parse_args()
    assert torch.cuda.is_available()
    if args.workload in ("all", "lstm"):
        print(json.dumps(bench_lstm_train()), flush=True)
    if args.workload in ("all", "ppo"):
        print(json.dumps(bench_ppo()), flush=True)
        print(json.dumps(bench_ppo(use_graph=True)), flush=True)


if __name__ == "__main__":
    main()
