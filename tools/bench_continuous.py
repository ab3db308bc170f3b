#!/usr/bin/env python3
"""Continuous-vs-segmented backtest throughput at the flagship shape
(pop=1024 x 64 symbols x 1M candles), with per-kernel breakdown of the
time-parallel path (flags vs trades) via CUDA events.

  python tools/bench_continuous.py [--pop 1024] [--symbols 64]
      [--candles 1000000] [--iters 5] [--nshards 0]
"""

from __future__ import annotations

import argparse
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--pop", type=int, default=1024)
    ap.add_argument("--symbols", type=int, default=64)
    ap.add_argument("--candles", type=int, default=1_000_000)
    ap.add_argument("--iters", type=int, default=5)
    ap.add_argument("--nshards", type=int, default=0)
    ap.add_argument("--tail", type=int, default=2048)
    ap.add_argument("--skip-seg", action="store_true")
    args = ap.parse_args()

    from ai_crypto_trader_amd.backtesting.strategy import random_population
    from ai_crypto_trader_amd.data.synthetic import (
        candles_chl_v, generate_ohlcv,
    )
    from ai_crypto_trader_amd.ops import require_hip_ops
    from ai_crypto_trader_amd.ops.backtest import (
        pick_nshards, run_backtest_continuous_gpu, run_backtest_gpu,
    )

    assert torch.cuda.is_available()
    dev = torch.device("cuda:0")
    ops = require_hip_ops()

    nsym, T, P = args.symbols, args.candles, args.pop
    candles = candles_chl_v(generate_ohlcv(T, nsym, seed=0))
    c_t = torch.from_numpy(candles).to(dev)
    pop = random_population(P, seed=1)
    p_t = torch.from_numpy(pop).to(dev)
    evals = P * nsym * T
    out = {"shape": {"pop": P, "nsym": nsym, "T": T}}

    def timeit(fn, iters=args.iters):
        fn()  # warm
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters

    nshards = args.nshards or pick_nshards(nsym, T, P, tail=args.tail)
    out["nshards"] = nshards

    # full continuous pipeline (auto time-group overlap)
    dt = timeit(lambda: run_backtest_continuous_gpu(
        c_t, p_t, nshards=nshards, tail=args.tail))
    out["continuous"] = {"s": dt, "gcandles_per_s": evals / dt / 1e9}
    # activity-sorted lanes: permute the population by measured trade
    # count (what GAEngine does from generation 2 on) and re-measure
    m0 = run_backtest_continuous_gpu(c_t, p_t, nshards=nshards,
                                     tail=args.tail)
    order = torch.argsort(m0[..., 1].sum(dim=1))
    p_sorted = p_t[order].contiguous()
    dts = timeit(lambda: run_backtest_continuous_gpu(
        c_t, p_sorted, nshards=nshards, tail=args.tail))
    out["continuous_activity_sorted"] = {
        "s": dts, "gcandles_per_s": evals / dts / 1e9}

    # time-group sweep (flags/trades overlap granularity)
    gsweep = {}
    for g in (1, 2, 4, 8, 16):
        if g > nshards:
            continue
        d = timeit(lambda g=g: run_backtest_continuous_gpu(
            c_t, p_t, nshards=nshards, tail=args.tail, time_groups=g),
            iters=3)
        gsweep[g] = round(evals / d / 1e9, 1)
    out["time_group_sweep_gcps"] = gsweep

    # per-kernel breakdown
    from ai_crypto_trader_amd.ops.backtest import _flag_cache
    (eflags, xflags, carry), = [v for v in _flag_cache.values()]
    metrics = torch.empty((P, nsym, 10), dtype=torch.float32, device=dev)
    stream = torch.cuda.current_stream(dev).cuda_stream

    dt_f = timeit(lambda: ops.bt_flags(
        c_t.data_ptr(), p_t.data_ptr(), eflags.data_ptr(),
        xflags.data_ptr(), nsym, T, P, nshards, args.tail, 0, nshards,
        stream))
    dt_t = timeit(lambda: ops.bt_trades(
        c_t.data_ptr(), p_t.data_ptr(), eflags.data_ptr(),
        xflags.data_ptr(), metrics.data_ptr(), nsym, T, P, 1.0, 0, nsym,
        0, T, 0, stream))
    out["flags_kernel"] = {"s": dt_f, "gcandles_per_s": evals / dt_f / 1e9}
    out["trades_kernel"] = {"s": dt_t, "gcandles_per_s": evals / dt_t / 1e9}

    # shard-count sweep on the flags kernel
    sweep = {}
    for s in (1, 2, 4, 8, 16, 32):
        if s > 1 and (T // s) < max(4096, 4 * args.tail):
            continue
        d = timeit(lambda s=s: ops.bt_flags(
            c_t.data_ptr(), p_t.data_ptr(), eflags.data_ptr(),
            xflags.data_ptr(), nsym, T, P, s, args.tail, 0, s, stream),
            iters=3)
        sweep[s] = round(evals / d / 1e9, 1)
    out["flags_shard_sweep_gcps"] = sweep

    if not args.skip_seg:
        # classic kernel at seg=1 (the round-1 weak case) and seg=64
        dt1 = timeit(lambda: run_backtest_gpu(c_t, p_t), iters=3)
        out["classic_seg1"] = {"s": dt1, "gcandles_per_s": evals / dt1 / 1e9}
        if T % 64 == 0:
            c64 = torch.from_numpy(
                candles.reshape(nsym * 64, T // 64, 4)).to(dev)
            dt64 = timeit(lambda: run_backtest_gpu(c64, p_t), iters=3)
            out["classic_seg64"] = {
                "s": dt64, "gcandles_per_s": evals / dt64 / 1e9}

    print(json.dumps(out, indent=1), flush=True)


if __name__ == "__main__":
    main()
