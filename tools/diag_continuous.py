#!/usr/bin/env python3
"""Diagnose the time-parallel continuous backtest against the CPU
engine at bit level: compares flag bitmaps (GPU, nshards=1 and sharded)
against CPU-derived flags, and isolates phase-2 (trades kernel) by
feeding it CPU-exact flags. Prints where the first differences are."""

from __future__ import annotations

import json
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import numpy as np
import torch


def unpack_flags(t: torch.Tensor, T: int) -> np.ndarray:
    """(nsym, nwords, P) int64 -> (P, nsym, T) bool."""
    a = t.cpu().numpy().view(np.uint64)
    nsym, nwords, P = a.shape
    bits = np.unpackbits(
        a.view(np.uint8).reshape(nsym, nwords, P, 8),
        axis=-1, bitorder="little",
    )  # (nsym, nwords, P, 64)
    bits = bits.transpose(2, 0, 1, 3).reshape(P, nsym, nwords * 64)
    return bits[:, :, :T].astype(bool)


def pack_flags(f: np.ndarray) -> torch.Tensor:
    """(P, nsym, T) bool -> (nsym, nwords, P) int64."""
    P, nsym, T = f.shape
    nwords = (T + 63) // 64
    pad = np.zeros((P, nsym, nwords * 64 - T), bool)
    bits = np.concatenate([f, pad], axis=-1)
    by = np.packbits(bits.reshape(P, nsym, nwords, 64),
                     axis=-1, bitorder="little")       # (P,nsym,nwords,8)
    w = by.view(np.uint64)[..., 0]                     # (P, nsym, nwords)
    return torch.from_numpy(
        np.ascontiguousarray(w.transpose(1, 2, 0)).view(np.int64))


def main():
    from ai_crypto_trader_amd.backtesting.engine_cpu import run_backtest_cpu
    from ai_crypto_trader_amd.backtesting.strategy import random_population
    from ai_crypto_trader_amd.data.synthetic import (
        candles_chl_v, generate_ohlcv,
    )
    from ai_crypto_trader_amd.ops import require_hip_ops
    from ai_crypto_trader_amd.ops.backtest import NMETRIC

    ops = require_hip_ops()
    dev = torch.device("cuda:0")
    print("[diag] ops loaded", flush=True)
    T, nsym, P = 18432, 2, 32
    candles = candles_chl_v(generate_ohlcv(T, nsym, seed=21))
    pop = random_population(P, seed=9)
    print("[diag] running CPU reference...", flush=True)
    m_cpu, nets = run_backtest_cpu(candles, pop, record_net=True)
    print("[diag] CPU done", flush=True)
    entry_v = pop[:, 10].astype(np.int32)[:, None, None]
    exit_v = pop[:, 11].astype(np.int32)[:, None, None]
    ef_cpu = nets >= entry_v
    xf_cpu = nets <= -exit_v

    c_t = torch.from_numpy(candles).to(dev)
    p_t = torch.from_numpy(pop).to(dev)
    nwords = (T + 63) // 64
    eflags = torch.zeros((nsym, nwords, P), dtype=torch.int64, device=dev)
    xflags = torch.zeros_like(eflags)
    metrics = torch.empty((P, nsym, NMETRIC), dtype=torch.float32,
                          device=dev)
    stream = torch.cuda.current_stream(dev).cuda_stream

    out = {}
    for ns in (1, 4):
        eflags.zero_(); xflags.zero_()
        ops.bt_flags(c_t.data_ptr(), p_t.data_ptr(), eflags.data_ptr(),
                     xflags.data_ptr(), nsym, T, P, ns, 2048, 0, ns,
                     stream)
        torch.cuda.synchronize()
        print(f"[diag] bt_flags ns={ns} ok", flush=True)
        ef = unpack_flags(eflags, T)
        xf = unpack_flags(xflags, T)
        ed = np.argwhere(ef != ef_cpu)
        xd = np.argwhere(xf != xf_cpu)
        out[f"nshards{ns}"] = {
            "entry_diffs": int(len(ed)), "exit_diffs": int(len(xd)),
            "first_entry_diffs": ed[:5].tolist(),
            "first_exit_diffs": xd[:5].tolist(),
        }

    # phase 2 isolation: CPU-exact flags -> trades kernel
    e_cpu_t = pack_flags(ef_cpu).to(dev)
    x_cpu_t = pack_flags(xf_cpu).to(dev)
    ops.bt_trades(c_t.data_ptr(), p_t.data_ptr(), e_cpu_t.data_ptr(),
                  x_cpu_t.data_ptr(), metrics.data_ptr(), nsym, T, P,
                  1.0, 0, nsym, 0, T, 0, stream)
    torch.cuda.synchronize()
    print("[diag] bt_trades(cpu flags) ok", flush=True)
    m_tr = metrics.cpu().numpy()
    d = np.argwhere(m_tr != m_cpu)
    out["trades_with_cpu_flags"] = {
        "metric_diffs": int(len(d)),
        "diff_cols": sorted(set(int(k) for _, _, k in d)),
        "first": [
            [int(i), int(s), int(k), float(m_cpu[i, s, k]),
             float(m_tr[i, s, k])]
            for i, s, k in d[:8]
        ],
    }
    if len(d):
        # bit-level finalize introspection for the first differing lane:
        # recompute sharpe from the GPU's own accumulator columns in
        # numpy f32 and show where it lands vs CPU col8 and GPU col8
        i, s, _ = d[0]
        f32 = np.float32
        acc = m_tr[i, s]
        n = f32(T)
        mean = f32(acc[6] / n)
        var = np.maximum(f32(acc[7] / n) - mean * mean, f32(0.0))
        std = np.sqrt(var)
        den = np.maximum(std, f32(1e-9))
        q = f32(mean / den)
        ann = f32(np.sqrt(np.float64(525600.0)))
        sharpe_np = f32(q * ann)

        def bits(x):
            return hex(np.float32(x).view(np.uint32))

        out["finalize_introspect"] = {
            "lane": [int(i), int(s)],
            "acc_equal_cpu": bool(np.array_equal(acc[:8], m_cpu[i, s, :8])),
            "sum_ret": bits(acc[6]), "sum_ret2": bits(acc[7]),
            "mean": bits(mean), "var": bits(var), "std": bits(std),
            "q_mean_over_std": bits(q),
            "ann_np": bits(ann), "ann_kernel": hex(0x44353ee5),
            "sharpe_np_from_gpu_acc": bits(sharpe_np),
            "sharpe_cpu": bits(m_cpu[i, s, 8]),
            "sharpe_gpu": bits(m_tr[i, s, 8]),
        }
    print(json.dumps(out, indent=1))


if __name__ == "__main__":
    main()
