#!/usr/bin/env python3
"""Serving-path latency on GPU: in-process ASGI calls against serve.py.

Measures what a serving deployment sees per request (no network): model
predict (fused LSTM inference path), analyzer decision, and the GPU
backtest endpoint.

    python tools/bench_serve.py [--reps 50]
"""

from __future__ import annotations

import argparse
import asyncio
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import numpy as np


async def run(reps: int):
    import httpx
    import torch

    from ai_crypto_trader_amd.data.synthetic import (
        candles_chl_v, generate_ohlcv,
    )
    from serve import build_server

    device = "cuda:0" if torch.cuda.is_available() else "cpu"
    app, _nn = build_server("models_store_bench", device)
    candles = candles_chl_v(generate_ohlcv(2000, 1, seed=4))[0]
    payload_predict = {"symbol": "BTCUSDC", "candles": candles.tolist()}
    payload_analyze = {
        "symbol": "BTCUSDC", "current_price": 100.0, "avg_volume": 5.0,
        "rsi": 28.0, "stoch_k": 15.0, "williams_r": -85.0,
        "price_change_1m": 0.2, "price_change_5m": 0.4,
        "price_change_15m": 0.1, "trend": "uptrend", "trend_strength": 60.0,
    }
    payload_backtest = {"symbol": "BTCUSDC", "candles": candles.tolist(),
                        "strategy": "momentum"}

    out = {}
    transport = httpx.ASGITransport(app=app)
    async with httpx.AsyncClient(transport=transport,
                                 base_url="http://b") as c:
        # warm (includes cold-start training for predict)
        await c.post("/predict", json=payload_predict)
        for name, path, body in [
            ("predict", "/predict", payload_predict),
            ("analyze", "/analyze", payload_analyze),
            ("backtest", "/backtest", payload_backtest),
        ]:
            await c.post(path, json=body)
            lat = []
            for _ in range(reps):
                t0 = time.perf_counter()
                r = await c.post(path, json=body)
                lat.append(time.perf_counter() - t0)
                assert r.status_code == 200
            lat = np.asarray(lat) * 1e3
            out[name] = {"p50_ms": round(float(np.percentile(lat, 50)), 3),
                         "p95_ms": round(float(np.percentile(lat, 95)), 3)}
    out["device"] = device
    print(json.dumps(out))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--reps", type=int, default=50)
    args = ap.parse_args()
    asyncio.run(run(args.reps))


if __name__ == "__main__":
    main()
