#!/usr/bin/env python3
"""Model-serving entrypoint: low-latency prediction + analysis API.

Loads trained predictor checkpoints (services/neural_network save format)
and serves:
  POST /predict        {symbol, candles: [[close,high,low,vol],...]}
                       -> next-change prediction (fused-kernel inference
                          path: no backward saves, models/lstm.py)
  POST /analyze        market-update dict -> LocalAnalyst decision
  POST /scan           {symbol: candles} -> ranked opportunities (one
                          batched GPU indicator launch)
  POST /backtest       {candles, params?} -> stats (GPU kernel)
  GET  /healthz, /metrics

  python serve.py [--port 8060] [--model-dir models_store] [--gpu]
"""

from __future__ import annotations

import argparse

import numpy as np


def build_server(model_dir: str, device: str):
    import torch
    from fastapi import FastAPI
    from fastapi.responses import PlainTextResponse

    from ai_crypto_trader_amd.analysis import CryptoScanner
    from ai_crypto_trader_amd.bus.message_bus import InProcessBus
    from ai_crypto_trader_amd.config import AppConfig
    from ai_crypto_trader_amd.services.analyzer import LocalAnalyst
    from ai_crypto_trader_amd.services.neural_network import (
        NeuralNetworkService,
    )
    from ai_crypto_trader_amd.utils.metrics import GpuTimer, get_metrics

    app = FastAPI(title="ai-crypto-trader-amd serving")
    cfg = AppConfig.load()
    nn = NeuralNetworkService(InProcessBus(), cfg, device=device)
    n_loaded = nn.load(model_dir)
    analyst = LocalAnalyst()
    scanner = CryptoScanner()
    metrics = get_metrics("serving")

    @app.get("/healthz")
    async def healthz():
        return {"ok": True, "models": sorted(nn.models),
                "device": device}

    @app.get("/metrics", response_class=PlainTextResponse)
    async def prom():
        return metrics.export().decode()

    @app.post("/predict")
    async def predict(body: dict):
        sym = body["symbol"]
        candles = np.asarray(body["candles"], np.float32)
        if sym not in nn.models:
            # cold start: train a quick model on the provided history
            nn.train(sym, candles, epochs=2)
        with GpuTimer(metrics, "serve_predict"):
            with torch.no_grad():
                out = nn.predict(sym, candles)
        return out or {"error": "insufficient history"}

    @app.post("/analyze")
    async def analyze(body: dict):
        with GpuTimer(metrics, "serve_analyze"):
            return analyst.analyze(body)

    @app.post("/scan")
    async def scan(body: dict):
        market = {s: np.asarray(c, np.float32)
                  for s, c in body.items()}
        with GpuTimer(metrics, "serve_scan"):
            return scanner.scan_market(market, top_k=body.get("_top_k", 10)
                                       if isinstance(body.get("_top_k"),
                                                     int) else 10)

    @app.post("/backtest")
    async def backtest(body: dict):
        from ai_crypto_trader_amd.backtesting.engine_cpu import (
            run_backtest_cpu,
        )
        from ai_crypto_trader_amd.backtesting.engine import metrics_to_stats
        from ai_crypto_trader_amd.backtesting.strategy import (
            clip_params, dict_to_params,
        )

        candles = np.asarray(body["candles"], np.float32)[None]
        vec = clip_params(dict_to_params(body.get("params", {}))[None])
        if device.startswith("cuda"):
            from ai_crypto_trader_amd.ops.backtest import run_backtest_gpu

            with GpuTimer(metrics, "serve_backtest"):
                m = run_backtest_gpu(
                    torch.from_numpy(candles).to(device),
                    torch.from_numpy(vec).to(device)).cpu().numpy()
        else:
            m = run_backtest_cpu(candles, vec)
        return metrics_to_stats(m[0, 0], candles.shape[1])

    return app, nn


def main():
    import torch
    import uvicorn

    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int, default=8060)
    ap.add_argument("--model-dir", default="models_store")
    ap.add_argument("--gpu", action="store_true")
    args = ap.parse_args()
    device = "cuda:0" if (args.gpu and torch.cuda.is_available()) else "cpu"
    app, _ = build_server(args.model_dir, device)
    uvicorn.run(app, host="127.0.0.1", port=args.port, log_level="info")


if __name__ == "__main__":
    main()
