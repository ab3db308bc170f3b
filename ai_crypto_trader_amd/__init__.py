"""ai_crypto_trader_amd — MI355X-native quant crypto trading + backtesting framework.

A from-scratch rebuild of the capabilities of `zd87pl/ai-crypto-trader`
 designed MI355X-first:

- compute plane: hand-written HIP/CDNA4 (gfx950) kernels for the per-candle
  backtest engine, rolling technical indicators, Monte-Carlo GBM path
  simulation, portfolio covariance/VaR (MFMA GEMM), genetic-algorithm
  evolution ops and recurrent-cell / GAE reductions, driven from
  PyTorch-ROCm; data-parallel sharding over RCCL/xGMI
  (one process per GPU, torch.distributed backend "nccl" == RCCL).
- control plane: asyncio services around a Redis-schema-compatible message
  bus reproducing the reference's pub/sub channels and key-value state
  (SURVEY.md §1.1).

Package layout:
  ops/          HIP kernels + torch extension bindings + CPU reference impls
  backtesting/  engines (CPU reference + GPU), strategies, metrics, analyzers
  models/       LSTM/GRU price predictors, DQN/PPO agents, regime detectors
  parallel/     torch.distributed (RCCL) shard manager and collectives
  bus/          message bus with the reference channel/payload schema
  data/         synthetic OHLCV generation + historical data management
  services/     control-plane services (market monitor, analyzer, risk, ...)
  utils/        circuit breaker, rate limiter, metrics, exchange interface
"""

__version__ = "0.1.0"
