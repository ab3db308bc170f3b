"""Single-service entrypoint — the reference's one-container-per-service
topology (docker-compose.yml:30-304: 11 app containers coordinating only
through Redis).

    python -m ai_crypto_trader_amd.services.runner --service market_monitor \
        --bus redis --redis-host redis --minutes 0

Each container runs exactly one service against the shared Redis bus
(RedisBus is wire-compatible with the reference's channel schema); the
single-process all-in-one alternative is run_trader.py. Market data is the
seeded synthetic feed (offline parity of the reference's Binance
websocket); services that need an exchange get the fake fill engine —
plug a real exchange behind utils/exchange.ExchangeFactory for live use.
"""

from __future__ import annotations

import argparse
import asyncio

from ..bus.message_bus import make_bus
from ..config import AppConfig, set_config
from ..data.feed import SyntheticFeed
from ..data.synthetic import candles_chl_v, generate_ohlcv
from ..utils.exchange import ExchangeFactory


def build_service(name: str, bus, cfg: AppConfig, args):
    """Construct one service by its reference-topology name."""
    device = args.device
    symbols = cfg.trading.symbols
    market = candles_chl_v(
        generate_ohlcv(args.candles, len(symbols), seed=cfg.seed))

    def feed():
        return SyntheticFeed(market, symbols, start=0, speed=args.speed)

    def exchange():
        name = getattr(cfg.trading, "exchange", "fake")
        if name == "binance":
            import os
            return ExchangeFactory.create_exchange(
                "binance", quote=cfg.trading.quote_asset,
                api_key=os.environ.get("BINANCE_API_KEY", ""),
                api_secret=os.environ.get("BINANCE_API_SECRET", ""))
        return ExchangeFactory.create_exchange(
            "fake", fee_rate=cfg.trading.fee_rate,
            quote=cfg.trading.quote_asset)

    if name == "market_monitor":
        from .market_monitor import MarketMonitorService
        return MarketMonitorService(bus, feed(), cfg)
    if name == "ai_analyzer":
        from .analyzer import AnalyzerService
        return AnalyzerService(bus, cfg)
    if name == "portfolio_risk":
        from .portfolio_risk import PortfolioRiskService
        return PortfolioRiskService(bus, cfg)
    if name == "monte_carlo":
        from .monte_carlo import MonteCarloService
        return MonteCarloService(bus, cfg)
    if name == "market_regime":
        from .market_regime import MarketRegimeService
        return MarketRegimeService(bus, cfg, device=device)
    if name == "neural_network":
        from .neural_network import NeuralNetworkService
        return NeuralNetworkService(bus, cfg, device=device)
    if name == "strategy_evolution":
        from .strategy_evolution import StrategyEvolutionService
        return StrategyEvolutionService(bus, cfg, candles=market,
                                        device=device)
    if name == "trade_executor":
        from .trade_executor import TradeExecutorService
        return TradeExecutorService(bus, exchange(), cfg)
    if name == "social_monitor":
        from .social import SocialMonitorService
        src = None
        if getattr(cfg.social, "source", "synthetic") == "lunarcrush":
            import os

            from ..live.social_news import LunarCrushSocialSource
            src = LunarCrushSocialSource(
                api_key=os.environ.get("LUNARCRUSH_API_KEY", ""))
        return SocialMonitorService(bus, cfg, source=src)
    if name == "enhanced_social":
        from .social import EnhancedSocialMonitorService
        return EnhancedSocialMonitorService(bus, cfg)
    if name == "social_risk":
        from .social import SocialRiskAdjuster
        return SocialRiskAdjuster(bus, cfg)
    if name == "news_analysis":
        from .news import NewsAnalysisService
        src = None
        if getattr(cfg.news, "source", "synthetic") == "live":
            import os

            from ..live.social_news import LiveNewsHeadlines
            src = LiveNewsHeadlines(
                cryptopanic_key=os.environ.get("CRYPTOPANIC_API_KEY", ""),
                lunarcrush_key=os.environ.get("LUNARCRUSH_API_KEY", ""))
        return NewsAnalysisService(bus, cfg, source=src)
    if name == "order_book":
        from .order_book import OrderBookAnalysisService
        return OrderBookAnalysisService(bus, exchange(), cfg)
    if name == "pattern_recognition":
        from .pattern_recognition import PatternRecognitionService
        return PatternRecognitionService(bus, cfg, device=device)
    if name == "strategy_selection":
        from .strategy_selection import StrategySelectionService
        return StrategySelectionService(bus, cfg)
    if name == "model_registry":
        from .registry import ModelRegistryService
        return ModelRegistryService(bus, cfg)
    if name == "ai_explainability":
        from .registry import AIExplainabilityService
        return AIExplainabilityService(bus, cfg)
    if name == "feature_importance":
        from .registry import FeatureImportanceAnalyzer
        return FeatureImportanceAnalyzer(bus, cfg)
    if name == "grid_trading":
        from .grid_dca import GridTradingStrategy
        return GridTradingStrategy(bus, exchange(), symbols[0], cfg)
    if name == "dca":
        from .grid_dca import DCAStrategy
        return DCAStrategy(bus, exchange(), symbols[0], cfg)
    if name == "arbitrage":
        from .arbitrage import ArbitrageDetectionService
        quote = cfg.trading.quote_asset
        pairs = [(s[: -len(quote)], quote) for s in symbols]
        return ArbitrageDetectionService(bus, exchange(), cfg, pairs=pairs)
    raise SystemExit(f"unknown service {name!r}")


SERVICES = [
    "market_monitor", "ai_analyzer", "portfolio_risk", "monte_carlo",
    "market_regime", "neural_network", "strategy_evolution",
    "trade_executor", "social_monitor", "enhanced_social", "social_risk",
    "news_analysis", "order_book", "pattern_recognition",
    "strategy_selection", "model_registry", "ai_explainability",
    "feature_importance", "grid_trading", "dca", "arbitrage",
]


def parse_args():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--service", required=True, choices=SERVICES)
    ap.add_argument("--bus", default="inprocess",
                    choices=["inprocess", "redis"])
    ap.add_argument("--redis-host", default="redis")
    ap.add_argument("--redis-port", type=int, default=6379)
    ap.add_argument("--minutes", type=float, default=0.0,
                    help="0 = run forever")
    ap.add_argument("--symbols", default="BTCUSDC,ETHUSDC,SOLUSDC")
    ap.add_argument("--candles", type=int, default=20_000)
    ap.add_argument("--speed", type=float, default=1.0)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--device", default="cpu")
    ap.add_argument("--health-port", type=int, default=0,
                    help="serve a TCP health endpoint on this port "
                         "(reference: per-service health_check_server, "
                         "ports 8001-8016, probed with `nc -z` in "
                         "docker-compose healthchecks); 0 = off")
    return ap.parse_args()


async def serve_health(svc, port: int):
    """One-line JSON health over raw TCP — `nc -z` (connect probe) and
    `curl`-style reads both work (reference market_monitor_service.py:
    635-656 health_check_server)."""
    import json as _json

    async def handle(reader, writer):
        h = svc.health()
        writer.write((_json.dumps(h) + "\n").encode())
        try:
            await writer.drain()
        finally:
            writer.close()

    server = await asyncio.start_server(handle, "0.0.0.0", port)
    return server


async def amain():
    args = parse_args()
    cfg = AppConfig.load()
    cfg.trading.symbols = args.symbols.split(",")
    cfg.seed = args.seed
    set_config(cfg)
    if args.bus == "redis":
        bus = make_bus("redis", host=args.redis_host, port=args.redis_port)
    else:
        bus = make_bus()
    svc = build_service(args.service, bus, cfg, args)
    await svc.start()
    health_server = None
    if args.health_port:
        health_server = await serve_health(svc, args.health_port)
    try:
        if args.minutes > 0:
            await asyncio.sleep(args.minutes * 60)
        else:
            while True:
                await asyncio.sleep(3600)
    finally:
        if health_server is not None:
            health_server.close()
        await svc.stop()
        await bus.close()


def main():
    asyncio.run(amain())


if __name__ == "__main__":
    main()
