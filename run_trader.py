#!/usr/bin/env python3
"""All-in-one trader launcher (reference parity: run_trader.py — which
ships with a SyntaxError at :865/:989 and cannot run; this one works).

Starts the full service topology in one asyncio process over the
in-process bus: market monitor (synthetic replay or live seam), analyzer,
portfolio risk, Monte-Carlo, regime detection, neural-network prediction,
strategy evolution, trade executor — and prints a periodic status report
(reference print_status :39-1247 reads ~25 Redis keys; this reads the
same keys off the bus).

Usage:
  python run_trader.py [--minutes N] [--symbols BTCUSDC,ETHUSDC]
                       [--candles 20000] [--speed 0] [--gpu]
"""

from __future__ import annotations

import argparse
import asyncio
import time


from ai_crypto_trader_amd.bus.message_bus import InProcessBus
from ai_crypto_trader_amd.bus.schema import Keys
from ai_crypto_trader_amd.config import AppConfig, set_config
from ai_crypto_trader_amd.data.feed import SyntheticFeed
from ai_crypto_trader_amd.data.synthetic import (
    candles_chl_v, generate_ohlcv,
)
from ai_crypto_trader_amd.services.analyzer import AnalyzerService
from ai_crypto_trader_amd.services.market_monitor import MarketMonitorService
from ai_crypto_trader_amd.services.market_regime import MarketRegimeService
from ai_crypto_trader_amd.services.monte_carlo import MonteCarloService
from ai_crypto_trader_amd.services.neural_network import NeuralNetworkService
from ai_crypto_trader_amd.services.portfolio_risk import PortfolioRiskService
from ai_crypto_trader_amd.services.strategy_evolution import (
    StrategyEvolutionService,
)
from ai_crypto_trader_amd.services.trade_executor import TradeExecutorService
from ai_crypto_trader_amd.utils.exchange import ExchangeFactory


def parse_args():
    ap = argparse.ArgumentParser()
    ap.add_argument("--minutes", type=float, default=2.0,
                    help="wall-clock run time")
    ap.add_argument("--symbols", type=str, default="BTCUSDC,ETHUSDC,SOLUSDC,XRPUSDC")
    ap.add_argument("--candles", type=int, default=20_000)
    ap.add_argument("--speed", type=float, default=0.0,
                    help="0 = replay as fast as possible")
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--market", choices=["calm", "regime"], default="calm",
                    help="synthetic market: calm GBM or regime-switching "
                         "episodes (bull/calm/bear/volatile)")
    ap.add_argument("--gpu", action="store_true",
                    help="use cuda:0 for the numeric services")
    ap.add_argument("--status-interval", type=float, default=5.0)
    ap.add_argument("--core-only", action="store_true",
                    help="run only the 8 core services (default: the full "
                         "17+-service topology)")
    ap.add_argument("--dashboard", action="store_true",
                    help="serve the dashboard API on :8050")
    return ap.parse_args()


async def print_status(bus: InProcessBus, services: list, interval: float):
    """Status report (reference print_status run_trader.py:39-1247 reads
    ~25 Redis keys; this reads the same keys off the bus)."""
    while True:
        await asyncio.sleep(interval)
        holdings = await bus.get_json(Keys.HOLDINGS) or {}
        risk = await bus.get_json(Keys.PORTFOLIO_RISK) or {}
        regime = await bus.get_json(Keys.CURRENT_MARKET_REGIME) or {}
        prices = await bus.hgetall(Keys.CURRENT_PRICES)
        active = await bus.get_json(Keys.ACTIVE_TRADES) or {}
        mc = await bus.get_json(Keys.MONTE_CARLO_LATEST_REPORT) or {}
        patt = await bus.get_json(Keys.PATTERN_ANALYSIS_REPORT) or {}
        news = await bus.get_json(Keys.NEWS_SUMMARY_REPORT) or {}
        grid = await bus.get_json(Keys.GRID_PERFORMANCE) or {}
        dca = await bus.get_json(Keys.DCA_PERFORMANCE) or {}
        params = await bus.get_json(Keys.STRATEGY_PARAMS) or {}
        social = await bus.hgetall(Keys.SOCIAL_METRICS)
        diver = await bus.get_json(Keys.PORTFOLIO_DIVERSIFICATION) or {}
        print("=" * 72)
        print(f"status @ {time.strftime('%H:%M:%S')}  | "
              f"portfolio ${holdings.get('total_value', 0):,.2f}  | "
              f"open trades {len(active)}  | "
              f"regime {regime.get('regime', '?')}"
              f" ({regime.get('confidence', 0):.2f})")
        print(f"portfolio VaR: ${risk.get('portfolio_var', 0):,.2f}"
              f" ({risk.get('portfolio_var_pct', 0):.2f}%)  | "
              f"avg corr {risk.get('avg_correlation', 0):.2f}")
        if prices:
            ps = "  ".join(f"{s}={float(p):.4f}"
                           for s, p in sorted(prices.items())[:6])
            print(f"prices: {ps}")
        extras = []
        if mc.get("base_var_95") is not None:
            extras.append(f"MC var95={mc['base_var_95']:.4f}")
        if news.get("market_sentiment") is not None:
            extras.append(f"news={news['market_sentiment']:.2f}")
        if social:
            import json as _j
            vals = [_j.loads(v).get("sentiment", 0.5)
                    for v in social.values()]
            extras.append(f"social={sum(vals) / len(vals):.2f}")
        if patt.get("patterns"):
            found = [f"{s}:{d['pattern']}"
                     for s, d in patt["patterns"].items()
                     if d["pattern"] != "none"][:3]
            if found:
                extras.append("patterns=" + ",".join(found))
        if grid.get("fills"):
            extras.append(f"grid fills={grid['fills']}")
        if dca.get("n_purchases"):
            extras.append(f"dca buys={dca['n_purchases']}")
        if params:
            extras.append("params=evolved")
        if diver.get("avg_correlation") is not None:
            extras.append(f"divers={1 - diver['avg_correlation']:.2f}")
        if extras:
            print(" | ".join(extras))
        for s in services:
            h = s.health()
            extra = ""
            for attr in ("updates_published", "signals_published",
                         "trades_done", "evolutions", "runs", "trained",
                         "predictions"):
                if hasattr(s, attr):
                    extra += f" {attr}={getattr(s, attr)}"
            print(f"  [{'OK' if h['healthy'] else 'DOWN'}] "
                  f"{h['service']:<20}{extra}")


async def main():
    args = parse_args()
    symbols = args.symbols.split(",")
    cfg = AppConfig.load()
    cfg.trading.symbols = symbols
    cfg.trading.ai_analysis_interval = 0.5     # replay-speed analysis
    cfg.monte_carlo.interval_s = 20.0
    cfg.monte_carlo.num_simulations = 20_000
    cfg.evolution.interval_s = 45.0
    cfg.evolution.population_size = 256
    cfg.evolution.generations = 3
    cfg.seed = args.seed
    set_config(cfg)

    device = "cuda:0" if args.gpu else "cpu"
    bus = InProcessBus()
    if args.market == "regime":
        from ai_crypto_trader_amd.data.synthetic import (
            generate_regime_ohlcv,
        )
        market = candles_chl_v(generate_regime_ohlcv(
            args.candles, len(symbols), seed=args.seed))
    else:
        market = candles_chl_v(
            generate_ohlcv(args.candles, len(symbols), seed=args.seed))
    feed = SyntheticFeed(market, symbols, start=0, speed=args.speed)
    exchange = ExchangeFactory.create_exchange(
        "fake", fee_rate=cfg.trading.fee_rate,
        quote=cfg.trading.quote_asset)

    # keep the fake exchange's tickers in sync with the feed
    monitor = MarketMonitorService(bus, feed, cfg)
    orig_push = monitor._push

    def push_and_tick(c):
        exchange.set_price(c.symbol, c.close)
        return orig_push(c)

    monitor._push = push_and_tick

    services = [
        monitor,
        AnalyzerService(bus, cfg),
        PortfolioRiskService(bus, cfg),
        MonteCarloService(bus, cfg),
        MarketRegimeService(bus, cfg, device=device),
        NeuralNetworkService(bus, cfg, device=device),
        StrategyEvolutionService(bus, cfg, candles=market, device=device),
        TradeExecutorService(bus, exchange, cfg),
    ]
    if not args.core_only:
        # full reference topology
        from ai_crypto_trader_amd.services.arbitrage import (
            ArbitrageDetectionService,
        )
        from ai_crypto_trader_amd.services.grid_dca import (
            DCAStrategy, GridTradingStrategy,
        )
        from ai_crypto_trader_amd.services.news import NewsAnalysisService
        from ai_crypto_trader_amd.services.order_book import (
            OrderBookAnalysisService,
        )
        from ai_crypto_trader_amd.services.pattern_recognition import (
            PatternRecognitionService,
        )
        from ai_crypto_trader_amd.services.registry import (
            AIExplainabilityService, FeatureImportanceAnalyzer,
            ModelRegistryService,
        )
        from ai_crypto_trader_amd.services.social import (
            EnhancedSocialMonitorService, SocialMonitorService,
            SocialRiskAdjuster,
        )
        from ai_crypto_trader_amd.services.strategy_selection import (
            StrategySelectionService,
        )
        quote = cfg.trading.quote_asset
        pairs = [(s[: -len(quote)], quote) for s in symbols]
        services += [
            SocialMonitorService(bus, cfg),
            EnhancedSocialMonitorService(bus, cfg),
            SocialRiskAdjuster(bus, cfg),
            NewsAnalysisService(bus, cfg),
            OrderBookAnalysisService(bus, exchange, cfg),
            PatternRecognitionService(bus, cfg, device=device),
            StrategySelectionService(bus, cfg),
            ModelRegistryService(bus, cfg),
            AIExplainabilityService(bus, cfg),
            FeatureImportanceAnalyzer(bus, cfg),
            GridTradingStrategy(bus, exchange, symbols[0], cfg),
            DCAStrategy(bus, exchange, symbols[0], cfg),
            ArbitrageDetectionService(bus, exchange, cfg, pairs=pairs),
        ]
    for s in services:
        await s.start()
    from ai_crypto_trader_amd.services.supervisor import ServiceSupervisor
    supervisor = ServiceSupervisor(bus, services, cfg)
    await supervisor.start()
    services = services + [supervisor]

    dash_server = None
    if args.dashboard:
        import uvicorn

        from dashboard import DataStore, build_app
        store = DataStore(bus)
        app = build_app(bus, store)
        dash_server = uvicorn.Server(uvicorn.Config(
            app, host="127.0.0.1", port=8050, log_level="warning"))
        asyncio.create_task(dash_server.serve())
    status = asyncio.create_task(
        print_status(bus, services, args.status_interval))

    deadline = time.monotonic() + args.minutes * 60.0
    try:
        while time.monotonic() < deadline and monitor.running:
            await asyncio.sleep(0.5)
    finally:
        status.cancel()
        for s in services:
            await s.stop()
        holdings = await bus.get_json(Keys.HOLDINGS) or {}
        print(f"\nfinal portfolio value: "
              f"${holdings.get('total_value', 0):,.2f}")


if __name__ == "__main__":
    asyncio.run(main())
