#!/usr/bin/env python3
"""Standalone trader (reference parity: auto_trader.py:33-774 — the legacy
all-in-one scanner -> analyst -> executor loop without the service bus).

MarketMonitor thread-equivalent: a replay/live feed pushing opportunities
into a queue (:45-92); TradeExecutor loop: scan -> analyze -> size ->
order with market buy + stop-loss + take-profit (:345-431, :569-641);
AutoTrader facade with config load and start/stop (:664-757).

  python auto_trader.py --candles 20000 --minutes 1
"""

from __future__ import annotations

import argparse
import queue
import threading
import time

import numpy as np

from ai_crypto_trader_amd.analysis import (
    CryptoScanner, PositionSizer, TechnicalAnalyzer, TradingSignalVotes,
)
from ai_crypto_trader_amd.config import AppConfig
from ai_crypto_trader_amd.data.synthetic import candles_chl_v, generate_ohlcv
from ai_crypto_trader_amd.services.analyzer import LocalAnalyst
from ai_crypto_trader_amd.utils.exchange import ExchangeFactory


class MarketMonitor(threading.Thread):
    """Feeds per-symbol candle history and flags opportunities (:33-92)."""

    def __init__(self, market: dict[str, np.ndarray], out_q: queue.Queue,
                 min_move_pct: float = 0.2):
        super().__init__(daemon=True)
        self.market = market
        self.out_q = out_q
        self.min_move = min_move_pct
        self.pos = 200
        self.running = True
        self.history: dict[str, np.ndarray] = {}

    def run(self):
        T = min(len(c) for c in self.market.values())
        while self.running and self.pos < T:
            for sym, candles in self.market.items():
                self.history[sym] = candles[: self.pos]
                move = (candles[self.pos - 1, 0] /
                        candles[self.pos - 2, 0] - 1) * 100
                if abs(move) >= self.min_move:
                    try:
                        self.out_q.put_nowait((sym, self.pos))
                    except queue.Full:
                        pass
            self.pos += 1
        self.running = False


class TradeExecutor(threading.Thread):
    """Opportunity queue consumer: votes + analyst gate + sized orders
    (:125-641)."""

    def __init__(self, monitor: MarketMonitor, exchange, config: AppConfig):
        super().__init__(daemon=True)
        self.monitor = monitor
        self.exchange = exchange
        self.config = config
        self.analyst = LocalAnalyst()
        self.sizer = PositionSizer()
        self.positions: dict[str, dict] = {}
        self.trades = 0
        self.running = True

    def should_execute_trade(self, analysis: dict, signal: dict) -> bool:
        """(:485) dual gate: analyst BUY + votes agreement."""
        return (analysis["decision"] == "BUY"
                and analysis["confidence"]
                >= self.config.trading.min_confidence * 0.7
                and signal["decision"] == "BUY")

    def execute_trade(self, sym: str, price: float, analysis: dict,
                      ta: TechnicalAnalyzer):
        """(:345-431): market buy + STOP_LOSS_LIMIT + LIMIT TP."""
        bal = self.exchange.get_balances().get(
            self.config.trading.quote_asset, 0.0)
        size = self.sizer.calculate_position_size(bal, ta.volatility())
        qty = size["position_usd"] / price
        if qty <= 0:
            return
        o = self.exchange.create_order(sym, "BUY", "MARKET", qty)
        if o.status != "FILLED":
            return
        stop = price * (1 - size["stop_loss_pct"])
        tp = price * (1 + size["take_profit_pct"])
        self.exchange.create_order(sym, "SELL", "STOP_LOSS_LIMIT",
                                   qty * 0.999, price=stop * 0.99,
                                   stop_price=stop)
        self.exchange.create_order(sym, "SELL", "LIMIT", 0.0, price=tp)
        self.positions[sym] = {"qty": qty, "entry": price, "stop": stop,
                               "tp": tp}
        self.trades += 1

    def run(self):
        q = queue.Queue(maxsize=1000)
        self.monitor.out_q = q
        while self.running and self.monitor.running:
            try:
                sym, pos = q.get(timeout=0.5)
            except queue.Empty:
                continue
            if sym in self.positions:
                continue
            hist = self.monitor.history.get(sym)
            if hist is None or len(hist) < 100:
                continue
            ta = TechnicalAnalyzer(hist[-512:])
            sigv = TradingSignalVotes(ta).signal()
            price = float(hist[-1, 0])
            bb = ta.bollinger()
            update = {
                "symbol": sym, "current_price": price,
                "avg_volume": float(hist[-20:, 3].mean()),
                "rsi": ta.rsi(), "stoch_k": ta.stochastic(),
                "williams_r": ta.williams_r(),
                "macd": ta.macd()[0], "bb_position": bb["position"],
                "trend": ta.trend(),
                "trend_strength": 50.0,
                "price_change_1m":
                    (price / float(hist[-2, 0]) - 1) * 100,
                "price_change_5m":
                    (price / float(hist[-6, 0]) - 1) * 100,
                "price_change_15m":
                    (price / float(hist[-16, 0]) - 1) * 100,
            }
            analysis = self.analyst.analyze(update)
            if self.should_execute_trade(analysis, sigv):
                self.exchange.set_price(sym, price)
                self.execute_trade(sym, price, analysis, ta)


class AutoTrader:
    """Facade (:664-757)."""

    def __init__(self, config: AppConfig | None = None,
                 market: dict[str, np.ndarray] | None = None):
        self.config = config or AppConfig()
        if market is None:
            syms = self.config.trading.symbols
            data = candles_chl_v(
                generate_ohlcv(20_000, len(syms), seed=self.config.seed))
            market = {s: data[i] for i, s in enumerate(syms)}
        self.exchange = ExchangeFactory.create_exchange(
            "fake", fee_rate=self.config.trading.fee_rate,
            quote=self.config.trading.quote_asset)
        for s, c in market.items():
            self.exchange.set_price(s, float(c[199, 0]))
        self.monitor = MarketMonitor(market, queue.Queue(maxsize=1000))
        self.executor = TradeExecutor(self.monitor, self.exchange,
                                      self.config)
        self.scanner = CryptoScanner(self.config.trading.quote_asset)
        self.market = market

    def scan(self, top_k: int = 5):
        return self.scanner.scan_market(
            {s: c[:5000] for s, c in self.market.items()}, top_k=top_k)

    def start(self):
        self.monitor.start()
        self.executor.start()

    def stop(self):
        self.monitor.running = False
        self.executor.running = False
        self.monitor.join(timeout=5)
        self.executor.join(timeout=5)

    def status(self) -> dict:
        return {
            "positions": dict(self.executor.positions),
            "trades": self.executor.trades,
            "balances": self.exchange.get_balances(),
            "feed_pos": self.monitor.pos,
        }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--minutes", type=float, default=1.0)
    ap.add_argument("--candles", type=int, default=20_000)
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()
    cfg = AppConfig.load()
    cfg.seed = args.seed
    syms = cfg.trading.symbols
    data = candles_chl_v(generate_ohlcv(args.candles, len(syms),
                                        seed=args.seed))
    at = AutoTrader(cfg, {s: data[i] for i, s in enumerate(syms)})
    print("scan:", [(r["symbol"], round(r["score"], 1))
                    for r in at.scan()])
    at.start()
    deadline = time.time() + args.minutes * 60
    while time.time() < deadline and at.monitor.running:
        time.sleep(2)
        st = at.status()
        print(f"t={st['feed_pos']} trades={st['trades']} "
              f"positions={list(st['positions'])}")
    at.stop()
    print("final:", at.status()["balances"])


if __name__ == "__main__":
    main()
