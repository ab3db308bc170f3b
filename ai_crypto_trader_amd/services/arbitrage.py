"""Arbitrage detection (reference parity:
services/arbitrage_detection_service.py:17-783).

Triangle arbitrage over a currency-pair digraph: cycle enumeration
bounded by max steps (:309-340) and cycle profitability = product of
fee-adjusted rates (:342-432); cross-exchange spread scaffold (:434-522);
opportunity history + notifications (:523-675). Cycle search is a
self-contained bounded DFS (the reference used networkx.simple_cycles)."""

from __future__ import annotations

import time
from collections import defaultdict

from ..bus.schema import Channels, Keys
from .base import Service


class PairGraph:
    def __init__(self, fee: float = 0.001):
        self.fee = fee
        # rates[a][b] = how much b per 1 a (fee-free)
        self.rates: dict[str, dict[str, float]] = defaultdict(dict)

    def add_pair(self, base: str, quote: str, price: float):
        """A market base/quote at `price` quote per base."""
        if price <= 0:
            return
        self.rates[base][quote] = price
        self.rates[quote][base] = 1.0 / price

    def cycles(self, start: str, max_steps: int = 4) -> list[list[str]]:
        """Bounded DFS cycle enumeration from `start` (:309-340)."""
        out = []

        def dfs(node, path):
            if len(path) > max_steps:
                return
            for nxt in self.rates.get(node, {}):
                if nxt == start and len(path) >= 3:
                    out.append(path + [start])
                elif nxt not in path:
                    dfs(nxt, path + [nxt])

        dfs(start, [start])
        return out

    def cycle_profit(self, cycle: list[str]) -> float:
        """Fee-adjusted product of rates around the cycle (:342-432).
        Returns net multiplier (1.002 = +0.2%)."""
        x = 1.0
        for a, b in zip(cycle[:-1], cycle[1:]):
            r = self.rates.get(a, {}).get(b)
            if r is None:
                return 0.0
            x *= r * (1.0 - self.fee)
        return x


class ArbitrageDetectionService(Service):
    name = "arbitrage_detection"

    def __init__(self, bus, exchange, config=None,
                 min_profit_pct: float | None = None,
                 base_currency: str = "USDC",
                 pairs: list[tuple[str, str]] | None = None):
        super().__init__(bus, config)
        self.exchange = exchange
        self.min_profit_pct = (min_profit_pct if min_profit_pct is not None
                               else self.config.arbitrage.min_profit_pct)
        self.base = base_currency
        self.pairs = pairs or []
        self.opportunities: list[dict] = []
        self.scans = 0

    def scan_once(self) -> list[dict]:
        g = PairGraph(fee=self.config.trading.fee_rate)
        for base, quote in self.pairs:
            t = self.exchange.get_ticker(base + quote)
            g.add_pair(base, quote, t["price"])
        found = []
        for cyc in g.cycles(self.base, max_steps=4):
            profit = g.cycle_profit(cyc)
            pct = (profit - 1.0) * 100.0
            if pct >= self.min_profit_pct:
                found.append({
                    "cycle": cyc, "profit_pct": pct, "at": time.time(),
                })
        self.scans += 1
        return found

    async def run(self):
        while self.running:
            try:
                found = self.scan_once()
            except Exception as e:
                self.log.warning("scan failed: %r", e)
                found = []
            if found:
                self.opportunities.extend(found)
                del self.opportunities[:-200]
                await self.bus.set(Keys.ARBITRAGE_OPPORTUNITIES,
                                   self.opportunities[-20:])
                for op in found:
                    await self.bus.publish(
                        Channels.ARBITRAGE_NOTIFICATIONS, op)
            await self.sleep(2.0)


class CrossExchangeDetector:
    """Cross-exchange spread detection (reference
    arbitrage_detection_service.py:434-522): compares the same symbol's
    quotes across two ExchangeInterface venues and reports spreads that
    survive both venues' taker fees."""

    def __init__(self, venues: dict, min_profit_pct: float = 0.05):
        self.venues = dict(venues)            # name -> ExchangeInterface
        self.min_profit_pct = min_profit_pct

    def scan(self, symbols: list[str]) -> list[dict]:
        out = []
        names = list(self.venues)
        for sym in symbols:
            quotes = {}
            for name in names:
                try:
                    tk = self.venues[name].get_ticker(sym)
                    fee = self.venues[name].get_trading_fees(sym) \
                        .get("taker", 0.001)
                except Exception:
                    continue
                if tk.get("ask", 0) > 0 and tk.get("bid", 0) > 0:
                    quotes[name] = (tk["bid"], tk["ask"], fee)
            for buy_v in quotes:
                for sell_v in quotes:
                    if buy_v == sell_v:
                        continue
                    bid_s, _, fee_s = quotes[sell_v]
                    _, ask_b, fee_b = quotes[buy_v]
                    # buy at buy_v's ask, sell at sell_v's bid, both fees
                    net = bid_s * (1 - fee_s) / (ask_b * (1 + fee_b)) - 1
                    if net * 100 >= self.min_profit_pct:
                        out.append({
                            "symbol": sym, "buy_on": buy_v,
                            "sell_on": sell_v,
                            "buy_ask": ask_b, "sell_bid": bid_s,
                            "net_profit_pct": round(net * 100, 4),
                            "at": time.time(),
                        })
        return sorted(out, key=lambda o: -o["net_profit_pct"])
