"""Order-book analytics (reference parity:
services/utils/order_book_analyzer.py:9-824 +
services/order_book_analysis_service.py:15-423).

Depth analytics: liquidity imbalance / bid-ask pressure (:373-447),
price impact of simulated trade sizes (:127-244), support/resistance from
depth clusters via KMeans (:245-372), microstructure (Gini concentration
:528, spoofing/iceberg heuristics :473-606), signal generation
(:667-824). Works on any ExchangeInterface.get_order_book() snapshot."""

from __future__ import annotations

import time

import numpy as np

from ..bus.schema import Keys
from .base import Service

IMPACT_SIZES = [10_000, 50_000, 100_000, 500_000, 1_000_000]   # (:127)


class OrderBookAnalyzer:
    def __init__(self, impact_sizes=None):
        self.impact_sizes = list(impact_sizes or IMPACT_SIZES)

    def analyze(self, book: dict) -> dict:
        bids = np.asarray(book.get("bids", []), dtype=np.float64)
        asks = np.asarray(book.get("asks", []), dtype=np.float64)
        if bids.size == 0 or asks.size == 0:
            return {"ok": False}
        mid = (bids[0, 0] + asks[0, 0]) / 2.0
        out = {"ok": True, "mid": mid,
               "spread_bps": (asks[0, 0] - bids[0, 0]) / mid * 1e4}
        out.update(self.liquidity(bids, asks))
        out["price_impact"] = self.price_impact(bids, asks, mid)
        out["levels"] = self.support_resistance(bids, asks)
        out["microstructure"] = self.microstructure(bids, asks)
        out["signal"] = self.signal(out)
        return out

    def liquidity(self, bids, asks) -> dict:
        """Imbalance + pressure (:373-447)."""
        bv = float((bids[:, 0] * bids[:, 1]).sum())
        av = float((asks[:, 0] * asks[:, 1]).sum())
        tot = bv + av or 1.0
        return {
            "bid_value": bv, "ask_value": av,
            "imbalance": (bv - av) / tot,
            "bid_ask_ratio": bv / max(av, 1e-9),
            "pressure": "buy" if bv > 1.2 * av else
                        ("sell" if av > 1.2 * bv else "balanced"),
        }

    def price_impact(self, bids, asks, mid) -> dict:
        """Walk the book for simulated sizes (:127-244)."""
        out = {}
        for size in self.impact_sizes:
            out[str(size)] = {
                "buy_impact_bps": self._walk(asks, size, mid, +1),
                "sell_impact_bps": self._walk(bids, size, mid, -1),
            }
        return out

    @staticmethod
    def _walk(levels, usd, mid, side) -> float:
        remaining = usd
        cost = 0.0
        qty = 0.0
        for price, q in levels:
            lv = price * q
            take = min(remaining, lv)
            cost += take
            qty += take / price
            remaining -= take
            if remaining <= 0:
                break
        if qty == 0:
            return float("inf")
        vwap = cost / qty
        return (vwap - mid) / mid * 1e4 * side

    def support_resistance(self, bids, asks, k: int = 3) -> dict:
        """Depth clusters via KMeans (:245-372)."""
        from sklearn.cluster import KMeans

        def clusters(levels):
            if len(levels) < k:
                return [float(p) for p in levels[:, 0]]
            X = levels[:, :1]
            w = levels[:, 1]
            km = KMeans(n_clusters=k, n_init=4, random_state=0).fit(
                X, sample_weight=w)
            return sorted(float(c) for c in km.cluster_centers_[:, 0])

        return {"support": clusters(bids), "resistance": clusters(asks)}

    def microstructure(self, bids, asks) -> dict:
        """Gini concentration + spoofing/iceberg heuristics (:473-606)."""
        def gini(x):
            x = np.sort(x)
            n = len(x)
            if n == 0 or x.sum() == 0:
                return 0.0
            cum = np.cumsum(x)
            return float((n + 1 - 2 * (cum / cum[-1]).sum()) / n)

        all_q = np.concatenate([bids[:, 1], asks[:, 1]])
        big = all_q.mean() + 3 * all_q.std()
        spoof_bid = bool((bids[5:, 1] > big).any()) if len(bids) > 5 else False
        spoof_ask = bool((asks[5:, 1] > big).any()) if len(asks) > 5 else False
        top_q = np.concatenate([bids[:3, 1], asks[:3, 1]])
        iceberg = bool(top_q.std() < 0.05 * max(top_q.mean(), 1e-9))
        return {
            "gini": gini(all_q),
            "possible_spoofing": spoof_bid or spoof_ask,
            "possible_iceberg": iceberg,
        }

    def signal(self, a: dict) -> dict:
        """(:667-824) combined book signal."""
        imb = a.get("imbalance", 0.0)
        micro = a.get("microstructure", {})
        direction = "bullish" if imb > 0.15 else (
            "bearish" if imb < -0.15 else "neutral")
        strength = min(abs(imb) * 3, 1.0)
        if micro.get("possible_spoofing"):
            strength *= 0.5
        return {"direction": direction, "strength": strength}


class OrderBookAnalysisService(Service):
    name = "order_book_analysis"

    def __init__(self, bus, exchange, config=None,
                 interval_s: float | None = None):
        super().__init__(bus, config)
        self.exchange = exchange
        ob = self.config.order_book
        self.analyzer = OrderBookAnalyzer(
            impact_sizes=ob.impact_trade_sizes)
        self.interval_s = (interval_s if interval_s is not None
                           else ob.interval_s)
        self.analyzed = 0

    async def run(self):
        while self.running:
            for sym in self.config.trading.symbols:
                try:
                    book = self.exchange.get_order_book(sym)
                    res = self.analyzer.analyze(book)
                except Exception as e:
                    self.log.warning("book %s failed: %r", sym, e)
                    continue
                if res.get("ok"):
                    slim = {k: v for k, v in res.items()
                            if k not in ("levels",)}
                    await self.bus.set(Keys.order_book(sym), slim)
                    await self.bus.set(Keys.order_book_agg(sym), {
                        "imbalance": res["imbalance"],
                        "signal": res["signal"], "at": time.time(),
                    })
                    self.analyzed += 1
            await self.sleep(self.interval_s)
