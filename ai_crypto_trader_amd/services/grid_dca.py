"""Grid + DCA strategy bots (reference parity:
services/grid_trading_strategy.py:15-978 + services/dca_strategy.py:15-1121).

GridTradingStrategy: arithmetic/geometric level grids (:347-386), auto
boundaries +-range_pct or regime-adapted counts (:840-906), live order
placement against the exchange seam vs the simulation fill engine
(FakeExchange), rebalance on breakout (:781-839), performance metrics.

DCAStrategy: fixed / regime-based / value-averaging schedules (:347-451),
volatility+sentiment-adjusted sizing (:651-741), dip detection with extra
buys (:817-863), performance tracking."""

from __future__ import annotations

import time

import numpy as np

from ..bus.schema import Channels, Keys
from .base import Service


class GridTradingStrategy(Service):
    name = "grid_trading"

    def __init__(self, bus, exchange, symbol: str, config=None,
                 live: bool = False, order_qty: float = 1.0):
        """live=False: internal simulation fill engine (:679-780).
        live=True: place resting LIMIT orders on the exchange seam and
        track their fills (:418-509)."""
        super().__init__(bus, config)
        self.exchange = exchange
        self.symbol = symbol
        self.live = live
        self.order_qty = order_qty
        self.levels: list[dict] = []
        self.center = 0.0
        self.fills = 0
        self.pnl = 0.0
        self._orders: dict[str, dict] = {}    # order_id -> level

    def build_grid(self, center: float, regime: str = "ranging"):
        """(:347-386, :840-906) level construction."""
        g = self.config.grid
        n = g.levels + (4 if regime == "volatile" else 0)
        rng = g.range_pct * (1.5 if regime == "volatile" else 1.0)
        if g.spacing == "geometric":
            ratios = np.geomspace(1 - rng, 1 + rng, n)
        else:
            ratios = np.linspace(1 - rng, 1 + rng, n)
        self.center = center
        if self.live:
            for oid in list(self._orders):
                self.exchange.cancel_order(self.symbol, oid)
            self._orders.clear()
        self.levels = [
            {"price": float(center * r),
             "side": "BUY" if r < 1.0 else "SELL",
             "filled": False}
            for r in ratios if abs(r - 1.0) > 1e-9
        ]
        if self.live:
            for lv in self.levels:
                o = self.exchange.create_order(
                    self.symbol, lv["side"], "LIMIT", self.order_qty,
                    price=lv["price"])
                if o.status == "NEW":
                    self._orders[o.order_id] = lv
                elif o.status == "FILLED":
                    lv["filled"] = True
        return self.levels

    def poll_live_fills(self) -> list[dict]:
        """Live mode: reconcile resting orders with exchange fills and
        re-arm the grid (:418-509)."""
        fills = []
        for oid, lv in list(self._orders.items()):
            o = self.exchange.get_order(self.symbol, oid)
            if o is None:
                self._orders.pop(oid, None)
                continue
            if o.status == "FILLED":
                lv["filled"] = True
                self._orders.pop(oid)
                fills.append({**lv, "order_id": oid, "at": time.time()})
                if lv["side"] == "SELL":
                    self.pnl += (lv["price"] - self.center) / self.center
            elif o.status == "CANCELED":
                self._orders.pop(oid)
        self.fills += len(fills)
        return fills

    def on_price(self, price: float) -> list[dict]:
        """Per-tick driver: simulation fill engine (:679-780), or live
        order reconciliation when live=True."""
        fills = []
        if not self.levels:
            self.build_grid(price)
            return fills
        if self.live:
            fills = self.poll_live_fills()
            lo = min(lv["price"] for lv in self.levels)
            hi = max(lv["price"] for lv in self.levels)
            if price < lo * 0.995 or price > hi * 1.005:
                self.build_grid(price)
            return fills
        for lv in self.levels:
            if lv["filled"]:
                continue
            if lv["side"] == "BUY" and price <= lv["price"]:
                lv["filled"] = True
                fills.append({**lv, "at": time.time()})
            elif lv["side"] == "SELL" and price >= lv["price"]:
                lv["filled"] = True
                fills.append({**lv, "at": time.time()})
        self.fills += len(fills)
        for f in fills:
            # round-trip profit realized when a sell above center fills
            if f["side"] == "SELL":
                self.pnl += (f["price"] - self.center) / self.center
        # breakout rebalance (:781-839)
        lo = min(lv["price"] for lv in self.levels)
        hi = max(lv["price"] for lv in self.levels)
        if price < lo * 0.995 or price > hi * 1.005:
            self.build_grid(price)
        return fills

    async def run(self):
        sub = self.bus.subscribe(Channels.MARKET_UPDATES)

        async def on_msg(_, m):
            if m.get("symbol") != self.symbol:
                return
            fills = self.on_price(m["current_price"])
            for f in fills:
                await self.bus.publish(Channels.GRID_TRADE_NOTIFICATIONS, {
                    "symbol": self.symbol, **{k: f[k] for k in
                                              ("price", "side")},
                })
            await self.bus.set(Keys.GRID_PERFORMANCE, {
                "symbol": self.symbol, "fills": self.fills,
                "realized_pnl_pct": self.pnl * 100, "center": self.center,
            })
            await self.bus.set(Keys.grid_config(self.symbol), {
                "levels": len(self.levels),
                "low": min((lv["price"] for lv in self.levels),
                           default=0.0),
                "high": max((lv["price"] for lv in self.levels),
                            default=0.0),
            })

        await self.consume(sub, on_msg)


class DCAStrategy(Service):
    name = "dca_strategy"

    def __init__(self, bus, exchange, symbol: str, config=None,
                 candles_per_period: int = 60):
        super().__init__(bus, config)
        self.exchange = exchange
        self.symbol = symbol
        self.period = candles_per_period
        self.counter = 0
        self.purchases: list[dict] = []
        self.invested = 0.0
        self.units = 0.0
        self.banked = 0.0              # rebalance proceeds
        self.recent: list[float] = []

    def order_size(self, price: float, sentiment: float = 0.5,
                   regime: str = "ranging") -> float:
        """(:651-741) volatility+sentiment-adjusted sizing."""
        base = self.config.dca.base_order_usd
        if len(self.recent) > 20:
            vol = float(np.std(np.diff(np.log(self.recent[-60:]))))
            base *= float(np.clip(1.0 / max(vol * 300, 0.5), 0.5, 2.0))
        base *= 0.8 + 0.4 * sentiment
        if regime == "bear":
            base *= 1.2          # accumulate harder in drawdowns
        return base

    def is_dip(self, price: float) -> bool:
        """(:817-863) dip detection vs recent high."""
        if len(self.recent) < 30:
            return False
        hi = max(self.recent[-120:])
        return price < hi * (1 - self.config.dca.dip_threshold_pct)

    def _effective_period(self, regime: str) -> int:
        """regime_based schedule (:347-451): accumulate twice as often in
        bear markets, half as often in bull runs."""
        if self.config.dca.schedule != "regime_based":
            return self.period
        mult = {"bear": 0.5, "volatile": 0.75, "bull": 2.0}.get(regime, 1.0)
        return max(int(self.period * mult), 1)

    async def maybe_buy(self, price: float, sentiment: float,
                        regime: str):
        self.counter += 1
        scheduled = self.counter % self._effective_period(regime) == 0
        dip = self.is_dip(price)
        if not (scheduled or dip):
            return None
        if scheduled and self.config.dca.schedule == "value_averaging":
            # value averaging (:347-451): buy the gap to a linearly
            # growing target value instead of a fixed amount
            periods_done = self.counter // self.period
            target = self.config.dca.base_order_usd * periods_done
            gap = target - self.units * price
            usd = float(np.clip(gap, 0.0,
                                self.config.dca.base_order_usd * 4))
            if usd < 1e-6 and not dip:
                return None
        else:
            usd = self.order_size(price, sentiment, regime)
        if dip:
            usd *= self.config.dca.dip_multiplier
        qty = usd / price
        self.invested += usd
        self.units += qty
        rec = {"symbol": self.symbol, "usd": usd, "qty": qty,
               "price": price, "dip": dip, "at": time.time()}
        self.purchases.append(rec)
        await self.bus.publish(Channels.DCA_PURCHASE_NOTIFICATIONS, rec)
        await self.bus.set(Keys.DCA_PURCHASE_LIST, self.purchases[-50:])
        value = self.units * price
        await self.bus.set(Keys.DCA_PERFORMANCE, {
            "symbol": self.symbol, "invested": self.invested,
            "value": value,
            "pnl_pct": (value / self.invested - 1) * 100
            if self.invested else 0.0,
            "avg_cost": self.invested / self.units if self.units else 0.0,
            "n_purchases": len(self.purchases),
        })
        return rec

    def maybe_rebalance(self, price: float,
                        max_alloc_pct: float = 0.6,
                        period_mult: int = 30) -> dict | None:
        """Periodic ("monthly") rebalancing (reference :864-1022): when
        the position has run far past the target allocation of total
        DCA capital, trim back to target and bank the proceeds. Runs every
        `period_mult` DCA periods."""
        if self.counter == 0 or self.counter % (self.period * period_mult):
            return None
        value = self.units * price
        capital = value + self.banked
        if capital <= 0 or value / capital <= max_alloc_pct:
            return None
        target_value = capital * max_alloc_pct
        sell_units = (value - target_value) / price
        self.units -= sell_units
        proceeds = sell_units * price
        self.banked += proceeds
        rec = {"symbol": self.symbol, "action": "rebalance_sell",
               "qty": sell_units, "price": price, "usd": proceeds,
               "at": time.time()}
        self.purchases.append(rec)
        return rec

    async def run(self):
        sub = self.bus.subscribe(Channels.MARKET_UPDATES)

        async def on_msg(_, m):
            if m.get("symbol") != self.symbol:
                return
            price = m["current_price"]
            self.recent.append(price)
            del self.recent[:-512]
            reb = self.maybe_rebalance(price)
            if reb:
                await self.bus.publish(
                    Channels.DCA_PURCHASE_NOTIFICATIONS, reb)
            sent_raw = await self.bus.hget(Keys.SOCIAL_METRICS, self.symbol)
            sentiment = 0.5
            if sent_raw:
                import json
                sentiment = json.loads(sent_raw).get("sentiment", 0.5)
            regime_d = await self.bus.get_json(Keys.CURRENT_MARKET_REGIME)
            regime = (regime_d or {}).get("regime", "ranging")
            await self.maybe_buy(price, sentiment, regime)

        await self.consume(sub, on_msg)
