"""Binance-schema fixture server: a stateful transport that speaks the
venue's REST wire format.

Where ReplayTransport serves a static tape, this simulates the venue at
the HTTP layer — balances move, MARKET orders fill at the posted price,
STOP_LOSS_LIMIT orders rest and trigger on set_price() — so integration
tests drive the FULL executor order lifecycle through the real
BinanceExchange adapter (signing, params, JSON parsing) with
schema-faithful responses. Field shapes mirror what the reference's
live code consumes (services/utils/exchange_interface.py:67-207).
"""

from __future__ import annotations

import itertools
import json
import urllib.parse


class BinanceFixtureServer:
    def __init__(self, prices: dict[str, float] | None = None,
                 balances: dict[str, float] | None = None,
                 quote: str = "USDC", fee: float = 0.001,
                 tick: float = 0.01, step: float = 1e-5,
                 min_notional: float = 5.0):
        self.prices = dict(prices or {})
        self.balances = dict(balances or {quote: 10_000.0})
        self.quote = quote
        self.fee = fee
        self.tick = tick
        self.step = step
        self.min_notional = min_notional
        self.orders: dict[str, dict] = {}
        self._ids = itertools.count(1001)
        self.requests: list[tuple] = []      # audit for assertions

    # --- market control ---------------------------------------------------
    def set_price(self, symbol: str, price: float):
        self.prices[symbol] = price
        for o in self.orders.values():
            if o["symbol"] != symbol or o["status"] != "NEW":
                continue
            if o["type"] == "STOP_LOSS_LIMIT" and o["side"] == "SELL" \
                    and price <= o["stopPrice"]:
                self._fill(o, o["price"] or price)
            elif o["type"] == "LIMIT":
                if (o["side"] == "SELL" and price >= o["price"]) or \
                        (o["side"] == "BUY" and price <= o["price"]):
                    self._fill(o, o["price"])

    def _base(self, symbol):
        return symbol[:-len(self.quote)] \
            if symbol.endswith(self.quote) else symbol

    def _fill(self, o: dict, price: float):
        base = self._base(o["symbol"])
        qty = o["origQty"]
        if o["side"] == "BUY":
            cost = qty * price
            if self.balances.get(self.quote, 0.0) < cost:
                o["status"] = "REJECTED"
                return
            self.balances[self.quote] -= cost
            self.balances[base] = self.balances.get(base, 0.0) \
                + qty * (1 - self.fee)
        else:
            if self.balances.get(base, 0.0) < qty:
                o["status"] = "REJECTED"
                return
            self.balances[base] -= qty
            self.balances[self.quote] = \
                self.balances.get(self.quote, 0.0) \
                + qty * price * (1 - self.fee)
        o["status"] = "FILLED"
        o["executedQty"] = qty
        o["cummulativeQuoteQty"] = qty * price

    # --- transport interface ---------------------------------------------
    def request(self, method, url, params=None, headers=None, data=None):
        path = urllib.parse.urlparse(url).path
        p = {k: v for k, v in (params or {}).items()}
        self.requests.append((method, path, dict(p)))
        fn = {
            ("GET", "/api/v3/ticker/bookTicker"): self._ticker,
            ("GET", "/api/v3/depth"): self._depth,
            ("POST", "/api/v3/order"): self._new_order,
            ("DELETE", "/api/v3/order"): self._cancel,
            ("GET", "/api/v3/order"): self._query_order,
            ("GET", "/api/v3/account"): self._account,
            ("GET", "/sapi/v1/asset/tradeFee"): self._trade_fee,
            ("GET", "/api/v3/exchangeInfo"): self._exchange_info,
            ("GET", "/api/v3/klines"): self._klines,
        }.get((method.upper(), path))
        if fn is None:
            return 404, json.dumps({"code": -1000,
                                    "msg": f"unknown {path}"})
        return fn(p)

    def _ticker(self, p):
        px = self.prices.get(p.get("symbol", ""), 0.0)
        return 200, json.dumps({
            "symbol": p.get("symbol"),
            "bidPrice": f"{px * 0.9999:.8f}",
            "bidQty": "1.0",
            "askPrice": f"{px * 1.0001:.8f}",
            "askQty": "1.0",
        })

    def _depth(self, p):
        px = self.prices.get(p.get("symbol", ""), 1.0)
        n = int(p.get("limit", 100))
        return 200, json.dumps({
            "lastUpdateId": 1,
            "bids": [[f"{px * (1 - 1e-4 * (i + 1)):.8f}",
                      f"{1.0 + 0.1 * i:.4f}"] for i in range(n)],
            "asks": [[f"{px * (1 + 1e-4 * (i + 1)):.8f}",
                      f"{1.0 + 0.1 * i:.4f}"] for i in range(n)],
        })

    def _new_order(self, p):
        sym = p["symbol"]
        o = {
            "symbol": sym, "orderId": next(self._ids),
            "side": p["side"], "type": p["type"],
            "origQty": float(p["quantity"]),
            "price": float(p.get("price", 0) or 0),
            "stopPrice": float(p.get("stopPrice", 0) or 0),
            "status": "NEW", "executedQty": 0.0,
            "cummulativeQuoteQty": 0.0,
        }
        self.orders[str(o["orderId"])] = o
        if o["type"] == "MARKET":
            self._fill(o, self.prices.get(sym, 0.0))
        return 200, json.dumps(self._order_json(o))

    def _order_json(self, o):
        return {
            "symbol": o["symbol"], "orderId": o["orderId"],
            "status": o["status"], "side": o["side"], "type": o["type"],
            "origQty": f"{o['origQty']:.8f}",
            "executedQty": f"{o['executedQty']:.8f}",
            "cummulativeQuoteQty": f"{o['cummulativeQuoteQty']:.8f}",
            "price": f"{o['price']:.8f}",
            "stopPrice": f"{o['stopPrice']:.8f}",
        }

    def _cancel(self, p):
        o = self.orders.get(str(p.get("orderId", "")))
        if o is None or o["status"] != "NEW":
            return 400, json.dumps({"code": -2011,
                                    "msg": "Unknown order sent."})
        o["status"] = "CANCELED"
        return 200, json.dumps(self._order_json(o))

    def _query_order(self, p):
        o = self.orders.get(str(p.get("orderId", "")))
        if o is None:
            return 400, json.dumps({"code": -2013,
                                    "msg": "Order does not exist."})
        return 200, json.dumps(self._order_json(o))

    def _account(self, p):
        return 200, json.dumps({
            "balances": [
                {"asset": a, "free": f"{v:.8f}", "locked": "0.00000000"}
                for a, v in self.balances.items()
            ],
        })

    def _trade_fee(self, p):
        return 200, json.dumps([{
            "symbol": p.get("symbol", ""),
            "makerCommission": f"{self.fee}",
            "takerCommission": f"{self.fee}",
        }])

    def _exchange_info(self, p):
        sym = p.get("symbol", "")
        return 200, json.dumps({
            "symbols": [{
                "symbol": sym,
                "filters": [
                    {"filterType": "PRICE_FILTER",
                     "tickSize": f"{self.tick}"},
                    {"filterType": "LOT_SIZE",
                     "stepSize": f"{self.step}"},
                    {"filterType": "NOTIONAL",
                     "minNotional": f"{self.min_notional}"},
                ],
            }],
        })

    def _klines(self, p):
        # deterministic synthetic kline pages keyed on startTime: tests
        # exercise the PAGINATION mechanics; shapes match the venue
        sym = p.get("symbol", "")
        start = int(p.get("startTime", 0))
        end = int(p.get("endTime", 1 << 62))
        limit = int(p.get("limit", 1000))
        step = 60_000
        base_px = self.prices.get(sym, 100.0)
        rows = []
        t = start
        while len(rows) < limit and t <= end and t < self.kline_end_ms:
            px = base_px * (1 + 1e-6 * ((t // step) % 997))
            rows.append([
                t, f"{px:.8f}", f"{px * 1.001:.8f}",
                f"{px * 0.999:.8f}", f"{px * 1.0005:.8f}", "10.0",
                t + step - 1, "1000.0", 10, "5.0", "500.0", "0",
            ])
            t += step
        return 200, json.dumps(rows)

    kline_end_ms = 90_000_000     # default ~1500 minutes of history
