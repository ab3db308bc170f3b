"""Binance live adapters behind the framework's seams.

  - BinanceExchange: the concrete venue implementation of
    utils.exchange.ExchangeInterface — signed REST orders, balances,
    filters, fees (reference:
    services/utils/exchange_interface.py:67-207 BinanceExchange;
    services/trade_executor_service.py:909-992 MARKET /
    STOP_LOSS_LIMIT / LIMIT order placement).
  - fetch_klines: paginated /api/v3/klines history fetch, 1000 rows per
    request (reference: backtesting/data_manager.py:47-114).
  - BinanceWSFeed: `!miniTicker@arr` stream -> Candle objects for the
    market monitor (reference: market_monitor_service.py:615-633). The
    websocket is a message-source seam, so tests replay recorded frames
    and production plugs a socket reader in.

All HTTP goes through live.transport (UrllibTransport in production,
ReplayTransport in the offline tests — this container has no egress).
"""

from __future__ import annotations

import hashlib
import hmac
import json
import time
import urllib.parse

import numpy as np
import pandas as pd

from ..data.feed import Candle
from ..utils.exchange import ExchangeInterface, Order
from .transport import UrllibTransport

REST_URL = "https://api.binance.com"

_INTERVAL_MS = {
    "1m": 60_000, "3m": 180_000, "5m": 300_000, "15m": 900_000,
    "30m": 1_800_000, "1h": 3_600_000, "4h": 14_400_000,
    "1d": 86_400_000,
}


class BinanceError(Exception):
    pass


class BinanceExchange(ExchangeInterface):
    """ExchangeInterface over Binance spot REST.

    `exchange: binance` in config is the only switch callers need
    (ExchangeFactory registration in live/__init__); every call site
    already talks to the ABC."""

    def __init__(self, api_key: str = "", api_secret: str = "",
                 transport=None, base_url: str = REST_URL,
                 quote: str = "USDC", time_fn=time.time):
        self.key = api_key
        self.secret = api_secret.encode() if api_secret else b""
        self.http = transport or UrllibTransport()
        self.base = base_url.rstrip("/")
        self.quote = quote
        self.time_fn = time_fn
        self._filters: dict[str, dict] = {}
        self._fees: dict[str, dict] = {}

    # --- plumbing --------------------------------------------------------
    def _headers(self):
        return {"X-MBX-APIKEY": self.key} if self.key else {}

    def _sign(self, params: dict) -> dict:
        params = dict(params)
        params["timestamp"] = int(self.time_fn() * 1000)
        qs = urllib.parse.urlencode(params)
        params["signature"] = hmac.new(
            self.secret, qs.encode(), hashlib.sha256).hexdigest()
        return params

    def _get(self, path: str, params: dict | None = None,
             signed: bool = False):
        p = self._sign(params or {}) if signed else (params or {})
        status, body = self.http.request(
            "GET", self.base + path, p, self._headers())
        return self._parse(status, body, path)

    def _post(self, path: str, params: dict):
        status, body = self.http.request(
            "POST", self.base + path, self._sign(params), self._headers())
        return self._parse(status, body, path)

    def _delete(self, path: str, params: dict):
        status, body = self.http.request(
            "DELETE", self.base + path, self._sign(params),
            self._headers())
        return self._parse(status, body, path)

    @staticmethod
    def _parse(status: int, body: str, path: str):
        try:
            data = json.loads(body) if body else {}
        except json.JSONDecodeError as e:
            raise BinanceError(f"{path}: non-JSON response") from e
        if status >= 400 or (isinstance(data, dict) and "code" in data
                             and data.get("code", 0) < 0):
            raise BinanceError(f"{path}: HTTP {status}: {data}")
        return data

    # --- ExchangeInterface ----------------------------------------------
    def get_ticker(self, symbol: str) -> dict:
        d = self._get("/api/v3/ticker/bookTicker", {"symbol": symbol})
        bid = float(d.get("bidPrice", 0.0))
        ask = float(d.get("askPrice", 0.0))
        mid = (bid + ask) / 2 if bid and ask else bid or ask
        return {"symbol": symbol, "price": mid, "bid": bid, "ask": ask}

    def get_order_book(self, symbol: str, limit: int = 100) -> dict:
        d = self._get("/api/v3/depth", {"symbol": symbol,
                                        "limit": limit})
        return {
            "symbol": symbol,
            "bids": [[float(p), float(q)] for p, q in d.get("bids", [])],
            "asks": [[float(p), float(q)] for p, q in d.get("asks", [])],
        }

    def create_order(self, symbol, side, type, qty, price=None,
                     stop_price=None) -> Order:
        # order param shapes per the reference executor's three order
        # kinds (trade_executor_service.py:909-992)
        params: dict = {"symbol": symbol, "side": side, "type": type,
                        "quantity": self._fmt(qty)}
        if type == "LIMIT":
            params.update(timeInForce="GTC", price=self._fmt(price))
        elif type == "STOP_LOSS_LIMIT":
            params.update(timeInForce="GTC", price=self._fmt(price),
                          stopPrice=self._fmt(stop_price))
        d = self._post("/api/v3/order", params)
        filled = float(d.get("executedQty", 0.0) or 0.0)
        quote_filled = float(d.get("cummulativeQuoteQty", 0.0) or 0.0)
        return Order(
            order_id=str(d.get("orderId", "")), symbol=symbol, side=side,
            type=type, qty=qty, price=price, stop_price=stop_price,
            status=d.get("status", "NEW"),
            filled_qty=filled,
            filled_price=(quote_filled / filled) if filled else 0.0,
        )

    def cancel_order(self, symbol, order_id) -> bool:
        try:
            d = self._delete("/api/v3/order",
                             {"symbol": symbol, "orderId": order_id})
        except BinanceError:
            return False
        return d.get("status") in ("CANCELED", "PENDING_CANCEL")

    def get_order(self, symbol, order_id) -> Order | None:
        try:
            d = self._get("/api/v3/order",
                          {"symbol": symbol, "orderId": order_id},
                          signed=True)
        except BinanceError:
            return None
        filled = float(d.get("executedQty", 0.0) or 0.0)
        quote_filled = float(d.get("cummulativeQuoteQty", 0.0) or 0.0)
        return Order(
            order_id=str(d.get("orderId", order_id)), symbol=symbol,
            side=d.get("side", ""), type=d.get("type", ""),
            qty=float(d.get("origQty", 0.0) or 0.0),
            price=float(d.get("price", 0.0) or 0.0) or None,
            stop_price=float(d.get("stopPrice", 0.0) or 0.0) or None,
            status=d.get("status", "NEW"),
            filled_qty=filled,
            filled_price=(quote_filled / filled) if filled else 0.0,
        )

    def get_balances(self) -> dict[str, float]:
        d = self._get("/api/v3/account", signed=True)
        out = {}
        for b in d.get("balances", []):
            total = float(b.get("free", 0.0)) + float(b.get("locked", 0.0))
            if total > 0:
                out[b["asset"]] = total
        return out

    def get_trading_fees(self, symbol) -> dict:
        if symbol not in self._fees:
            d = self._get("/sapi/v1/asset/tradeFee",
                          {"symbol": symbol}, signed=True)
            row = d[0] if isinstance(d, list) and d else {}
            self._fees[symbol] = {
                "maker": float(row.get("makerCommission", 0.001)),
                "taker": float(row.get("takerCommission", 0.001)),
            }
        return self._fees[symbol]

    def get_symbol_filters(self, symbol) -> dict:
        if symbol not in self._filters:
            d = self._get("/api/v3/exchangeInfo", {"symbol": symbol})
            tick = step = notional = 0.0
            for s in d.get("symbols", []):
                if s.get("symbol") != symbol:
                    continue
                for f in s.get("filters", []):
                    ft = f.get("filterType")
                    if ft == "PRICE_FILTER":
                        tick = float(f.get("tickSize", 0.0))
                    elif ft == "LOT_SIZE":
                        step = float(f.get("stepSize", 0.0))
                    elif ft in ("NOTIONAL", "MIN_NOTIONAL"):
                        notional = float(f.get("minNotional", 0.0))
            self._filters[symbol] = {"tick_size": tick,
                                     "step_size": step,
                                     "min_notional": notional}
        return self._filters[symbol]

    @staticmethod
    def _fmt(x) -> str:
        return np.format_float_positional(
            float(x), trim="-", precision=8)


def fetch_klines(symbol: str, interval: str = "1m",
                 start_ms: int | None = None, end_ms: int | None = None,
                 limit: int | None = None, transport=None,
                 base_url: str = REST_URL) -> pd.DataFrame:
    """Paginated /api/v3/klines fetch, 1000 rows per request, stitched
    into a timestamp/open/high/low/close/volume frame (reference
    backtesting/data_manager.py:47-114). Pages forward from start_ms
    until end_ms (or until the venue returns a short page)."""
    http = transport or UrllibTransport()
    step = _INTERVAL_MS.get(interval, 60_000)
    rows: list[list] = []
    cursor = start_ms
    max_pages = 5000                    # safety: ~5M rows / request run
    while max_pages > 0:
        max_pages -= 1
        params = {"symbol": symbol, "interval": interval, "limit": 1000}
        if cursor is not None:
            params["startTime"] = int(cursor)
        if end_ms is not None:
            params["endTime"] = int(end_ms)
        status, body = http.request("GET", base_url + "/api/v3/klines",
                                    params, {})
        page = BinanceExchange._parse(status, body, "/api/v3/klines")
        if not page:
            break
        rows.extend(page)
        if limit is not None and len(rows) >= limit:
            rows = rows[:limit]
            break
        if len(page) < 1000:
            break
        cursor = page[-1][0] + step
        if end_ms is not None and cursor > end_ms:
            break
    df = pd.DataFrame(
        [[r[0], float(r[1]), float(r[2]), float(r[3]), float(r[4]),
          float(r[5])] for r in rows],
        columns=["timestamp", "open", "high", "low", "close", "volume"],
    )
    return df


class BinanceWSFeed:
    """`!miniTicker@arr` frames -> per-symbol Candle stream.

    `frames` is any async iterator of websocket text messages (the
    seam): production wires a socket reader, tests replay a recorded
    JSONL tape. Mirrors the reference's handler fields — symbol `s`,
    close `c`, high `h`, low `l`, quote volume `q`
    (market_monitor_service.py:615-633)."""

    def __init__(self, frames, symbols: list[str] | None = None,
                 quote: str = "USDC"):
        self.frames = frames
        self.symbols = set(symbols) if symbols else None
        self.quote = quote
        self._t = 0

    async def __aiter__(self):
        async for msg in self.frames:
            try:
                tickers = json.loads(msg)
            except json.JSONDecodeError:
                continue
            if isinstance(tickers, dict):
                tickers = tickers.get("data", [])
            emitted = False
            for tk in tickers:
                sym = tk.get("s", "")
                if not sym.endswith(self.quote):
                    continue
                if self.symbols and sym not in self.symbols:
                    continue
                yield Candle(
                    symbol=sym, t=self._t,
                    close=float(tk.get("c", 0.0)),
                    high=float(tk.get("h", 0.0)),
                    low=float(tk.get("l", 0.0)),
                    volume=float(tk.get("q", 0.0)),
                )
                emitted = True
            if emitted:
                self._t += 1


async def jsonl_frames(path):
    """Replay helper: one websocket frame per JSONL line."""
    from pathlib import Path

    for line in Path(path).read_text().splitlines():
        if line.strip():
            yield line
