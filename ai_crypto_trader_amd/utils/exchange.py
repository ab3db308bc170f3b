"""Exchange abstraction (reference parity: services/utils/
exchange_interface.py:10-219 — abc ExchangeInterface, factory) plus what
the reference lacks for offline work: a deterministic FakeExchange with a
simulation fill engine (modeled on the grid bot's simulated fills,
grid_trading_strategy.py:679-780), which is what tests/backtests/paper
trading run against. A live-exchange adapter can be registered with the
factory without touching callers."""

from __future__ import annotations

import itertools
import time
from abc import ABC, abstractmethod
from dataclasses import dataclass, field


@dataclass
class Order:
    order_id: str
    symbol: str
    side: str                    # BUY | SELL
    type: str                    # MARKET | LIMIT | STOP_LOSS_LIMIT
    qty: float
    price: float | None = None
    stop_price: float | None = None
    status: str = "NEW"          # NEW | FILLED | CANCELED
    filled_qty: float = 0.0
    filled_price: float = 0.0
    ts: float = field(default_factory=time.time)


class ExchangeInterface(ABC):
    """exchange_interface.py:10-65 surface."""

    @abstractmethod
    def get_ticker(self, symbol: str) -> dict: ...

    @abstractmethod
    def get_order_book(self, symbol: str, limit: int = 100) -> dict: ...

    @abstractmethod
    def create_order(self, symbol: str, side: str, type: str, qty: float,
                     price: float | None = None,
                     stop_price: float | None = None) -> Order: ...

    @abstractmethod
    def cancel_order(self, symbol: str, order_id: str) -> bool: ...

    @abstractmethod
    def get_order(self, symbol: str, order_id: str) -> Order | None:
        """Order status lookup (reference BinanceExchange get_order,
        exchange_interface.py:67-207) — callers must use this instead of
        touching adapter internals."""
        ...

    @abstractmethod
    def get_balances(self) -> dict[str, float]: ...

    @abstractmethod
    def get_trading_fees(self, symbol: str) -> dict: ...

    def get_symbol_filters(self, symbol: str) -> dict:
        """Tick/step/min-notional filters; adapters override with the
        venue's real values (reference exchange filters :789-797)."""
        return {"tick_size": 0.0, "step_size": 0.0, "min_notional": 0.0}


class FakeExchange(ExchangeInterface):
    """Deterministic in-memory exchange driven by set_price() ticks.

    MARKET orders fill at the current price; LIMIT/STOP_LOSS_LIMIT orders
    rest and fill when on_tick() sees the trigger condition — the fill
    semantics the reference's executor assumes of Binance
    (trade_executor_service.py:816-1046)."""

    def __init__(self, fee_rate: float = 0.001, quote: str = "USDC",
                 initial_balance: float = 10_000.0):
        self.fee = fee_rate
        self.quote = quote
        self.prices: dict[str, float] = {}
        self.balances: dict[str, float] = {quote: initial_balance}
        self.orders: dict[str, Order] = {}
        self.fills: list[Order] = []
        self._ids = itertools.count(1)

    # --- market data -----------------------------------------------------
    def set_price(self, symbol: str, price: float):
        self.prices[symbol] = price
        self.on_tick(symbol)

    def get_ticker(self, symbol: str) -> dict:
        p = self.prices.get(symbol, 0.0)
        return {"symbol": symbol, "price": p, "bid": p * 0.9999,
                "ask": p * 1.0001}

    def get_order_book(self, symbol: str, limit: int = 100) -> dict:
        p = self.prices.get(symbol, 1.0)
        bids = [[p * (1 - 0.0001 * (i + 1)), 1.0 + 0.1 * i]
                for i in range(limit)]
        asks = [[p * (1 + 0.0001 * (i + 1)), 1.0 + 0.1 * i]
                for i in range(limit)]
        return {"symbol": symbol, "bids": bids, "asks": asks}

    # --- trading ---------------------------------------------------------
    def _base(self, symbol: str) -> str:
        return symbol[:-len(self.quote)] if symbol.endswith(self.quote) \
            else symbol

    def _fill(self, o: Order, price: float):
        base = self._base(o.symbol)
        if o.side == "BUY":
            cost = o.qty * price
            if self.balances.get(self.quote, 0.0) < cost:
                o.status = "CANCELED"
                return
            self.balances[self.quote] -= cost
            self.balances[base] = self.balances.get(base, 0.0) \
                + o.qty * (1 - self.fee)
        else:
            if self.balances.get(base, 0.0) < o.qty:
                o.status = "CANCELED"
                return
            self.balances[base] -= o.qty
            self.balances[self.quote] = self.balances.get(self.quote, 0.0) \
                + o.qty * price * (1 - self.fee)
        o.status = "FILLED"
        o.filled_qty = o.qty
        o.filled_price = price
        self.fills.append(o)

    def create_order(self, symbol, side, type, qty, price=None,
                     stop_price=None) -> Order:
        o = Order(order_id=str(next(self._ids)), symbol=symbol, side=side,
                  type=type, qty=qty, price=price, stop_price=stop_price)
        self.orders[o.order_id] = o
        if type == "MARKET":
            self._fill(o, self.prices.get(symbol, price or 0.0))
        return o

    def on_tick(self, symbol: str):
        """Resting-order fill engine."""
        p = self.prices[symbol]
        for o in list(self.orders.values()):
            if o.symbol != symbol or o.status != "NEW":
                continue
            if o.type == "LIMIT":
                if (o.side == "SELL" and p >= o.price) or \
                        (o.side == "BUY" and p <= o.price):
                    self._fill(o, o.price)
            elif o.type == "STOP_LOSS_LIMIT":
                if o.side == "SELL" and p <= o.stop_price:
                    self._fill(o, o.price if o.price else p)

    def cancel_order(self, symbol, order_id) -> bool:
        o = self.orders.get(order_id)
        if o and o.status == "NEW":
            o.status = "CANCELED"
            return True
        return False

    def get_order(self, symbol, order_id) -> Order | None:
        return self.orders.get(order_id)

    def get_balances(self) -> dict[str, float]:
        return dict(self.balances)

    def get_trading_fees(self, symbol) -> dict:
        return {"maker": self.fee, "taker": self.fee}

    def get_symbol_filters(self, symbol) -> dict:
        """Exchange trading filters (reference tick/step rounding
        :789-797): price tick size, quantity step size, min notional."""
        return {"tick_size": 1e-6, "step_size": 1e-6, "min_notional": 1e-6}

    def portfolio_value(self) -> float:
        v = self.balances.get(self.quote, 0.0)
        for asset, qty in self.balances.items():
            if asset == self.quote:
                continue
            v += qty * self.prices.get(asset + self.quote, 0.0)
        return v


class ExchangeFactory:
    """exchange_interface.py:209-215 pattern."""

    _registry: dict[str, type] = {"fake": FakeExchange}

    @classmethod
    def register(cls, name: str, impl: type):
        cls._registry[name] = impl

    @classmethod
    def create_exchange(cls, name: str = "fake", **kw) -> ExchangeInterface:
        if name not in cls._registry and name == "binance":
            from .. import live  # noqa: F401  registers the adapter
        if name not in cls._registry:
            raise ValueError(
                f"unknown exchange '{name}' (known: {list(cls._registry)})")
        return cls._registry[name](**kw)
