"""Trade executor service (reference parity:
services/trade_executor_service.py:401-1428 + TrailingStopManager :55-399).

Subscribes `trading_signals` + `strategy_update`, gates on confidence and
portfolio-VaR / correlation conditions (:719-787), executes through the
exchange seam (FakeExchange by default), manages SL/TP/trailing stops, and
maintains the `holdings` / `active_trades` / `trailing_stops` keys."""

from __future__ import annotations

import json
import time
from decimal import Decimal

from ..bus.schema import Channels, Keys
from ..utils.exchange import ExchangeInterface
from .base import Service


def _quantize_down(value: float, step: float) -> float:
    """Floor `value` to a multiple of `step` in decimal arithmetic —
    binary-float division (0.3/0.1 == 2.999...) would floor one step low
    and carry residue a live venue's filters reject."""
    v = Decimal(str(value))
    s = Decimal(str(step))
    return float((v // s) * s)


def round_to_filters(qty: float, price: float, filters: dict):
    """Tick/step rounding (reference trade_executor_service.py:789-797):
    qty floored to step_size, price floored to tick_size (Decimal
    quantization, as live-venue filter code must); returns
    (qty, price, ok) with ok=False below min notional."""
    step = filters.get("step_size", 0.0) or 0.0
    tick = filters.get("tick_size", 0.0) or 0.0
    if step > 0:
        qty = _quantize_down(qty, step)
    if tick > 0:
        price = _quantize_down(price, tick)
    ok = qty * price >= filters.get("min_notional", 0.0)
    return qty, price, ok


class TrailingStopManager:
    """4 trailing strategies (trade_executor_service.py:168-247):
    percent_based / atr_based / volatility_based / fixed_amount, with an
    activation threshold and monotonic stop raises (:333-371)."""

    def __init__(self, strategy: str = "percent_based",
                 trail_pct: float = 0.02, activation_pct: float = 0.01,
                 fixed_amount: float = 0.0, atr_mult: float = 2.0):
        self.strategy = strategy
        self.trail_pct = trail_pct
        self.activation_pct = activation_pct
        self.fixed_amount = fixed_amount
        self.atr_mult = atr_mult
        self.stops: dict[str, dict] = {}

    def register(self, symbol: str, entry_price: float, stop_price: float,
                 atr: float = 0.0):
        self.stops[symbol] = {
            "entry": entry_price, "stop": stop_price, "peak": entry_price,
            "atr": atr, "active": False, "updates": 0,
        }

    def remove(self, symbol: str):
        self.stops.pop(symbol, None)

    def update(self, symbol: str, price: float,
               recent_vol: float = 0.0) -> float | None:
        """Returns a new (raised) stop price or None."""
        s = self.stops.get(symbol)
        if s is None:
            return None
        s["peak"] = max(s["peak"], price)
        if not s["active"]:
            if s["peak"] >= s["entry"] * (1 + self.activation_pct):
                s["active"] = True
            else:
                return None
        if self.strategy == "percent_based":
            cand = s["peak"] * (1 - self.trail_pct)
        elif self.strategy == "atr_based":
            cand = s["peak"] - self.atr_mult * s["atr"]
        elif self.strategy == "volatility_based":
            cand = s["peak"] * (1 - max(recent_vol * 2.0, 0.005))
        else:  # fixed_amount
            cand = s["peak"] - self.fixed_amount
        if cand > s["stop"]:
            s["stop"] = cand
            s["updates"] += 1
            return cand
        return None

    def status(self) -> dict:
        return {k: dict(v) for k, v in self.stops.items()}


class TradeExecutorService(Service):
    name = "trade_executor"

    def __init__(self, bus, exchange: ExchangeInterface, config=None):
        super().__init__(bus, config)
        self.exchange = exchange
        self.trailing = TrailingStopManager()
        self.active: dict[str, dict] = {}     # symbol -> trade record
        self.strategy_params: dict = {}
        self.trades_done = 0
        self.day_start_value: float | None = None
        self.day_start_at = 0.0

    def run_tasks(self):
        return [self._consume_signals(), self._monitor_trades()]

    def _portfolio_value(self) -> float:
        balances = self.exchange.get_balances()
        total = balances.get(self.config.trading.quote_asset, 0.0)
        for asset, qty in balances.items():
            if asset != self.config.trading.quote_asset:
                total += qty * self.exchange.get_ticker(
                    asset + self.config.trading.quote_asset)["price"]
        return total

    def daily_drawdown_exceeded(self) -> bool:
        """Max-daily-drawdown halt (trading_strategy.md: 6%): stop
        opening positions when today's portfolio drawdown from the
        day-open anchor passes the configured limit. The anchor is the
        value at the start of the 24h window (NOT the intraday peak —
        day-anchor semantics per trading_strategy.md)."""
        now = time.time()
        value = self._portfolio_value()
        if self.day_start_value is None or now - self.day_start_at > 86_400:
            self.day_start_value = value
            self.day_start_at = now
            return False
        dd = 1.0 - value / max(self.day_start_value, 1e-9)
        return dd > self.config.risk.max_daily_drawdown_pct

    # --- gates (trade_executor_service.py:719-787) -----------------------
    async def check_trading_conditions(self, signal: dict) -> tuple[bool, str]:
        if len(self.active) >= self.config.trading.max_positions:
            return False, "max_positions"
        if self.daily_drawdown_exceeded():
            return False, "daily_drawdown_halt"
        if signal["symbol"] in self.active:
            return False, "already_in_position"
        risk = await self.bus.get_json(Keys.PORTFOLIO_RISK)
        if risk and risk.get("portfolio_var_pct", 0.0) > \
                self.config.risk.max_portfolio_var_pct * 100:
            return False, "portfolio_var_exceeded"
        if risk:
            corr_block = risk.get("high_correlation_symbols", [])
            if signal["symbol"] in corr_block:
                return False, "correlation_limit"
        return True, "ok"

    async def _get_risk_info(self, symbol: str) -> dict:
        sl = await self.bus.get_json(Keys.ADAPTIVE_STOP_LOSSES) or {}
        return sl.get(symbol, {})

    async def execute_buy(self, signal: dict):
        sym = signal["symbol"]
        price = self.exchange.get_ticker(sym)["price"]
        if price <= 0:
            return
        balances = self.exchange.get_balances()
        quote = self.config.trading.quote_asset
        cash = balances.get(quote, 0.0)
        risk_info = signal.get("risk_info", {})
        pos_pct = risk_info.get("optimal_position_pct",
                                self.config.risk.fixed_position_pct)
        # social risk adjustment (trade_executor_service.py:799-814)
        sra = await self.bus.hget(Keys.SOCIAL_RISK_ADJUSTMENTS, sym)
        if sra:
            adj = json.loads(sra)
            pos_pct *= adj.get("position_multiplier", 1.0)
        cost = min(cash * pos_pct, cash)
        if cost < self.config.trading.min_trade_usd:
            return      # minimum trade amount (trading_strategy.md)
        qty = cost / price
        filters = self.exchange.get_symbol_filters(sym) \
            if hasattr(self.exchange, "get_symbol_filters") else {}
        qty, _, ok = round_to_filters(qty, price, filters)
        if not ok or qty <= 0:
            return
        order = self.exchange.create_order(sym, "BUY", "MARKET", qty)
        if order.status != "FILLED":
            return
        # the fill credits base NET of the taker fee — size the protective
        # stop at what we actually hold, or it can never execute
        fee = self.exchange.get_trading_fees(sym).get("taker", 0.0)
        qty = order.filled_qty * (1 - fee)
        stop_pct = risk_info.get("adaptive_stop_pct",
                                 self.config.risk.base_stop_loss_pct)
        stop = price * (1 - stop_pct)
        tp = price * (1 + 2 * stop_pct)        # 2:1 RR (PositionSizer :251)
        # STOP_LOSS_LIMIT at stop*0.99 limit (reference :980)
        stop_order = self.exchange.create_order(
            sym, "SELL", "STOP_LOSS_LIMIT", qty,
            price=stop * 0.99, stop_price=stop)
        self.trailing.register(sym, price, stop)
        self.active[sym] = {
            "symbol": sym, "qty": qty, "entry_price": price,
            "stop_price": stop, "tp_price": tp,
            "stop_order_id": stop_order.order_id,
            "opened_at": time.time(),
            "signal_confidence": signal.get("confidence", 0.0),
        }
        self.trades_done += 1
        self.metrics.executions.labels(sym, "BUY").inc()
        await self._write_state()
        await self.bus.publish(Channels.TRADE_EXECUTIONS, {
            "symbol": sym, "side": "BUY", "qty": qty, "price": price,
        })

    async def execute_sell(self, sym: str, reason: str):
        trade = self.active.pop(sym, None)
        if trade is None:
            return
        price = self.exchange.get_ticker(sym)["price"]
        # cancel the resting protective stop first (reference :333-371:
        # order replacement discipline — never leave an orphaned stop)
        oid = trade.get("stop_order_id")
        stop_o = self.exchange.get_order(sym, oid) if oid else None
        if stop_o is not None and stop_o.status == "FILLED":
            # the exchange-side stop already closed the position
            price = stop_o.filled_price
            reason = "stop_loss"
        else:
            if oid:
                self.exchange.cancel_order(sym, oid)
            o = self.exchange.create_order(sym, "SELL", "MARKET",
                                           trade["qty"])
            if o.status == "FILLED":
                price = o.filled_price
        self.trailing.remove(sym)
        self.trades_done += 1
        self.metrics.executions.labels(sym, "SELL").inc()
        await self._write_state()
        await self.bus.publish(Channels.TRADE_EXECUTIONS, {
            "symbol": sym, "side": "SELL", "qty": trade["qty"],
            "price": price, "reason": reason,
            "pnl_pct": (price / trade["entry_price"] - 1) * 100,
        })

    async def _write_state(self):
        """holdings / active_trades / trailing_stops keys
        (trade_executor_service.py:709-714, :1212, :1126)."""
        balances = self.exchange.get_balances()
        prices = {s: self.exchange.get_ticker(s)["price"]
                  for s in self.active}
        total = balances.get(self.config.trading.quote_asset, 0.0)
        holdings = {}
        for asset, qty in balances.items():
            if asset == self.config.trading.quote_asset:
                holdings[asset] = {"qty": qty, "value": qty}
            else:
                p = prices.get(asset + self.config.trading.quote_asset, 0.0)
                holdings[asset] = {"qty": qty, "value": qty * p}
                total += qty * p
        await self.bus.set(Keys.HOLDINGS, {
            "holdings": holdings, "total_value": total,
            "timestamp": time.time(),
        })
        await self.bus.set(Keys.ACTIVE_TRADES, self.active)
        await self.bus.set(Keys.TRAILING_STOPS, self.trailing.status())
        self.metrics.portfolio_value.set(total)
        self.metrics.active_trades.set(len(self.active))

    async def cleanup_positions(self) -> int:
        """Startup liquidation (reference trade_executor_service.py:
        488-547): sell any non-quote balances to the quote asset before
        trading begins, so the executor starts from a known-flat book.
        Returns the number of positions liquidated."""
        balances = self.exchange.get_balances()
        quote = self.config.trading.quote_asset
        n = 0
        for asset, qty in balances.items():
            if asset == quote or qty <= 0:
                continue
            sym = asset + quote
            price = self.exchange.get_ticker(sym)["price"]
            if price <= 0 or qty * price < self.config.trading.min_trade_usd:
                continue           # dust stays (reference skips dust too)
            filters = self.exchange.get_symbol_filters(sym)
            sell_qty, _, ok = round_to_filters(qty, price, filters)
            if not ok or sell_qty <= 0:
                continue
            o = self.exchange.create_order(sym, "SELL", "MARKET", sell_qty)
            if o.status == "FILLED":
                n += 1
                self.log.info("startup liquidation: sold %s %s @ %s",
                              sell_qty, asset, o.filled_price)
        if n:
            await self._write_state()
        return n

    # --- loops -----------------------------------------------------------
    async def _consume_signals(self):
        sub = self.bus.subscribe(Channels.TRADING_SIGNALS,
                                 Channels.RISK_ENRICHED_SIGNALS,
                                 Channels.STRATEGY_UPDATE)
        min_conf = self.config.trading.min_confidence

        async def on_msg(chan, msg):
            if chan == Channels.STRATEGY_UPDATE:
                # 'reload' hot-swap trigger (strategy_evolution:356)
                params = await self.bus.get_json(Keys.STRATEGY_PARAMS)
                if params:
                    self.strategy_params = params
                    self.log.info("strategy params hot-swapped")
                return
            if not isinstance(msg, dict) or "decision" not in msg:
                return
            sym = msg.get("symbol")
            if msg["decision"] == "BUY" and msg.get("confidence", 0) \
                    >= min_conf:
                ok, why = await self.check_trading_conditions(msg)
                if ok:
                    await self.execute_buy(msg)
                else:
                    self.log.debug("buy blocked: %s", why)
            elif msg["decision"] == "SELL" and sym in self.active:
                await self.execute_sell(sym, "signal")

        await self.consume(sub, on_msg)

    async def _monitor_trades(self):
        """SL/TP/trailing monitor loop (reference :1104-1217)."""
        await self._write_state()          # publish holdings at startup
        last_state = 0.0
        while self.running:
            if time.monotonic() - last_state > 2.0:
                last_state = time.monotonic()
                await self._write_state()
            for sym in list(self.active):
                trade = self.active.get(sym)
                if trade is None:
                    continue
                price = self.exchange.get_ticker(sym)["price"]
                if price <= 0:
                    continue
                new_stop = self.trailing.update(sym, price)
                if new_stop is not None:
                    trade["stop_price"] = new_stop
                    # replace the exchange-side stop at the raised level
                    # (reference :333-371)
                    oid = trade.get("stop_order_id")
                    if oid and self.exchange.cancel_order(sym, oid):
                        o = self.exchange.create_order(
                            sym, "SELL", "STOP_LOSS_LIMIT", trade["qty"],
                            price=new_stop * 0.99, stop_price=new_stop)
                        trade["stop_order_id"] = o.order_id
                if price <= trade["stop_price"]:
                    await self.execute_sell(sym, "stop_loss")
                elif price >= trade["tp_price"]:
                    await self.execute_sell(sym, "take_profit")
            await self.sleep(0.1)

    async def run(self):
        pass
