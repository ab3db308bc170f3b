"""Candle feeds for the live path.

The reference ingests Binance `!miniTicker@arr` over websocket
(market_monitor_service.py:615-633). Offline-first here: SyntheticFeed
replays a seeded GBM market (or CSV history) candle-by-candle with
configurable pacing; a live websocket feed implements the same interface
behind the exchange seam.
"""

from __future__ import annotations

import asyncio
from dataclasses import dataclass

import numpy as np


@dataclass
class Candle:
    symbol: str
    t: int                  # candle index
    close: float
    high: float
    low: float
    volume: float


class SyntheticFeed:
    """Replays (nsym, T, 4) [close, high, low, volume] candles in time
    order, all symbols per step. speed=0 -> no pacing (tests/backfill);
    speed=1 -> a candle interval per interval_s seconds."""

    def __init__(self, candles: np.ndarray, symbols: list[str],
                 start: int = 0, speed: float = 0.0,
                 interval_s: float = 60.0):
        assert candles.ndim == 3 and candles.shape[2] == 4
        assert candles.shape[0] == len(symbols)
        self.candles = candles
        self.symbols = symbols
        self.start = start
        self.speed = speed
        self.interval_s = interval_s

    async def __aiter__(self):
        nsym, T, _ = self.candles.shape
        for t in range(self.start, T):
            for s in range(nsym):
                c = self.candles[s, t]
                yield Candle(self.symbols[s], t, float(c[0]), float(c[1]),
                             float(c[2]), float(c[3]))
            if self.speed > 0:
                await asyncio.sleep(self.interval_s / self.speed)
            else:
                await asyncio.sleep(0)       # yield to the event loop
