"""Live-venue connectors (Binance REST/WS, LunarCrush, news feeds).

Importing this package registers the Binance adapter with the exchange
factory, so `ExchangeFactory.create_exchange("binance", ...)` — or
`exchange: binance` in config — is the only switch a deployment flips
(reference seam: services/utils/exchange_interface.py:209-215).
"""

from ..utils.exchange import ExchangeFactory
from .binance import BinanceExchange, BinanceWSFeed, fetch_klines
from .fixture_server import BinanceFixtureServer
from .social_news import (
    CryptoPanicClient, LiveNewsHeadlines, LiveNewsSource,
    LunarCrushClient, LunarCrushSocialSource, RssClient,
)
from .transport import RecordingTransport, ReplayTransport, UrllibTransport

ExchangeFactory.register("binance", BinanceExchange)

__all__ = [
    "BinanceExchange", "BinanceWSFeed", "fetch_klines",
    "BinanceFixtureServer", "LunarCrushClient", "CryptoPanicClient",
    "RssClient", "LiveNewsSource", "LiveNewsHeadlines",
    "LunarCrushSocialSource", "UrllibTransport", "ReplayTransport",
    "RecordingTransport",
]
