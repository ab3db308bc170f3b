"""News analysis (reference parity: services/utils/news_analyzer.py:50-915
+ services/news_analysis_service.py:33-420).

The reference pulls 4 live feeds (CryptoPanic/LunarCrush/CoinDesk/
Cointelegraph) and scores with VADER or a HF transformer. Offline-first
here: a deterministic synthetic headline generator + a self-contained
lexicon sentiment scorer with the same API surface (sentiment, entity
extraction, relevance scoring, extractive summarization, topic
extraction). A live fetcher/scorer can be registered behind the same
seams."""

from __future__ import annotations

import re
import time

import numpy as np

from ..bus.schema import Channels, Keys
from .base import Service

POSITIVE = {
    "surge", "rally", "bull", "bullish", "gain", "gains", "soar", "record",
    "adoption", "approve", "approval", "breakout", "upgrade", "partnership",
    "growth", "profit", "institutional", "etf", "green", "high", "rebound",
}
NEGATIVE = {
    "crash", "plunge", "bear", "bearish", "loss", "losses", "dump", "hack",
    "exploit", "ban", "lawsuit", "sec", "fraud", "liquidation", "fear",
    "selloff", "decline", "red", "low", "bankruptcy", "downgrade",
}
INTENSIFIERS = {"massive": 1.5, "huge": 1.4, "major": 1.3, "slight": 0.6,
                "minor": 0.6}

TOPICS = {
    "regulation": {"sec", "regulation", "ban", "lawsuit", "approval",
                   "etf", "legal"},
    "defi": {"defi", "protocol", "yield", "liquidity", "dex"},
    "security": {"hack", "exploit", "vulnerability", "breach"},
    "markets": {"rally", "crash", "volume", "breakout", "selloff",
                "liquidation"},
    "adoption": {"adoption", "partnership", "institutional", "payment"},
}

KNOWN_ENTITIES = {"bitcoin": "BTC", "btc": "BTC", "ethereum": "ETH",
                  "eth": "ETH", "solana": "SOL", "sol": "SOL",
                  "ripple": "XRP", "xrp": "XRP", "binance": "BNB",
                  "cardano": "ADA", "dogecoin": "DOGE"}


class NewsAnalyzer:
    """Lexicon sentiment + entity/relevance/summary/topics
    (news_analyzer.py:409/:502/:554/:596/:644 API surface)."""

    def sentiment(self, text: str) -> float:
        """[0, 1] with 0.5 neutral (VADER-compound-shaped)."""
        words = re.findall(r"[a-z']+", text.lower())
        score = 0.0
        mult = 1.0
        for w in words:
            if w in INTENSIFIERS:
                mult = INTENSIFIERS[w]
                continue
            if w in POSITIVE:
                score += mult
            elif w in NEGATIVE:
                score -= mult
            mult = 1.0
        norm = score / max(np.sqrt(len(words)), 1.0)
        return float(np.clip(0.5 + norm * 0.35, 0.0, 1.0))

    def entities(self, text: str) -> list[str]:
        """Regex/dict entity extraction (:502)."""
        found = []
        for w in re.findall(r"[A-Za-z]+", text):
            t = KNOWN_ENTITIES.get(w.lower())
            if t and t not in found:
                found.append(t)
        # $TICKER mentions
        for m in re.findall(r"\$([A-Z]{2,6})\b", text):
            if m not in found:
                found.append(m)
        return found

    def relevance(self, text: str, symbol: str) -> float:
        """(:554) relevance of an article to a symbol."""
        base = symbol.replace("USDC", "").replace("USDT", "")
        ents = self.entities(text)
        if base in ents:
            return 1.0 if len(ents) == 1 else 0.7
        return 0.1 if ents else 0.3

    def summarize(self, texts: list[str], k: int = 3) -> list[str]:
        """Extractive: pick the k most lexicon-loaded sentences (:596)."""
        scored = sorted(
            texts, key=lambda t: abs(self.sentiment(t) - 0.5), reverse=True)
        return scored[:k]

    def topics(self, text: str) -> list[str]:
        words = set(re.findall(r"[a-z]+", text.lower()))
        return [t for t, kws in TOPICS.items() if words & kws]


class SyntheticNewsSource:
    """Deterministic headline stream whose tone drifts with a seeded
    random walk per symbol (offline stand-in for the 4 live feeds)."""

    POS_T = ["{sym} surges to new record as institutional adoption grows",
             "Major partnership sends {sym} soaring",
             "{sym} rally continues on ETF approval hopes"]
    NEG_T = ["{sym} plunges amid massive liquidations",
             "Exchange hack triggers {sym} selloff",
             "SEC lawsuit fears drive {sym} to new low"]
    NEU_T = ["{sym} trades sideways as volume declines",
             "Analysts split on {sym} outlook"]

    def __init__(self, seed: int = 0):
        self.rng = np.random.default_rng(seed)
        self.tone: dict[str, float] = {}

    def headlines(self, symbol: str, n: int = 3) -> list[str]:
        base = symbol.replace("USDC", "")
        tone = self.tone.get(symbol, 0.0) + \
            self.rng.standard_normal() * 0.3
        tone = float(np.clip(tone, -1, 1))
        self.tone[symbol] = tone
        out = []
        for _ in range(n):
            u = self.rng.random() + tone * 0.3
            tpl = (self.POS_T if u > 0.66 else
                   (self.NEG_T if u < 0.33 else self.NEU_T))
            out.append(str(self.rng.choice(tpl)).format(sym=base))
        return out


class NewsAnalysisService(Service):
    """Per-symbol news scoring + market-wide summary
    (news_analysis_service.py:125-343); writes the `news_analysis` hash the
    analyzer service consumes (ai_analyzer_service.py:429)."""

    name = "news_analysis"

    def __init__(self, bus, config=None, source=None):
        super().__init__(bus, config)
        self.analyzer = NewsAnalyzer()
        self.source = source or SyntheticNewsSource(self.config.seed)
        self.tracked: set[str] = set()
        self.analyzed = 0

    def run_tasks(self):
        return [self._track(), self._news_loop()]

    async def _track(self):
        sub = self.bus.subscribe(Channels.MARKET_UPDATES)

        def on_msg(_, m):
            if m.get("symbol"):
                self.tracked.add(m["symbol"])

        await self.consume(sub, on_msg)

    async def _news_loop(self):
        while self.running:
            all_heads = []
            for sym in sorted(self.tracked):
                heads = self.source.headlines(sym)
                sents = [self.analyzer.sentiment(h) for h in heads]
                entry = {
                    "symbol": sym,
                    "sentiment": float(np.mean(sents)) if sents else 0.5,
                    "n_articles": len(heads),
                    "topics": sorted({t for h in heads
                                      for t in self.analyzer.topics(h)}),
                    "top_headlines": self.analyzer.summarize(heads, 2),
                    "at": time.time(),
                }
                await self.bus.hset(Keys.NEWS_ANALYSIS, sym, entry)
                all_heads.extend(heads)
                self.analyzed += 1
            if all_heads:
                await self.bus.set(Keys.NEWS_SUMMARY_REPORT, {
                    "at": time.time(),
                    "market_sentiment": float(np.mean(
                        [self.analyzer.sentiment(h) for h in all_heads])),
                    "summary": self.analyzer.summarize(all_heads, 3),
                })
                await self.bus.publish(Channels.NEWS_ANALYSIS_UPDATES, {
                    "n": len(all_heads), "at": time.time(),
                })
            await self.sleep(3.0)

    async def run(self):
        pass
