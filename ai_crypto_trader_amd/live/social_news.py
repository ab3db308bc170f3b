"""Live social / news connectors behind the existing service seams.

  - LunarCrushClient: v4 /public/coins/<sym>/v1 asset metrics and
    /public/category/cryptocurrencies/news/v1 feeds with a TTL cache
    (reference: services/social_monitor_service.py:95-186).
  - CryptoPanicClient + RssClient: the reference's four news sources —
    CryptoPanic API, LunarCrush feeds, CoinDesk RSS, Cointelegraph RSS
    (services/utils/news_analyzer.py:178-448). RSS parsed with stdlib
    ElementTree; no feedparser dependency.

Both return plain dicts in the shapes services/social.py and
services/news.py consume, so swapping the offline synthetic sources for
these clients is a constructor argument. Offline tests replay recorded
tapes through live.transport.ReplayTransport.
"""

from __future__ import annotations

import html
import json
import re
import time
import xml.etree.ElementTree as ET

from .transport import UrllibTransport

LUNARCRUSH_URL = "https://lunarcrush.com/api4"
CRYPTOPANIC_URL = "https://cryptopanic.com/api/v1"
COINDESK_RSS = "https://www.coindesk.com/arc/outboundfeeds/rss/"
COINTELEGRAPH_RSS = "https://cointelegraph.com/rss"


class LunarCrushClient:
    def __init__(self, api_key: str = "", transport=None,
                 base_url: str = LUNARCRUSH_URL, cache_ttl: float = 300.0,
                 time_fn=time.time):
        self.key = api_key
        self.http = transport or UrllibTransport()
        self.base = base_url.rstrip("/")
        self.ttl = cache_ttl
        self.time_fn = time_fn
        self._cache: dict[str, tuple[float, dict]] = {}

    def _get(self, path: str) -> dict:
        now = self.time_fn()
        hit = self._cache.get(path)
        if hit and now - hit[0] < self.ttl:
            return hit[1]
        headers = {"Authorization": f"Bearer {self.key}"} \
            if self.key else {}
        status, body = self.http.request("GET", self.base + path, None,
                                         headers)
        data = json.loads(body) if status == 200 and body else {}
        self._cache[path] = (now, data)
        return data

    def asset_metrics(self, symbol: str) -> dict:
        """Raw metrics dict in the shape services/social.py's source
        seam expects (reference field mapping:
        social_monitor_service.py:120-160)."""
        base = symbol
        for q in ("USDC", "USDT", "BUSD"):
            if base.endswith(q):
                base = base[:-len(q)]
                break
        d = self._get(f"/public/coins/{base}/v1").get("data", {})
        return {
            "social_volume": float(d.get("social_volume_24h", 0.0)),
            "engagement": float(d.get("interactions_24h", 0.0)),
            "contributors": float(d.get("social_contributors", 0.0)),
            "sentiment": float(d.get("sentiment", 50.0)) / 100.0,
            "twitter_volume": float(d.get("num_posts", 0.0)),
            "reddit_volume": float(d.get("reddit_posts", 0.0)),
            "news_volume": float(d.get("news", 0.0)),
            "galaxy_score": float(d.get("galaxy_score", 0.0)),
            "alt_rank": float(d.get("alt_rank", 0.0)),
        }

    def feeds(self, limit: int = 20) -> list[dict]:
        d = self._get("/public/category/cryptocurrencies/news/v1")
        out = []
        for item in (d.get("data") or [])[:limit]:
            out.append({
                "title": item.get("post_title", ""),
                "url": item.get("post_link", ""),
                "source": "lunarcrush",
                "published": float(item.get("post_created", 0.0)),
                "sentiment": float(item.get("post_sentiment", 3.0)),
            })
        return out


class CryptoPanicClient:
    def __init__(self, api_key: str = "", transport=None,
                 base_url: str = CRYPTOPANIC_URL):
        self.key = api_key
        self.http = transport or UrllibTransport()
        self.base = base_url.rstrip("/")

    def posts(self, currency: str | None = None,
              limit: int = 50) -> list[dict]:
        params = {"auth_token": self.key, "public": "true"}
        if currency:
            params["currencies"] = currency
        status, body = self.http.request(
            "GET", self.base + "/posts/", params, {})
        if status != 200:
            return []
        data = json.loads(body) if body else {}
        out = []
        for r in (data.get("results") or [])[:limit]:
            out.append({
                "title": r.get("title", ""),
                "url": r.get("url", ""),
                "source": "cryptopanic",
                "published": r.get("published_at", ""),
                "currencies": [c.get("code", "")
                               for c in r.get("currencies", []) or []],
                "votes": r.get("votes", {}),
            })
        return out


_TAG_RE = re.compile(r"<[^>]+>")


class RssClient:
    """CoinDesk / Cointelegraph RSS (reference
    news_analyzer.py:270-368) via stdlib XML parsing."""

    def __init__(self, transport=None):
        self.http = transport or UrllibTransport()

    def fetch(self, url: str, source: str,
              limit: int = 30) -> list[dict]:
        status, body = self.http.request("GET", url, None, {})
        if status != 200 or not body:
            return []
        try:
            root = ET.fromstring(body)
        except ET.ParseError:
            return []
        items = []
        for item in root.iter("item"):
            if len(items) >= limit:
                break
            title = (item.findtext("title") or "").strip()
            desc = _TAG_RE.sub(" ", item.findtext("description") or "")
            items.append({
                "title": html.unescape(title),
                "summary": html.unescape(desc).strip(),
                "url": (item.findtext("link") or "").strip(),
                "source": source,
                "published": (item.findtext("pubDate") or "").strip(),
            })
        return items

    def coindesk(self, limit: int = 30) -> list[dict]:
        return self.fetch(COINDESK_RSS, "coindesk", limit)

    def cointelegraph(self, limit: int = 30) -> list[dict]:
        return self.fetch(COINTELEGRAPH_RSS, "cointelegraph", limit)


class LunarCrushSocialSource:
    """Drop-in for services.social.SyntheticSocialSource: same
    `.metrics(symbol, t)` / `.observe_price` interface, metrics pulled
    live from LunarCrush (reference social_monitor_service.py:95-186).
    Swap via SocialMonitorService(..., source=LunarCrushSocialSource())."""

    def __init__(self, api_key: str = "", transport=None,
                 client: LunarCrushClient | None = None):
        self.client = client or LunarCrushClient(api_key, transport)

    def observe_price(self, symbol: str, ret: float):
        pass                      # live source needs no price coupling

    def metrics(self, symbol: str, t: int):
        from ..bus.schema import SocialMetricsBlock

        m = self.client.asset_metrics(symbol)
        return SocialMetricsBlock(
            social_volume=m["social_volume"],
            social_engagement=m["engagement"],
            social_contributors=m["contributors"],
            social_sentiment=m["sentiment"],
            twitter_volume=m["twitter_volume"],
            reddit_volume=m["reddit_volume"],
            news_volume=m["news_volume"],
        )


class LiveNewsHeadlines:
    """Drop-in for services.news.SyntheticNewsSource: `.headlines(sym)`
    from the four live sources (news_analyzer.py:178-448)."""

    def __init__(self, transport=None, cryptopanic_key: str = "",
                 lunarcrush_key: str = ""):
        self.src = LiveNewsSource(transport, cryptopanic_key,
                                  lunarcrush_key)

    def headlines(self, symbol: str, n: int = 3) -> list[str]:
        base = symbol
        for q in ("USDC", "USDT", "BUSD"):
            if base.endswith(q):
                base = base[:-len(q)]
                break
        items = self.src.fetch_all(base)
        ranked = [i["title"] for i in items
                  if base.lower() in i["title"].lower()]
        rest = [i["title"] for i in items if i["title"] not in ranked]
        return (ranked + rest)[:n]


class LiveNewsSource:
    """Aggregates the four reference news sources into the item-dict
    list services/news.py scores (news_analyzer.py:178-448)."""

    def __init__(self, transport=None, cryptopanic_key: str = "",
                 lunarcrush_key: str = ""):
        self.cp = CryptoPanicClient(cryptopanic_key, transport)
        self.rss = RssClient(transport)
        self.lc = LunarCrushClient(lunarcrush_key, transport)

    def fetch_all(self, currency: str | None = None) -> list[dict]:
        items: list[dict] = []
        items += self.cp.posts(currency)
        items += self.rss.coindesk()
        items += self.rss.cointelegraph()
        items += self.lc.feeds()
        return items
