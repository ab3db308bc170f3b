"""Technical analysis toolkit (reference parity: binance_ml_strategy.py).

  TechnicalAnalyzer  :14-249  full indicator set + trend/volatility/
                              support-resistance accessors
  PositionSizer      :251-291 volatility-tiered sizing, 2:1 TP/SL
  CryptoScanner      :293-468 market scan + 0-100 opportunity score —
                              the reference's ThreadPool(10) scan becomes
                              one batched GPU indicator pass over all
                              symbols when a GPU is present
  TradingSignal      :470-581 6-indicator voting + 0-100 strength
"""

from __future__ import annotations

import numpy as np

from .ops import gpu_available
from .ops.indicators import IND_NAMES, indicators_cpu


class TechnicalAnalyzer:
    """Wraps the indicator engine for one symbol's candle history.
    candles: (T, 4) [close, high, low, volume] float32."""

    def __init__(self, candles: np.ndarray):
        self.candles = np.asarray(candles, np.float32)
        self.ind = indicators_cpu(self.candles[None])[0]   # (T, 13)

    def _col(self, name: str) -> np.ndarray:
        return self.ind[:, IND_NAMES.index(name)]

    # --- accessors (:40-182) --------------------------------------------
    def sma(self, n: int) -> float:
        return float(self.candles[-n:, 0].mean())

    def ema12(self) -> float:
        return float(self._col("ema12")[-1])

    def ema26(self) -> float:
        return float(self._col("ema26")[-1])

    def macd(self) -> tuple[float, float, float]:
        return (float(self._col("macd")[-1]),
                float(self._col("macd_signal")[-1]),
                float(self._col("macd_hist")[-1]))

    def rsi(self) -> float:
        return float(self._col("rsi14")[-1])

    def stochastic(self) -> float:
        return float(self._col("stoch_k")[-1])

    def williams_r(self) -> float:
        return float(self._col("williams_r")[-1])

    def bollinger(self) -> dict:
        up, mid, lo = (float(self._col("bb_up")[-1]),
                       float(self._col("bb_mid")[-1]),
                       float(self._col("bb_lo")[-1]))
        close = float(self.candles[-1, 0])
        width = (up - lo) / max(mid, 1e-9)
        pos = (close - lo) / max(up - lo, 1e-9)
        return {"upper": up, "middle": mid, "lower": lo,
                "width": width, "position": pos}

    def atr(self) -> float:
        return float(self._col("atr14")[-1])

    def ichimoku(self) -> dict:
        """Ichimoku cloud (reference binance_ml_strategy.py:40-182 via the
        `ta` lib): tenkan(9) / kijun(26) midpoints, senkou A/B spans,
        chikou = close 26 back, and the position vs the cloud."""
        h, lo = self.candles[:, 1], self.candles[:, 2]

        def midpoint(n):
            return (h[-n:].max() + lo[-n:].min()) / 2.0

        tenkan = float(midpoint(min(9, len(h))))
        kijun = float(midpoint(min(26, len(h))))
        senkou_a = (tenkan + kijun) / 2.0
        senkou_b = float(midpoint(min(52, len(h))))
        close = float(self.candles[-1, 0])
        chikou = float(self.candles[-min(26, len(h)), 0])
        top, bot = max(senkou_a, senkou_b), min(senkou_a, senkou_b)
        position = ("above_cloud" if close > top else
                    "below_cloud" if close < bot else "in_cloud")
        return {"tenkan": tenkan, "kijun": kijun, "senkou_a": senkou_a,
                "senkou_b": senkou_b, "chikou": chikou,
                "position": position,
                "bullish": close > top and tenkan > kijun}

    def vwap(self) -> float:
        return float(self._col("vwap20")[-1])

    # --- trend / volatility / S&R (:184-249) ----------------------------
    def trend(self) -> str:
        close = float(self.candles[-1, 0])
        s20, s50 = self.sma(20), self.sma(50)
        if close > s20 > s50:
            return "uptrend"
        if close < s20 < s50:
            return "downtrend"
        return "neutral"

    def volatility(self, n: int = 100) -> float:
        """Annualized close-to-close vol over the last n 1m candles."""
        c = self.candles[-n:, 0]
        if len(c) < 3:
            return 0.0
        return float(np.diff(np.log(c)).std() * np.sqrt(525_600))

    def support_resistance(self, n: int = 200) -> dict:
        w = self.candles[-n:]
        close = float(w[-1, 0])
        lows = w[:, 2]
        highs = w[:, 1]
        support = float(np.percentile(lows, 10))
        resistance = float(np.percentile(highs, 90))
        return {"support": support, "resistance": resistance,
                "close": close}


class PositionSizer:
    """Volatility-tiered position sizing with volume factor and 2:1 TP/SL
    (:251-291)."""

    def __init__(self, base_pct: float = 0.1, max_pct: float = 0.5):
        self.base_pct = base_pct
        self.max_pct = max_pct

    def calculate_position_size(self, equity: float, volatility: float,
                                avg_volume_usd: float = 1e6,
                                stop_loss_pct: float = 0.02) -> dict:
        if volatility > 1.0:
            tier = 0.5
        elif volatility > 0.6:
            tier = 0.75
        else:
            tier = 1.0
        vol_factor = min(avg_volume_usd / 1e6, 1.5)
        pct = min(self.base_pct * tier * vol_factor, self.max_pct)
        return {
            "position_pct": pct,
            "position_usd": equity * pct,
            "stop_loss_pct": stop_loss_pct,
            "take_profit_pct": 2.0 * stop_loss_pct,   # 2:1 RR
        }


class TradingSignalVotes:
    """6-indicator BUY/SELL/NEUTRAL voting + 0-100 strength (:470-581)."""

    def __init__(self, analyzer: TechnicalAnalyzer):
        self.a = analyzer

    def votes(self) -> dict:
        a = self.a
        close = float(a.candles[-1, 0])
        macd, sig, hist = a.macd()
        bb = a.bollinger()
        v = {
            "rsi": "BUY" if a.rsi() < 30 else
                   ("SELL" if a.rsi() > 70 else "NEUTRAL"),
            "macd": "BUY" if hist > 0 else
                    ("SELL" if hist < 0 else "NEUTRAL"),
            "stochastic": "BUY" if a.stochastic() < 20 else
                          ("SELL" if a.stochastic() > 80 else "NEUTRAL"),
            "williams": "BUY" if a.williams_r() < -80 else
                        ("SELL" if a.williams_r() > -20 else "NEUTRAL"),
            "bollinger": "BUY" if bb["position"] < 0.05 else
                         ("SELL" if bb["position"] > 0.95 else "NEUTRAL"),
            "trend": "BUY" if a.trend() == "uptrend" else
                     ("SELL" if a.trend() == "downtrend" else "NEUTRAL"),
        }
        return v

    def signal(self) -> dict:
        v = self.votes()
        buys = sum(1 for x in v.values() if x == "BUY")
        sells = sum(1 for x in v.values() if x == "SELL")
        if buys > sells and buys >= 2:
            decision = "BUY"
        elif sells > buys and sells >= 2:
            decision = "SELL"
        else:
            decision = "NEUTRAL"
        strength = min(abs(buys - sells) / 6.0 * 100.0 +
                       max(buys, sells) * 8.0, 100.0)
        return {"decision": decision, "strength": strength, "votes": v}


class CryptoScanner:
    """Market scan over a symbol universe with a 0-100 opportunity score
    (:293-468). On GPU the whole universe's indicators come from ONE
    batched kernel launch (replacing ThreadPoolExecutor(max_workers=10))."""

    def __init__(self, quote: str = "USDC"):
        self.quote = quote

    def opportunity_score(self, analyzer: TechnicalAnalyzer,
                          avg_volume_usd: float = 1e6) -> float:
        score = 50.0
        rsi = analyzer.rsi()
        if rsi < 30:
            score += (30 - rsi)              # oversold bonus
        elif rsi > 70:
            score -= (rsi - 70)
        _, _, hist = analyzer.macd()
        score += np.clip(hist * 1e4, -10, 10)
        bb = analyzer.bollinger()
        if bb["position"] < 0.1:
            score += 10
        elif bb["position"] > 0.9:
            score -= 10
        if analyzer.trend() == "uptrend":
            score += 10
        elif analyzer.trend() == "downtrend":
            score -= 10
        score += min(avg_volume_usd / 1e6, 1.0) * 10
        vol = analyzer.volatility()
        if vol > 1.5:
            score -= 15                      # too hot
        return float(np.clip(score, 0.0, 100.0))

    def scan_market(self, market: dict[str, np.ndarray],
                    top_k: int = 10) -> list[dict]:
        """market: symbol -> (T, 4) candles. Batched scoring; returns the
        top-k opportunities sorted by score."""
        symbols = sorted(market)
        if not symbols:
            return []
        if gpu_available():
            scores = self._scan_gpu(symbols, market)
        else:
            scores = {}
            for s in symbols:
                a = TechnicalAnalyzer(market[s])
                vol_usd = float(market[s][-20:, 3].mean()
                                * market[s][-1, 0])
                scores[s] = (self.opportunity_score(a, vol_usd),
                             TradingSignalVotes(a).signal())
        out = [{"symbol": s, "score": sc, "signal": sig}
               for s, (sc, sig) in scores.items()]
        out.sort(key=lambda r: -r["score"])
        return out[:top_k]

    def _scan_gpu(self, symbols, market):
        """One indicators_gpu launch over the whole (padded) universe."""
        import torch

        from .ops.indicators import indicators_gpu

        T = min(len(market[s]) for s in symbols)
        batch = np.stack([market[s][-T:] for s in symbols]).astype(
            np.float32)
        ind = indicators_gpu(torch.from_numpy(batch).cuda())
        torch.cuda.synchronize()
        ind_np = ind.cpu().numpy()
        scores = {}
        for i, s in enumerate(symbols):
            a = TechnicalAnalyzer.__new__(TechnicalAnalyzer)
            a.candles = batch[i]
            a.ind = ind_np[i]
            vol_usd = float(batch[i][-20:, 3].mean() * batch[i][-1, 0])
            scores[s] = (self.opportunity_score(a, vol_usd),
                         TradingSignalVotes(a).signal())
        return scores
