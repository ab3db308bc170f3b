"""Historical data management (reference parity:
backtesting/data_manager.py:18-415).

CSV store layout `{data_dir}/{market,social}/{symbol}/` (:26-33), cached
loads (:214-317), social resampling + as-of merge (:373-415). Fetching
is offline-first: `fetch` generates deterministic seeded-GBM history (no
network in this environment); a live fetcher can be registered behind the
same interface."""

from __future__ import annotations

import hashlib
from pathlib import Path

import numpy as np
import pandas as pd

from ..data.synthetic import generate_ohlcv

MARKET_COLS = ["timestamp", "open", "high", "low", "close", "volume"]


class HistoricalDataManager:
    def __init__(self, data_dir: str = "backtesting_data"):
        self.root = Path(data_dir)
        self.market_dir = self.root / "market"
        self.social_dir = self.root / "social"
        self.market_dir.mkdir(parents=True, exist_ok=True)
        self.social_dir.mkdir(parents=True, exist_ok=True)
        self._cache: dict[tuple, pd.DataFrame] = {}

    # --- paths -----------------------------------------------------------
    def market_path(self, symbol: str, interval: str) -> Path:
        d = self.market_dir / symbol
        d.mkdir(parents=True, exist_ok=True)
        return d / f"{symbol}_{interval}.csv"

    def social_path(self, symbol: str) -> Path:
        d = self.social_dir / symbol
        d.mkdir(parents=True, exist_ok=True)
        return d / f"{symbol}_social.csv"

    # --- fetch: seeded synthetic (default, no egress here) or the live
    # paginated Binance klines client (reference
    # backtesting/data_manager.py:47-114) ---------------------------------
    def fetch_market_data(self, symbol: str, interval: str = "1m",
                          n_candles: int = 10_000,
                          source: str = "synthetic",
                          transport=None,
                          start_ms: int | None = None,
                          end_ms: int | None = None) -> pd.DataFrame:
        if source == "binance":
            from ..live.binance import fetch_klines

            df = fetch_klines(symbol, interval, start_ms=start_ms,
                              end_ms=end_ms, limit=n_candles,
                              transport=transport)
            # open column is not part of the store schema's value set
            # used downstream; keep the venue's columns as returned
            df = df[MARKET_COLS]
            df.to_csv(self.market_path(symbol, interval), index=False)
            return df
        seed = int(hashlib.sha1(
            f"{symbol}:{interval}".encode()).hexdigest()[:8], 16)
        ohlcv = generate_ohlcv(n_candles, 1, seed=seed)[0]
        step_s = {"1m": 60, "3m": 180, "5m": 300, "15m": 900,
                  "1h": 3600, "1d": 86400}.get(interval, 60)
        df = pd.DataFrame(ohlcv, columns=MARKET_COLS[1:])
        df.insert(0, "timestamp",
                  np.arange(n_candles, dtype=np.int64) * step_s * 1000)
        df.to_csv(self.market_path(symbol, interval), index=False)
        return df

    def fetch_social_data(self, symbol: str, n_days: int = 30,
                          seed: int | None = None) -> pd.DataFrame:
        if seed is None:
            seed = int(hashlib.sha1(
                f"social:{symbol}".encode()).hexdigest()[:8], 16)
        rng = np.random.default_rng(seed)
        # slowly-varying daily sentiment + volume (LunarCrush-shaped rows)
        sent = 0.5 + 0.2 * np.cumsum(
            rng.standard_normal(n_days)) / np.sqrt(n_days)
        df = pd.DataFrame({
            "timestamp": np.arange(n_days, dtype=np.int64) * 86_400_000,
            "sentiment": np.clip(sent, 0.0, 1.0),
            "social_volume": rng.integers(100, 10_000, n_days),
            "engagement": rng.integers(1_000, 100_000, n_days),
            "contributors": rng.integers(10, 1_000, n_days),
        })
        df.to_csv(self.social_path(symbol), index=False)
        return df

    # --- load ------------------------------------------------------------
    def load_market_data(self, symbol: str, interval: str = "1m",
                         fetch_if_missing: bool = True,
                         n_candles: int = 10_000) -> pd.DataFrame | None:
        key = ("m", symbol, interval)
        if key in self._cache:
            return self._cache[key]
        p = self.market_path(symbol, interval)
        if not p.exists():
            if not fetch_if_missing:
                return None
            df = self.fetch_market_data(symbol, interval, n_candles)
        else:
            df = pd.read_csv(p)
        self._cache[key] = df
        return df

    def load_social_data(self, symbol: str,
                         fetch_if_missing: bool = True) -> pd.DataFrame | None:
        key = ("s", symbol)
        if key in self._cache:
            return self._cache[key]
        p = self.social_path(symbol)
        if not p.exists():
            if not fetch_if_missing:
                return None
            df = self.fetch_social_data(symbol)
        else:
            df = pd.read_csv(p)
        self._cache[key] = df
        return df

    def list_available(self) -> dict:
        out = {"market": [], "social": []}
        for p in self.market_dir.glob("*/*.csv"):
            out["market"].append(p.stem)
        for p in self.social_dir.glob("*/*.csv"):
            out["social"].append(p.stem)
        return out

    # --- merge (reference :373-415) --------------------------------------
    @staticmethod
    def merge_market_and_social_data(market: pd.DataFrame,
                                     social: pd.DataFrame) -> pd.DataFrame:
        """Resample daily social rows onto the bar frequency with an as-of
        (backward) join; bars before the first social row get the neutral
        defaults (social_data_provider.py:27-231)."""
        m = market.sort_values("timestamp")
        s = social.sort_values("timestamp")
        merged = pd.merge_asof(m, s, on="timestamp", direction="backward")
        merged["sentiment"] = merged["sentiment"].fillna(0.5)
        for c in ("social_volume", "engagement", "contributors"):
            if c in merged:
                merged[c] = merged[c].fillna(0)
        return merged

    def to_chlv(self, df: pd.DataFrame) -> np.ndarray:
        """DataFrame -> (1, T, 4) [close, high, low, volume] f32 for the
        engines (prices normalized to close[0] like data/synthetic.py)."""
        arr = np.stack([
            df["close"].to_numpy(np.float32),
            df["high"].to_numpy(np.float32),
            df["low"].to_numpy(np.float32),
            df["volume"].to_numpy(np.float32),
        ], axis=-1)[None]
        c0 = arr[0, 0, 0] or 1.0
        arr[..., :3] /= c0
        return np.ascontiguousarray(arr)


class SocialDataProvider:
    """Point-in-time social lookup with neutral defaults
    (backtesting/social_data_provider.py:27-231)."""

    DEFAULTS = {"sentiment": 0.5, "social_volume": 0.0,
                "engagement": 0.0, "contributors": 0.0}

    def __init__(self, dm: HistoricalDataManager):
        self.dm = dm

    def at(self, symbol: str, timestamp_ms: int) -> dict:
        df = self.dm.load_social_data(symbol, fetch_if_missing=False)
        if df is None or df.empty:
            return dict(self.DEFAULTS)
        past = df[df["timestamp"] <= timestamp_ms]
        if past.empty:
            return dict(self.DEFAULTS)
        row = past.iloc[-1]
        return {k: float(row.get(k, v)) for k, v in self.DEFAULTS.items()}
