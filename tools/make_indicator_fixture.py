#!/usr/bin/env python3
"""Generate the checked-in indicator golden fixture
(tests/fixtures/indicator_golden.csv).

Columns: a shared OHLCV series plus indicator columns computed with an
INDEPENDENT pandas implementation of the reference's `ta`-library
formulas (binance_ml_strategy.py:40-182, market_monitor_service.py:
219-298 — the exact conventions of ta.momentum.RSIIndicator /
ta.trend.MACD / ta.volatility.BollingerBands / AverageTrueRange /
StochasticOscillator / WilliamsRIndicator / VolumeWeightedAveragePrice):

  rsi      Wilder ewm(alpha=1/14, adjust=False) of clipped diffs,
           seeded by the ewm itself (ta's convention)
  macd     ewm(span=12/26, adjust=False) difference; signal span=9
  bb_*     rolling(20).mean +- 2 * rolling(20).std(ddof=0)
  atr      ewm(alpha=1/14, adjust=False) of the true range
  stoch_k  100*(close - ll14)/(hh14 - ll14)
  williams -100*(hh14 - close)/(hh14 - ll14)
  vwap     rolling-20 sum(tp*v)/sum(v), tp = (h+l+c)/3

The CPU twin (ops/indicators.indicators_cpu) and therefore the HIP
kernel are asserted against this file in
tests/test_indicator_golden.py. The `ta` package itself is not
installed in this image; these pandas expressions are transcribed from
its public formulas and serve as the independent second implementation
the round-1 verdict asked for.
"""

from __future__ import annotations

import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import numpy as np
import pandas as pd


def ta_indicators(df: pd.DataFrame) -> pd.DataFrame:
    c, h, lo, v = df["close"], df["high"], df["low"], df["volume"]
    out = pd.DataFrame(index=df.index)
    ema12 = c.ewm(span=12, adjust=False).mean()
    ema26 = c.ewm(span=26, adjust=False).mean()
    out["ema12"], out["ema26"] = ema12, ema26
    out["macd"] = ema12 - ema26
    out["macd_signal"] = out["macd"].ewm(span=9, adjust=False).mean()

    diff = c.diff(1)
    up = diff.clip(lower=0).fillna(0.0)
    dn = (-diff.clip(upper=0)).fillna(0.0)
    ag = up.ewm(alpha=1 / 14, adjust=False).mean()
    al = dn.ewm(alpha=1 / 14, adjust=False).mean()
    out["rsi"] = 100 - 100 / (1 + ag / al.clip(lower=1e-12))

    mid = c.rolling(20).mean()
    sd = c.rolling(20).std(ddof=0)
    out["bb_mid"] = mid
    out["bb_up"] = mid + 2 * sd
    out["bb_lo"] = mid - 2 * sd

    prev_c = c.shift(1)
    tr = pd.concat([h - lo, (h - prev_c).abs(), (lo - prev_c).abs()],
                   axis=1).max(axis=1)
    out["atr"] = tr.ewm(alpha=1 / 14, adjust=False).mean()

    hh = h.rolling(14).max()
    ll = lo.rolling(14).min()
    rng = (hh - ll).clip(lower=1e-12)
    out["stoch_k"] = 100 * (c - ll) / rng
    out["williams_r"] = -100 * (hh - c) / rng

    tp = (h + lo + c) / 3
    out["vwap"] = (tp * v).rolling(20).sum() / \
        v.rolling(20).sum().clip(lower=1e-12)
    return out


def main():
    from ai_crypto_trader_amd.data.synthetic import (
        candles_chl_v, generate_ohlcv,
    )

    candles = candles_chl_v(generate_ohlcv(3000, 1, seed=1234))[0]
    df = pd.DataFrame(candles, columns=["close", "high", "low", "volume"])
    ind = ta_indicators(df.astype(np.float64))
    full = pd.concat([df, ind], axis=1)
    out = Path(__file__).resolve().parent.parent / "tests" / "fixtures" \
        / "indicator_golden.csv"
    full.to_csv(out, index=False, float_format="%.10g")
    print(f"wrote {out} ({len(full)} rows)")


if __name__ == "__main__":
    main()
