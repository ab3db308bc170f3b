"""Composite indicator combinations (reference parity:
services/utils/indicator_combinations.py — 15 composites computed from a
single market-update dict). Pure dict math on the MarketUpdate payload."""

from __future__ import annotations


def _sign(x: float) -> int:
    return 1 if x > 0 else (-1 if x < 0 else 0)


def calculate_indicator_combinations(u: dict) -> dict:
    """u: a MarketUpdate-shaped dict (bus/schema.py). Returns the
    combined-indicator dict published inside `market_updates`
    (market_monitor_service.py:476-485)."""
    rsi = u.get("rsi", 50.0)
    rsi3 = u.get("rsi_3m", rsi)
    rsi5 = u.get("rsi_5m", rsi)
    macd = u.get("macd", 0.0)
    macd3 = u.get("macd_3m", macd)
    macd5 = u.get("macd_5m", macd)
    stoch = u.get("stoch_k", 50.0)
    willr = u.get("williams_r", -50.0)
    bb = u.get("bb_position", 0.5)
    trend = u.get("trend", "neutral")
    tstr = u.get("trend_strength", 0.0)
    p1 = u.get("price_change_1m", 0.0)
    p5 = u.get("price_change_5m", 0.0)
    p15 = u.get("price_change_15m", 0.0)
    vol = u.get("avg_volume", 0.0)

    out: dict = {}

    # trend_confirmation (:96): MACD sign agreement across timeframes
    agree = _sign(macd) + _sign(macd3) + _sign(macd5)
    out["trend_confirmation"] = {
        "value": agree / 3.0,
        "signal": "bullish" if agree >= 2 else
                  ("bearish" if agree <= -2 else "neutral"),
    }

    # momentum_trend_alignment (:113)
    mom = _sign(p1) + _sign(p5) + _sign(p15)
    t = 1 if trend == "uptrend" else (-1 if trend == "downtrend" else 0)
    out["momentum_trend_alignment"] = {
        "value": (mom / 3.0 + t) / 2.0,
        "aligned": _sign(mom) == t and t != 0,
    }

    # triple timeframe price momentum (:144)
    out["triple_ma"] = {
        "value": (p1 + p5 + p15) / 3.0,
        "signal": "bullish" if p1 > 0 and p5 > 0 and p15 > 0 else
                  ("bearish" if p1 < 0 and p5 < 0 and p15 < 0 else "mixed"),
    }

    # volatility_adjusted_momentum (:186)
    bw = max(abs(bb - 0.5) * 2.0, 1e-6)
    out["volatility_adjusted_momentum"] = {"value": p5 / bw}

    # oscillator_consensus (:244): RSI/stoch/williams voting
    votes = ((rsi < 30) + (stoch < 20) + (willr < -80)) - \
            ((rsi > 70) + (stoch > 80) + (willr > -20))
    out["oscillator_consensus"] = {
        "value": votes / 3.0,
        "signal": "oversold" if votes >= 2 else
                  ("overbought" if votes <= -2 else "neutral"),
    }

    # stochastic_rsi combination (:313)
    out["stoch_rsi"] = {
        "value": (rsi / 100.0 + stoch / 100.0) / 2.0,
        "oversold": rsi < 35 and stoch < 25,
        "overbought": rsi > 65 and stoch > 75,
    }

    # double_rsi (:332): fast vs slow timeframe RSI cross
    out["double_rsi"] = {
        "fast": rsi, "slow": rsi5,
        "signal": "bullish" if rsi > rsi5 else
                  ("bearish" if rsi < rsi5 else "neutral"),
    }

    # volume_weighted_momentum (:370)
    out["volume_weighted_momentum"] = {"value": p5 * (1.0 + min(vol, 1e9) /
                                                     max(vol + 1.0, 1.0))}

    # trend_strength_index (:441)
    out["trend_strength_index"] = {
        "value": tstr * (1 if t >= 0 else -1),
        "strong": tstr > 0.6,
    }

    # market_regime_indicator (:505)
    vol_regime = "high" if bw > 0.6 else ("low" if bw < 0.2 else "normal")
    out["market_regime"] = {
        "trend": trend, "volatility": vol_regime,
        "regime": f"{trend}/{vol_regime}",
    }

    # reversal_probability (:540)
    rev = 0.0
    if rsi > 70 and bb > 0.95:
        rev = min(1.0, (rsi - 70) / 30 + (bb - 0.95) * 10)
    elif rsi < 30 and bb < 0.05:
        rev = -min(1.0, (30 - rsi) / 30 + (0.05 - bb) * 10)
    out["reversal_probability"] = {"value": rev}

    # breakout_confirmation (:595)
    out["breakout_confirmation"] = {
        "breakout_up": bb > 1.0 and p1 > 0.2,
        "breakout_down": bb < 0.0 and p1 < -0.2,
    }

    # divergence detector (:636): price momentum vs oscillator direction
    out["divergence"] = {
        "bearish": p5 > 0.5 and rsi < rsi5,
        "bullish": p5 < -0.5 and rsi > rsi5,
    }

    # volatility_trend_score (:222): BB-band extremity blended with trend
    # strength — high when a strong trend rides the band edge
    tstr01 = min(abs(tstr) / 100.0, 1.0)
    out["volatility_trend_score"] = round(
        0.7 * abs(bb - 0.5) * 2.0 + 0.3 * tstr01, 4)

    # volume_price_confirmation (:401): does volume confirm the 1m move
    vol_now = u.get("volume", vol)
    vratio = vol_now / vol if vol > 0 else 1.0
    if abs(p1) < 0.1:
        conf, strength = "neutral", 0.0
    elif vratio > 1.2:
        conf = "strong_bullish" if p1 > 0 else "strong_bearish"
        strength = min(1.0, vratio - 1.0)
    else:
        conf = "weak_bullish" if p1 > 0 else "weak_bearish"
        strength = max(0.0, min(0.5, (vratio - 0.8) / 0.4))
    out["volume_price_confirmation"] = {
        "confirmation": conf, "strength": round(strength, 4),
    }

    return out
