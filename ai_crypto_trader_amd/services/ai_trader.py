"""LLM trading-analyst adapter (reference parity: services/ai_trader.py
:9-418).

The reference calls OpenAI chat.completions in JSON mode for every
analysis (:36-189). This environment has no egress, so the DEFAULT
analyst is the deterministic LocalAnalyst (services/analyzer.py) behind
the same interface; when an OpenAI-compatible endpoint + key exist, the
LLM path activates with the reference's prompt/response contract
(decision/confidence/reasoning/risk_level/key_indicators/explanation/
factor_weights + model-version UUID tracking :24-34)."""

from __future__ import annotations

import json
import os
import uuid

from .analyzer import LocalAnalyst


class AITrader:
    def __init__(self, api_key: str | None = None,
                 base_url: str | None = None, model: str = "gpt-4o-mini",
                 min_confidence: float = 0.7):
        self.api_key = api_key or os.environ.get("OPENAI_API_KEY")
        self.base_url = base_url or os.environ.get("OPENAI_BASE_URL")
        self.model = model
        self.min_confidence = min_confidence
        self.model_version = f"{model}-{uuid.uuid4().hex[:8]}"
        self.local = LocalAnalyst()
        self._client = None
        if self.api_key:
            try:
                import openai

                self._client = openai.OpenAI(
                    api_key=self.api_key, base_url=self.base_url)
            except Exception:
                self._client = None

    @property
    def llm_available(self) -> bool:
        return self._client is not None

    # --- prompts (reference :36-189 contract) ---------------------------
    @staticmethod
    def _analysis_prompt(market: dict, social: dict | None,
                         news: dict | None) -> str:
        return (
            "You are a crypto trading analyst. Given the market data "
            "below, respond ONLY with JSON: {\"decision\": BUY|SELL|HOLD, "
            "\"confidence\": 0-1, \"reasoning\": str, \"risk_level\": "
            "low|medium|high, \"key_indicators\": [str], \"explanation\": "
            "{factor: score}, \"factor_weights\": {factor: weight}}.\n"
            f"market: {json.dumps(market)}\n"
            f"social: {json.dumps(social or {})}\n"
            f"news: {json.dumps(news or {})}"
        )

    def analyze_trade_opportunity(self, market: dict,
                                  social: dict | None = None,
                                  news: dict | None = None,
                                  nn_pred: dict | None = None) -> dict:
        if self._client is None:
            out = self.local.analyze(market, social, news, nn_pred)
            out["model_version"] = f"local-{self.model_version}"
            return out
        resp = self._client.chat.completions.create(
            model=self.model,
            response_format={"type": "json_object"},
            messages=[{"role": "user",
                       "content": self._analysis_prompt(market, social,
                                                        news)}],
        )
        out = json.loads(resp.choices[0].message.content)
        out.setdefault("model_version", self.model_version)
        out.setdefault("model_id", self.model)
        return out

    def analyze_risk_setup(self, position: dict, market: dict) -> dict:
        """(:191-234) SL/TP recommendation."""
        vol = market.get("volatility", 0.5)
        base = 0.02 * max(min(vol / 0.6, 2.0), 0.5)
        return {
            "stop_loss_pct": base,
            "take_profit_pct": 2 * base,
            "risk_level": "high" if vol > 1.0 else
                          ("low" if vol < 0.3 else "medium"),
        }

    def analyze_market_conditions(self, updates: list[dict]) -> dict:
        """(:236-342) market-wide narrative."""
        if not updates:
            return {"condition": "unknown", "bias": 0.0}
        chg = [u.get("price_change_15m", 0.0) for u in updates]
        bias = sum(chg) / len(chg)
        return {
            "condition": "risk_on" if bias > 0.2 else
                         ("risk_off" if bias < -0.2 else "mixed"),
            "bias": bias,
            "breadth": sum(1 for c in chg if c > 0) / len(chg),
        }

    # --- gates (reference :368-418) --------------------------------------
    def should_take_trade(self, analysis: dict) -> bool:
        return (analysis.get("decision") == "BUY"
                and analysis.get("confidence", 0.0) >= self.min_confidence)

    def adjust_position_size(self, ai_size_pct: float,
                             technical_size_pct: float) -> float:
        """Average AI/technical sizes, conservative (:389-418)."""
        return min(ai_size_pct, technical_size_pct) * 0.5 + \
            (ai_size_pct + technical_size_pct) / 4.0
