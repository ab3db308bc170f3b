"""Portfolio risk service (reference parity:
services/portfolio_risk_service.py:36-948).

Per-asset historical VaR/CVaR (:217-284), covariance/correlation via the
MFMA GEMM kernel on GPU (ops/covar.py, replacing `returns_df.corr()`
:286-326), portfolio VaR quadratic form + PD check (:328-398), position
sizing equal-risk / half-Kelly / fixed (:400-487), adaptive stop-loss
scaled by annualized vol (:489-546), signal enrichment to
`risk_enriched_signals` (:796-856), `stop_loss_adjustments` and
`risk_alerts` channels, `portfolio_risk` / `adaptive_stop_losses` /
`portfolio_diversification` keys."""

from __future__ import annotations

import time

import numpy as np

from ..bus.schema import Channels, Keys, RiskAlert, RiskInfo
from ..ops import gpu_available
from ..ops.covar import (
    corr_from_cov, cov_cpu, historical_var_cvar, is_positive_definite,
    portfolio_var,
)
from .base import Service

Z95 = 1.645


class PortfolioRiskService(Service):
    name = "portfolio_risk"

    def __init__(self, bus, config=None, price_history=None):
        super().__init__(bus, config)
        # symbol -> list of closes (live-updated from market_updates)
        self.prices: dict[str, list[float]] = {}
        if price_history:
            self.prices.update(price_history)
        self.lookback = 512
        self.last_risk: dict = {}

    def run_tasks(self):
        return [self._consume_market(), self._enrich_signals(),
                self._risk_loop()]

    async def _consume_market(self):
        sub = self.bus.subscribe(Channels.MARKET_UPDATES)

        def on_msg(_, m):
            sym = m.get("symbol")
            if not sym:
                return
            h = self.prices.setdefault(sym, [])
            h.append(m["current_price"])
            if len(h) > self.lookback:
                del h[: len(h) - self.lookback]

        await self.consume(sub, on_msg)

    def _returns_matrix(self) -> tuple[list[str], np.ndarray] | None:
        syms = [s for s, h in self.prices.items() if len(h) >= 32]
        if len(syms) < 2:
            return None
        n = min(len(self.prices[s]) for s in syms)
        px = np.stack([np.asarray(self.prices[s][-n:], np.float64)
                       for s in syms], axis=1)
        rets = np.diff(np.log(px), axis=0).astype(np.float32)
        return syms, rets

    def compute_cov(self, rets: np.ndarray) -> np.ndarray:
        if gpu_available() and rets.shape[1] % 16 == 0 and \
                rets.shape[1] <= 64:
            import torch

            from ..ops.covar import cov_gpu
            from ..utils.metrics import GpuTimer
            with GpuTimer(self.metrics, "cov"):
                cov = cov_gpu(torch.from_numpy(rets).cuda()).cpu().numpy()
            return cov
        return cov_cpu(rets)

    async def compute_portfolio_risk(self) -> dict | None:
        out = self._returns_matrix()
        if out is None:
            return None
        syms, rets = out
        holdings = await self.bus.get_json(Keys.HOLDINGS) or {}
        total = holdings.get("total_value", 1.0) or 1.0
        values = []
        for s in syms:
            base = s[:-len(self.config.trading.quote_asset)]
            values.append(
                holdings.get("holdings", {}).get(base, {}).get("value", 0.0))
        values = np.asarray(values)
        vols = rets.std(axis=0).astype(np.float64)

        cov = self.compute_cov(rets)
        corr = corr_from_cov(cov)
        pd_ok = is_positive_definite(corr + 1e-9 * np.eye(len(syms)))
        pvar = portfolio_var(values, vols, corr, confidence_z=Z95)

        per_asset = {}
        for i, s in enumerate(syms):
            var, cvar = historical_var_cvar(rets[:, i], max(values[i], 0.0)
                                            or total * 0.1)
            per_asset[s] = {
                "var": var, "cvar": cvar, "vol": float(vols[i]),
                "value": float(values[i]),
            }

        # diversification check (:774): average |rho| off-diagonal
        n = len(syms)
        off = np.abs(corr[np.triu_indices(n, 1)])
        avg_corr = float(off.mean()) if off.size else 0.0
        high_corr = [
            syms[j]
            for i in range(n) for j in range(i + 1, n)
            if corr[i, j] > self.config.risk.correlation_threshold
            and values[i] > 0
        ]

        risk = {
            "timestamp": time.time(),
            "symbols": syms,
            "portfolio_var": pvar,
            "portfolio_var_pct": pvar / total * 100.0,
            "positive_definite": bool(pd_ok),
            "avg_correlation": avg_corr,
            "high_correlation_symbols": sorted(set(high_corr)),
            "correlation_matrix": np.round(corr, 4).tolist(),
            "per_asset": per_asset,
        }
        self.last_risk = risk
        return risk

    def position_size(self, sym: str, risk: dict) -> float:
        """equal_risk (1/VaR) / half-Kelly / fixed (:400-487)."""
        mode = self.config.risk.position_sizing
        pa = risk.get("per_asset", {}).get(sym)
        if mode == "fixed" or pa is None:
            return self.config.risk.fixed_position_pct
        if mode == "half_kelly":
            vol = max(pa["vol"], 1e-6)
            edge = 0.02       # conservative assumed edge
            return float(np.clip(0.5 * edge / (vol * vol) / 100, 0.01, 0.5))
        inv = {s: 1.0 / max(a["vol"], 1e-6)
               for s, a in risk["per_asset"].items()}
        return float(np.clip(inv[sym] / sum(inv.values()), 0.01, 0.5))

    def adaptive_stop(self, sym: str, risk: dict) -> float:
        """stop pct scaled by annualized vol factor in [0.5, 2.0] (:489)."""
        pa = risk.get("per_asset", {}).get(sym)
        base = self.config.risk.base_stop_loss_pct
        if pa is None:
            return base
        ann_vol = pa["vol"] * np.sqrt(525_600.0)
        factor = np.clip(ann_vol / 0.6,
                         self.config.risk.adaptive_stop_vol_factor_min,
                         self.config.risk.adaptive_stop_vol_factor_max)
        return float(base * factor)

    async def _enrich_signals(self):
        """Signal enrichment pipeline (reference :796-856)."""
        sub = self.bus.subscribe(Channels.TRADING_SIGNALS)

        async def on_sig(_, sig):
            if not (isinstance(sig, dict) and "decision" in sig):
                return
            risk = self.last_risk or {}
            sym = sig["symbol"]
            pa = risk.get("per_asset", {}).get(sym, {})
            stop_pct = self.adaptive_stop(sym, risk)
            info = RiskInfo(
                var=pa.get("var", 0.0),
                var_pct=pa.get("var", 0.0) /
                max(pa.get("value", 1.0), 1.0) * 100,
                cvar=pa.get("cvar", 0.0),
                portfolio_var=risk.get("portfolio_var", 0.0),
                optimal_position_pct=self.position_size(sym, risk),
                adaptive_stop_loss=sig.get("market_data", {}).get(
                    "current_price", 0.0) * (1 - stop_pct),
                adaptive_stop_pct=stop_pct,
            )
            enriched = dict(sig)
            enriched["risk_info"] = info.to_dict()
            await self.bus.publish(Channels.RISK_ENRICHED_SIGNALS, enriched)

        await self.consume(sub, on_sig)

    async def _risk_loop(self):
        while self.running:
            await self.sleep(5.0)
            try:
                risk = await self.compute_portfolio_risk()
                if risk is None:
                    continue
                await self.bus.set(Keys.PORTFOLIO_RISK, risk)
                self.metrics.var.set(risk["portfolio_var"])
                stops = {
                    s: {"adaptive_stop_pct": self.adaptive_stop(s, risk)}
                    for s in risk["symbols"]
                }
                await self.bus.set(Keys.ADAPTIVE_STOP_LOSSES, stops)
                await self.bus.set(Keys.PORTFOLIO_DIVERSIFICATION, {
                    "avg_correlation": risk["avg_correlation"],
                    "high_correlation_symbols":
                        risk["high_correlation_symbols"],
                })
                if risk["portfolio_var_pct"] > \
                        self.config.risk.max_portfolio_var_pct * 100:
                    await self.bus.publish(
                        Channels.RISK_ALERTS,
                        RiskAlert("portfolio_var_exceeded",
                                  {"var_pct": risk["portfolio_var_pct"]}
                                  ).to_dict())
                if risk["avg_correlation"] > 0.8:
                    await self.bus.publish(
                        Channels.RISK_ALERTS,
                        RiskAlert("poor_diversification",
                                  {"avg_correlation":
                                   risk["avg_correlation"]}).to_dict())
            except Exception as e:
                self.log.warning("risk computation failed: %r", e)

    async def run(self):
        pass
