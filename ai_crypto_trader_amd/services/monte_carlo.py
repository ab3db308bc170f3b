"""Monte-Carlo risk service (reference parity:
services/monte_carlo_service.py:41-943).

Portfolio GBM simulation per asset x scenario (:492-575) with the
base/bull/bear/volatile/crab scenario multipliers (:249-256), VaR/CVaR/
percentiles/prob-profit (:304-325) and per-path max drawdown (:327-336) —
all on the HIP Philox path kernel (ops/montecarlo.py) when a GPU is
present (10M+ correlated paths) and the numpy reference otherwise.
On-demand requests via the `monte_carlo_request` key (:776-823); results
to `monte_carlo_results` / `monte_carlo_latest_report` keys."""

from __future__ import annotations

import time

import numpy as np

from ..bus.schema import Channels, Keys
from ..ops import gpu_available
from ..ops.montecarlo import mc_paths_cpu, risk_stats
from .base import Service


class MonteCarloService(Service):
    name = "monte_carlo"

    def __init__(self, bus, config=None, price_history=None):
        super().__init__(bus, config)
        self.prices: dict[str, list[float]] = dict(price_history or {})
        self.last_report: dict = {}
        self.runs = 0

    def run_tasks(self):
        return [self._consume_market(), self._mc_loop(),
                self._request_loop()]

    async def _consume_market(self):
        sub = self.bus.subscribe(Channels.MARKET_UPDATES)

        def on_msg(_, m):
            if m.get("symbol"):
                h = self.prices.setdefault(m["symbol"], [])
                h.append(m["current_price"])
                if len(h) > 4096:
                    del h[:2048]

        await self.consume(sub, on_msg)

    def _estimate_params(self, syms):
        """Annualized mu/sigma from history (reference :238-247) +
        correlation for the correlated path generator."""
        n = min(len(self.prices[s]) for s in syms)
        px = np.stack([np.asarray(self.prices[s][-n:], np.float64)
                       for s in syms], axis=1)
        rets = np.diff(np.log(px), axis=0)
        per_year = 525_600.0
        # clamp the annualized drift: a 1m-candle sample mean annualizes
        # x525600, so short histories produce absurd mu estimates (the
        # reference has the same naive annualization bug at x252 from 30d;
        # monte_carlo_service.py:238-247 — we implement the intent)
        mu = np.clip(rets.mean(axis=0) * per_year, -2.0, 2.0)
        sigma = np.clip(rets.std(axis=0) * np.sqrt(per_year), 1e-4, 4.0)
        if len(syms) > 1:
            corr = np.corrcoef(rets.T)
            corr = np.nan_to_num(corr, nan=0.0)
            np.fill_diagonal(corr, 1.0)
            # PD repair
            w, v = np.linalg.eigh(corr)
            corr = v @ np.diag(np.maximum(w, 1e-6)) @ v.T
            d = np.sqrt(np.diag(corr))
            corr = corr / d[:, None] / d[None, :]
        else:
            corr = np.eye(1)
        return mu, sigma, np.linalg.cholesky(corr)

    def _log_return_rows(self, syms) -> np.ndarray | None:
        """(T-1, A) joint log-return rows for the historical bootstrap
        (None when histories are too short)."""
        n = min(len(self.prices[s]) for s in syms)
        if n < 32:
            return None
        px = np.stack([np.asarray(self.prices[s][-n:], np.float64)
                       for s in syms], axis=1)
        return np.diff(np.log(px), axis=0).astype(np.float32)

    def simulate(self, syms: list[str], scenario: str = "base",
                 n_paths: int | None = None) -> dict:
        """One portfolio simulation under a scenario (reference
        run_monte_carlo_simulation :197-336)."""
        mu, sigma, chol = self._estimate_params(syms)
        m_mu, m_sig = self.config.monte_carlo.scenarios.get(
            scenario, (1.0, 1.0))
        mu = mu * m_mu
        sigma = sigma * m_sig
        w = np.full(len(syms), 1.0 / len(syms))
        # pad to a multiple of 4 assets (path kernels consume float4 z's):
        # zero-weight dummies with unit diagonal correlation
        A = len(syms)
        pad = (-A) % 4
        if pad:
            mu = np.concatenate([mu, np.zeros(pad)])
            sigma = np.concatenate([sigma, np.full(pad, 1e-4)])
            w = np.concatenate([w, np.zeros(pad)])
            c2 = np.eye(A + pad)
            c2[:A, :A] = chol
            chol = c2
        days = self.config.monte_carlo.time_horizon_days
        dt = 1.0 / 365.0
        n_paths = n_paths or self.config.monte_carlo.num_simulations
        seed = (self.runs * 977 + 13) & 0xFFFFFFFF

        if self.config.monte_carlo.simulation_method == "historical":
            lr = self._log_return_rows(syms)
            if lr is not None:
                # scenario multipliers rescale the empirical rows around
                # their mean (shape-preserving; reference :275-298)
                lm = lr.mean(axis=0, keepdims=True)
                lr_s = lm * m_mu + (lr - lm) * m_sig
                if pad:
                    lr_s = np.concatenate(
                        [lr_s, np.zeros((len(lr_s), pad), np.float32)],
                        axis=1)
                if gpu_available() and lr_s.shape[1] in (4, 8, 16, 32, 64):
                    import torch
                    from ..ops.montecarlo import mc_bootstrap_gpu
                    fv, dd = mc_bootstrap_gpu(lr_s, w, n_steps=days,
                                              n_paths=n_paths, seed=seed)
                    torch.cuda.synchronize()
                    stats = risk_stats(fv, v0=1.0)
                    stats["max_drawdown_mean"] = float(dd.mean())
                    stats["max_drawdown_p95"] = float(
                        torch.quantile(dd, 0.95))
                else:
                    from ..ops.montecarlo import mc_bootstrap_cpu
                    fv, dd = mc_bootstrap_cpu(
                        lr_s, w, n_steps=days,
                        n_paths=min(n_paths, 20_000), seed=seed)
                    stats = risk_stats(fv, v0=1.0)
                    stats["max_drawdown_mean"] = float(dd.mean())
                    stats["max_drawdown_p95"] = float(
                        np.percentile(dd, 95))
                stats["method"] = "historical"
                self.runs += 1
                stats.update({
                    "scenario": scenario, "symbols": syms,
                    "n_paths": int(n_paths), "horizon_days": days,
                    "timestamp": time.time(),
                })
                return stats

        if gpu_available() and len(mu) in (4, 8, 16, 32, 64):
            import torch
            from ..ops.montecarlo import mc_paths_gpu
            t0 = time.perf_counter()
            # antithetic variates: unbiased with reduced estimator
            # variance at half the RNG cost (production risk runs; the
            # benchmark measures iid paths)
            fv, dd = mc_paths_gpu(chol, mu, sigma, w, antithetic=True,
                                  n_steps=days,
                                  n_paths=n_paths, dt=dt, seed=seed)
            torch.cuda.synchronize()
            el = time.perf_counter() - t0
            self.metrics.record_kernel_time("mc_paths", el)
            self.metrics.mc_paths_per_sec.set(n_paths / el)
            stats = risk_stats(fv, v0=1.0)
            stats["max_drawdown_mean"] = float(dd.mean())
            stats["max_drawdown_p95"] = float(
                torch.quantile(dd, 0.95))
        else:
            fv, dd = mc_paths_cpu(chol, mu, sigma, w, n_steps=days,
                                  n_paths=min(n_paths, 20_000), dt=dt,
                                  seed=seed)
            stats = risk_stats(fv, v0=1.0)
            stats["max_drawdown_mean"] = float(dd.mean())
            stats["max_drawdown_p95"] = float(np.percentile(dd, 95))
        self.runs += 1
        stats.update({
            "scenario": scenario, "symbols": syms, "n_paths": int(n_paths),
            "horizon_days": days, "timestamp": time.time(),
        })
        return stats

    def simulate_scenarios_gpu(self, syms: list[str]) -> dict:
        """All scenarios concurrently on separate HIP streams: kernels
        launch back-to-back with no host sync between scenarios, then one
        join before the (device-side) stats. Overlaps the path generation
        of the 5 scenarios across the GPU (SURVEY.md build-plan item:
        'overlap copies and collectives with compute on separate HIP
        streams')."""
        import torch

        from ..ops.montecarlo import (
            mc_bootstrap_gpu, mc_paths_gpu, risk_stats,
        )

        mu0, sigma0, chol = self._estimate_params(syms)
        w = np.full(len(syms), 1.0 / len(syms))
        days = self.config.monte_carlo.time_horizon_days
        n_paths = self.config.monte_carlo.num_simulations
        historical = (self.config.monte_carlo.simulation_method
                      == "historical")
        lr = self._log_return_rows(syms) if historical else None
        if lr is not None:
            lr_mean = lr.mean(axis=0, keepdims=True)
        from ..utils.metrics import GpuTimer

        streams = {}
        results = {}
        timer = GpuTimer(self.metrics, "mc_paths_scenarios").__enter__()
        for scen, (m_mu, m_sig) in self.config.monte_carlo.scenarios.items():
            st = torch.cuda.Stream()
            streams[scen] = st
            with torch.cuda.stream(st):
                if lr is not None:
                    # historical bootstrap (reference :275-298): scenario
                    # multipliers rescale the empirical return rows around
                    # their mean (shape-preserving)
                    lr_s = lr_mean * m_mu + (lr - lr_mean) * m_sig
                    fv, dd = mc_bootstrap_gpu(
                        lr_s, w, n_steps=days, n_paths=n_paths,
                        seed=(self.runs * 977 + 13) & 0xFFFFFFFF)
                else:
                    fv, dd = mc_paths_gpu(
                        chol, mu0 * m_mu,
                        np.maximum(sigma0 * m_sig, 1e-4), w,
                        n_steps=days, n_paths=n_paths, dt=1.0 / 365.0,
                        seed=(self.runs * 977 + 13) & 0xFFFFFFFF,
                        antithetic=True)
                results[scen] = (fv, dd)
            self.runs += 1
        for st in streams.values():
            st.synchronize()
        timer.__exit__()
        report = {}
        for scen, (fv, dd) in results.items():
            stats = risk_stats(fv, v0=1.0)
            stats["max_drawdown_mean"] = float(dd.mean())
            stats["max_drawdown_p95"] = float(torch.quantile(dd, 0.95))
            stats.update({"scenario": scen, "symbols": syms,
                          "n_paths": int(n_paths), "horizon_days": days,
                          "timestamp": time.time()})
            report[scen] = stats
        return report

    def fan_chart_data(self, syms: list[str], scenario: str = "base",
                       n_paths: int = 512) -> dict | None:
        """Per-step percentile bands of the portfolio value — the data
        behind the reference's matplotlib fan chart (:396-490), served as
        JSON for the dashboard. Small-n run on the numpy twin recording
        V_t per step."""
        if not all(len(self.prices.get(s, [])) >= 64 for s in syms):
            return None
        mu, sigma, chol = self._estimate_params(syms)
        m_mu, m_sig = self.config.monte_carlo.scenarios.get(
            scenario, (1.0, 1.0))
        mu = mu * m_mu
        sigma = np.maximum(sigma * m_sig, 1e-4)
        A = len(syms)
        pad = (-A) % 4
        w = np.full(A, 1.0 / A)
        if pad:
            mu = np.concatenate([mu, np.zeros(pad)])
            sigma = np.concatenate([sigma, np.full(pad, 1e-4)])
            w = np.concatenate([w, np.zeros(pad)])
            c2 = np.eye(A + pad)
            c2[:A, :A] = chol
            chol = c2
        days = self.config.monte_carlo.time_horizon_days
        from ..ops.montecarlo import _gbm_terms, philox_normal4_np
        cvol_k, drift = _gbm_terms(chol, mu, sigma, 1.0 / 365.0)
        Af = len(mu)
        logS = np.zeros((n_paths, Af), np.float32)
        paths = np.arange(n_paths, dtype=np.uint64)
        bands = {p: [1.0] for p in (5, 25, 50, 75, 95)}
        for step in range(days):
            for k4 in range(Af // 4):
                ctr = np.full(n_paths,
                              (np.uint64(step) << np.uint64(32))
                              | np.uint64(k4), np.uint64)
                z4 = philox_normal4_np(11, paths, ctr)
                for dz in range(4):
                    logS += np.outer(z4[:, dz], cvol_k[k4 * 4 + dz])
            logS += drift
            V = (w * np.exp2(logS)).sum(axis=1)
            for p in bands:
                bands[p].append(float(np.percentile(V, p)))
        return {"scenario": scenario, "symbols": syms,
                "horizon_days": days, "n_paths": n_paths,
                "percentiles": {str(p): v for p, v in bands.items()}}

    async def run_portfolio_mc(self) -> dict | None:
        syms = sorted(s for s, h in self.prices.items() if len(h) >= 64)
        if not syms:
            return None
        # pad/trim to a kernel-supported asset count
        for k in (64, 32, 16, 8, 4):
            if len(syms) >= k:
                syms = syms[:k]
                break
        if gpu_available() and len(syms) in (4, 8, 16, 32, 64):
            report = self.simulate_scenarios_gpu(syms)
        else:
            report = {}
            for scen in self.config.monte_carlo.scenarios:
                report[scen] = self.simulate(syms, scen)
        await self.bus.set(Keys.MONTE_CARLO_RESULTS, report)
        try:
            fan = self.fan_chart_data(syms)
            if fan:
                await self.bus.set(Keys.MC_FAN_CHART, fan)
        except Exception as e:
            self.log.debug("fan chart skipped: %r", e)
        await self.bus.set(Keys.MONTE_CARLO_LATEST_REPORT, {
            "generated_at": time.time(),
            "scenarios": list(report),
            "base_var_95": report.get("base", {}).get("var_95"),
        })
        self.last_report = report
        return report

    async def _mc_loop(self):
        while self.running:
            try:
                await self.run_portfolio_mc()
            except Exception as e:       # keep the service alive
                self.log.warning("mc failed: %s", e)
            await self.sleep(self.config.monte_carlo.interval_s)

    async def _request_loop(self):
        """On-demand per-symbol requests (reference :776-823)."""
        while self.running:
            req = await self.bus.get_json(Keys.MONTE_CARLO_REQUEST)
            if req and req.get("symbol") in self.prices and \
                    len(self.prices[req["symbol"]]) >= 64:
                try:
                    res = self.simulate([req["symbol"]],
                                        req.get("scenario", "base"),
                                        req.get("n_paths"))
                    await self.bus.set(
                        f"monte_carlo_result_{req['symbol']}", res)
                except Exception as e:
                    self.log.warning("mc request failed: %s", e)
                await self.bus.delete(Keys.MONTE_CARLO_REQUEST)
            await self.sleep(1.0)

    async def run(self):
        pass
