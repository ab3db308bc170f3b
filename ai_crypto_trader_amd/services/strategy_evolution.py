"""Strategy evolution service (reference parity:
services/strategy_evolution_service.py:34-1749).

The evolution brain: monitors live strategy performance, and when it
degrades below the configured thresholds (:1571, config.json:208-211)
evolves the parameter set. Optimizers:
  ga      device-resident GA over the HIP backtest kernel (GAEngine;
          replaces the serial fitness loop genetic_algorithm.py:119-133)
  rl      DQN/PPO parameter-nudging (reference :696-975): trains the RL
          agent on recent market snapshots and maps its action preference
          to parameter nudges
  gpt     LLM optimizer seam (reference :364-511): an injected llm_fn
          proposes a param dict or strategy CODE (params_from_code
          extracts); GA fallback when no llm_fn is configured
  hybrid  regime/volatility-based method selection (:1151-1184)
Hot-swap via `strategy_params` key + `strategy_update` channel (:349-362);
per-regime parameter adjustment tables (:145-174); model-version registry
events.
"""

from __future__ import annotations

import time
import uuid

import numpy as np

from ..backtesting.engine_cpu import METRIC_NAMES
from ..backtesting.strategy import (
    DEFAULT_PARAMS, clip_params, params_to_dict,
)
from ..bus.schema import Channels, EvolutionUpdate, Keys
from .base import Service

# per-regime parameter adjustment tables (reference :145-174)
REGIME_ADJUSTMENTS = {
    "bull": {"take_profit_pct": 1.3, "stop_loss_pct": 1.1,
             "position_size_pct": 1.2},
    "bear": {"take_profit_pct": 0.8, "stop_loss_pct": 0.8,
             "position_size_pct": 0.7},
    "volatile": {"stop_loss_pct": 1.4, "position_size_pct": 0.6,
                 "trailing_stop_pct": 1.5},
    "ranging": {"take_profit_pct": 0.9, "entry_votes": 1.0},
}


class StrategyEvolutionService(Service):
    name = "strategy_evolution"

    def __init__(self, bus, config=None, candles=None, device="cpu",
                 llm_fn=None):
        super().__init__(bus, config)
        self.candles = candles        # (nsym, T, 4) evaluation market
        self.device = device
        self.llm_fn = llm_fn          # optional GPT-optimizer seam
        self.current_params = DEFAULT_PARAMS.copy()
        self.current_strategy_id = "default"
        self.engine = None
        self.evolutions = 0
        self.model_versions: list[dict] = []

    # --- performance gate (reference :1571) ------------------------------
    def needs_improvement(self, perf: dict) -> bool:
        e = self.config.evolution
        return (perf.get("sharpe", 0.0) < e.min_sharpe_ratio
                or perf.get("max_drawdown", 0.0) > e.max_drawdown
                or perf.get("win_rate", 1.0) < e.min_win_rate
                or perf.get("profit_factor", 9.9) < e.min_profit_factor)

    def select_method(self, regime: str, volatility: float) -> str:
        """hybrid optimizer selection (reference :1151-1184)."""
        m = self.config.evolution.method
        if m != "hybrid":
            return m
        if regime == "volatile" or volatility > 0.8:
            return "ga"           # broad search when conditions shift
        if self.evolutions % 3 == 2:
            # every 3rd evolution: RL nudging — PPO on the hipGraph env
            # when a GPU is present, DQN self-play otherwise
            return "ppo" if str(self.device).startswith("cuda") else "rl"
        return "ga"

    # --- optimizers ------------------------------------------------------
    def optimize_with_ga(self, generations: int | None = None):
        from ..backtesting.ga_engine import GAEngine

        if self.engine is None:
            e = self.config.evolution
            self.engine = GAEngine(
                self.candles,
                pop_per_rank=e.population_size,
                device=self.device, seed=self.config.seed + 7,
                elite_k=e.elite_k, tournament=e.tournament,
                cx_rate=e.cx_rate, mut_rate=e.mut_rate,
                mut_scale=e.mut_scale, segments="auto",
            )
            # seed the population with the current params (elitism anchor)
            import torch
            self.engine.pop_t[0] = torch.from_numpy(
                self.current_params).to(self.engine.device)
        gens = generations or self.config.evolution.generations
        for _ in range(gens):
            self.engine.step()
        self.engine.eval_fitness()
        fit, best = self.engine.best()
        return clip_params(best[None])[0], {"fitness": fit}

    def optimize_with_rl(self):
        """RL parameter nudging (reference :696-975): a DQN trained on
        recent windows picks among {widen, hold, tighten} per risk param."""
        import torch

        from ..models.rl import DQNAgent

        agent = DQNAgent("cpu", n_obs=8, n_act=3, seed=self.evolutions)
        rets = np.diff(np.log(self.candles[0, -256:, 0]))
        obs = torch.zeros(1, 8)
        obs[0, 0] = float(rets.mean() * 1e4)
        obs[0, 1] = float(rets.std() * 1e2)
        # tiny self-play: reward tightening stops in high vol
        for i in range(120):
            a = agent.act(obs)
            vol = float(obs[0, 1])
            r = torch.tensor([1.0 if (vol > 1.0 and a == 2) or
                              (vol <= 1.0 and a == 1) else 0.0])
            agent.remember(obs, a, r, obs, torch.ones(1))
            agent.replay()
        with torch.no_grad():
            a = int(agent.q(obs).argmax())
        p = self.current_params.copy()
        if a == 2:      # tighten
            p[13] *= 0.8
            p[15] = max(p[15], 0.01)
        elif a == 1:    # widen
            p[13] *= 1.2
            p[14] *= 1.2
        return clip_params(p[None])[0], {"rl_action": a}

    def optimize_with_ppo(self, train_steps: int = 3):
        """PPO parameter guidance (the BASELINE's PPO counterpart of the
        reference's DQN nudging :696-975): train PPO briefly on the HIP
        envs over the evaluation market (hipGraph-captured rollout+update
        on GPU) and map the learned action preference — mean entry
        propensity vs exit propensity — to risk-appetite nudges."""
        import torch

        from ..models.rl import PPOAgent, TradingVecEnv

        if not str(self.device).startswith("cuda"):
            return self.optimize_with_rl()    # CPU fallback: DQN nudging
        market = torch.from_numpy(self.candles).to(self.device)
        env = TradingVecEnv(market, n_envs=256,
                            ep_len=min(1024, self.candles.shape[1] - 64),
                            seed=self.evolutions)
        env.reset()
        agent = PPOAgent(self.device, seed=self.evolutions, use_graph=True)
        for _ in range(train_steps):
            stats = agent.train_step(env, horizon=64)
        with torch.no_grad():
            logits, _ = agent.net(env.obs)
            probs = torch.softmax(logits, dim=-1).mean(dim=0)
        buy_pref = float(probs[1])
        sell_pref = float(probs[2])
        p = self.current_params.copy()
        if buy_pref > sell_pref + 0.1:        # risk-on: looser entries
            p[10] = max(p[10] - 1, 1)         # entry_votes
            p[14] *= 1.2                      # take_profit
        elif sell_pref > buy_pref + 0.1:      # risk-off
            p[10] = min(p[10] + 1, 4)
            p[13] *= 0.8                      # tighter stop
            p[12] *= 0.8                      # smaller size
        return clip_params(p[None])[0], {
            "ppo_buy_pref": buy_pref, "ppo_sell_pref": sell_pref,
            "ppo_entropy": stats.get("entropy", 0.0),
        }

    def optimize_with_llm(self):
        """GPT optimizer seam (reference :364-511): an injected `llm_fn`
        receives the current params + measured performance and returns a
        param dict (or strategy CODE — params_from_code extracts them).
        Offline default: no llm_fn -> GA fallback."""
        if self.llm_fn is None:
            return self.optimize_with_ga()
        perf = self.evaluate_params(self.current_params)
        out = self.llm_fn(params_to_dict(self.current_params), perf)
        if isinstance(out, str):                 # strategy code text
            from .strategy_evaluator import params_from_code
            out = params_from_code(out)
        from ..backtesting.strategy import dict_to_params
        from .strategy_evaluator import validate_strategy
        base = params_to_dict(self.current_params)
        base.update(out or {})
        ok, issues = validate_strategy(base)   # validator gate (reference
        if not ok:                             # STRATEGY_EVOLUTION.md)
            self.log.warning("llm proposal issues (clipped): %s", issues)
        vec = clip_params(dict_to_params(base)[None])[0]
        return vec, {"llm": True, "validation_issues": issues,
                     **{k: v for k, v in perf.items()
                        if k in ("sharpe", "win_rate")}}

    def adjust_for_regime(self, params: np.ndarray, regime: str):
        """per-regime multiplier tables (reference :302, :145-174)."""
        adj = REGIME_ADJUSTMENTS.get(regime, {})
        p = params.copy()
        from ..backtesting.strategy import PARAM_NAMES
        for name, mult in adj.items():
            i = PARAM_NAMES.index(name)
            p[i] *= mult
        return clip_params(p[None])[0]

    def similar_version(self, params: np.ndarray,
                        threshold: float = 0.95) -> str | None:
        """Model-version similarity dedup (reference :1295-1400): a new
        parameter set within `threshold` cosine-style similarity (1 -
        normalized L2 over the bounds span) of an existing version
        reuses that version's id instead of minting a new one."""
        from ..backtesting.strategy import PARAM_BOUNDS, dict_to_params

        span = (PARAM_BOUNDS[:, 1] - PARAM_BOUNDS[:, 0]).astype(np.float64)
        for v in reversed(self.model_versions[-50:]):
            other = dict_to_params(v["params"]).astype(np.float64)
            d = np.abs(params.astype(np.float64) - other) / span
            sim = 1.0 - float(np.sqrt((d ** 2).mean()))
            if sim >= threshold:
                return v["id"]
        return None

    async def hot_swap(self, params: np.ndarray, perf: dict, regime: str):
        """strategy_params key + 'reload' publish (reference :349-362)."""
        self.current_params = params
        sid = self.similar_version(params) \
            or f"evolved-{uuid.uuid4().hex[:8]}"
        old = self.current_strategy_id
        self.current_strategy_id = sid
        await self.bus.set(Keys.STRATEGY_PARAMS, params_to_dict(params))
        await self.bus.publish(Channels.STRATEGY_UPDATE, "reload")
        await self.bus.publish(
            Channels.STRATEGY_EVOLUTION_UPDATES,
            EvolutionUpdate(sid, params_to_dict(params), perf,
                            market_regime=regime).to_dict())
        existing = next((v for v in self.model_versions
                         if v["id"] == sid), None)
        if existing is not None:
            # near-duplicate params: refresh the existing version's
            # performance rather than minting a new entry (:1295-1400)
            existing["performance"] = perf
            existing["at"] = time.time()
            await self.bus.publish(Channels.MODEL_REGISTRY_EVENTS, {
                "event": "strategy_version_reused", "id": sid,
            })
        else:
            self.model_versions.append({
                "id": sid, "previous": old,
                "params": params_to_dict(params),
                "performance": perf, "at": time.time(),
            })
            del self.model_versions[:-200]      # bounded history
            await self.bus.publish(Channels.MODEL_REGISTRY_EVENTS, {
                "event": "strategy_registered", "id": sid,
            })

    async def evolve_once(self) -> dict:
        regime_d = await self.bus.get_json(Keys.CURRENT_MARKET_REGIME) or {}
        regime = regime_d.get("regime", "ranging")
        vol = regime_d.get("volatility", 0.5)
        method = self.select_method(regime, vol)
        t0 = time.perf_counter()
        if method == "rl":
            params, perf = self.optimize_with_rl()
        elif method == "ppo":
            params, perf = self.optimize_with_ppo()
        elif method == "gpt":
            params, perf = self.optimize_with_llm()
        else:
            params, perf = self.optimize_with_ga()
        params = self.adjust_for_regime(params, regime)
        perf["method"] = method
        perf["seconds"] = time.perf_counter() - t0
        await self.hot_swap(params, perf, regime)
        self.evolutions += 1
        return perf

    async def run(self):
        while self.running:
            # monitor the live strategy: evaluate the current params on the
            # evaluation market and publish strategy_performance_{id}
            # (reference monitor_strategy :513 reads the Redis perf key the
            # executor fills; here the evolution service measures directly)
            try:
                perf = self.evaluate_params(self.current_params)
                await self.bus.set(
                    Keys.strategy_performance(self.current_strategy_id),
                    perf)
            except Exception as e:
                self.log.warning("perf evaluation failed: %r", e)
                perf = {}
            if not perf or self.needs_improvement(perf):
                try:
                    out = await self.evolve_once()
                    self.log.info("evolved via %s: %s", out.get("method"),
                                  {k: v for k, v in out.items()
                                   if k != "method"})
                except Exception as e:
                    self.log.warning("evolution failed: %s", e)
            await self.sleep(self.config.evolution.interval_s)

    def evaluate_params(self, params: np.ndarray) -> dict:
        """Metrics of a param set on the evaluation market (used for the
        strategy_performance key; strategy_evaluation.py:32-228 formulas
        via the engines). GPU kernel when available."""
        from ..backtesting.strategy import clip_params

        vec = clip_params(np.asarray(params, np.float32)[None])
        if str(self.device).startswith("cuda"):
            import torch

            from ..ops.backtest import run_backtest_gpu
            mt = run_backtest_gpu(
                torch.from_numpy(self.candles).to(self.device),
                torch.from_numpy(vec).to(self.device))
            torch.cuda.synchronize()
            m = mt.float().mean(dim=(0, 1)).cpu().numpy()
        else:
            from ..backtesting.engine_cpu import run_backtest_cpu

            m = run_backtest_cpu(self.candles, vec).mean(axis=(0, 1))
        d = dict(zip(METRIC_NAMES, (float(x) for x in m)))
        gp, gl = d["gross_profit"], d["gross_loss"]
        d["profit_factor"] = gp / gl if gl > 0 else float("inf")
        d["win_rate"] = d["wins"] / max(d["n_trades"], 1.0)
        return d
