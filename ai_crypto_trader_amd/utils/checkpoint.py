"""Unified run-state checkpoint/resume.

The reference checkpoints only model artifacts (Keras .h5, RL weights,
regime pickles — SURVEY.md §5 'Checkpoint/resume') and has NO unified
run-state checkpointing. This module adds it: one directory captures
  - GA population + generation + RNG seed (backtesting/ga_engine.py)
  - torch model/optimizer state dicts (predictors, RL agents)
  - service KV state (bus key snapshot: holdings, strategy params, ...)
  - a manifest with versions + timestamps
so a killed run_trader/evolution/training job resumes where it stopped.
"""

from __future__ import annotations

import json
import time
from pathlib import Path

import numpy as np


class CheckpointManager:
    def __init__(self, root: str = "checkpoints", keep: int = 5):
        self.root = Path(root)
        self.keep = keep
        self.root.mkdir(parents=True, exist_ok=True)

    # --- generic ---------------------------------------------------------
    def new_dir(self, tag: str) -> Path:
        d = self.root / f"{tag}-{int(time.time())}"
        d.mkdir(parents=True, exist_ok=True)
        self._prune(tag)
        return d

    def latest(self, tag: str) -> Path | None:
        cands = sorted(self.root.glob(f"{tag}-*"))
        return cands[-1] if cands else None

    def _prune(self, tag: str):
        cands = sorted(self.root.glob(f"{tag}-*"))
        for old in cands[: max(len(cands) - self.keep, 0)]:
            import shutil
            shutil.rmtree(old, ignore_errors=True)

    # --- GA engine -------------------------------------------------------
    def save_ga(self, engine, tag: str = "ga") -> Path:
        d = self.new_dir(tag)
        np.save(d / "population.npy", engine.pop_t.cpu().numpy())
        manifest = {
            "kind": "ga", "gen": engine.gen, "seed": engine.seed,
            "pop_per_rank": engine.pop_per_rank, "world": engine.world,
            "elite_k": engine.elite_k, "at": time.time(),
        }
        if engine.last_fitness_global is not None:
            np.save(d / "fitness.npy",
                    engine.last_fitness_global.cpu().numpy())
        (d / "manifest.json").write_text(json.dumps(manifest, indent=2))
        return d

    def load_ga(self, engine, tag: str = "ga") -> bool:
        import torch

        d = self.latest(tag)
        if d is None:
            return False
        manifest = json.loads((d / "manifest.json").read_text())
        pop = np.load(d / "population.npy")
        engine.pop_t = torch.from_numpy(pop).to(engine.device)
        engine.gen = manifest["gen"]
        fit = d / "fitness.npy"
        if fit.exists():
            engine.last_fitness_global = torch.from_numpy(
                np.load(fit)).to(engine.device)
        return True

    # --- torch models ----------------------------------------------------
    def save_model(self, model, optimizer=None, tag: str = "model",
                   meta: dict | None = None) -> Path:
        import torch

        d = self.new_dir(tag)
        torch.save(model.state_dict(), d / "model.pt")
        if optimizer is not None:
            torch.save(optimizer.state_dict(), d / "optimizer.pt")
        (d / "manifest.json").write_text(json.dumps(
            {"kind": "model", "at": time.time(), **(meta or {})}, indent=2))
        return d

    def load_model(self, model, optimizer=None, tag: str = "model",
                   map_location="cpu") -> dict | None:
        import torch

        d = self.latest(tag)
        if d is None:
            return None
        model.load_state_dict(torch.load(d / "model.pt",
                                         map_location=map_location))
        opt_p = d / "optimizer.pt"
        if optimizer is not None and opt_p.exists():
            optimizer.load_state_dict(torch.load(
                opt_p, map_location=map_location))
        return json.loads((d / "manifest.json").read_text())

    # --- bus state -------------------------------------------------------
    async def save_bus_state(self, bus, keys: list[str],
                             tag: str = "bus") -> Path:
        d = self.new_dir(tag)
        state = {}
        for k in keys:
            v = await bus.get(k)
            if v is not None:
                state[k] = v
        (d / "state.json").write_text(json.dumps(state, indent=2))
        (d / "manifest.json").write_text(json.dumps(
            {"kind": "bus", "n_keys": len(state), "at": time.time()},
            indent=2))
        return d

    async def load_bus_state(self, bus, tag: str = "bus") -> int:
        d = self.latest(tag)
        if d is None:
            return 0
        state = json.loads((d / "state.json").read_text())
        for k, v in state.items():
            await bus.set(k, v)
        return len(state)
