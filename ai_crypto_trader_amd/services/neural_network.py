"""Neural-network prediction service (reference parity:
services/neural_network_service.py:30-1617).

Trains the LSTM price predictor (models/lstm.py — custom gfx950 recurrent
kernels on GPU) on rolling candle windows (seq_len 60 x 9 features,
config.json:403-499 semantics), predicts next-candle change, and
publishes `neural_network_predictions` + the `nn_prediction_{sym}_{intv}`
keys. Feature scaling is min-max over the training window (reference
:530-586); checkpoints to model_dir (:148, :907)."""

from __future__ import annotations

import time
from pathlib import Path

import numpy as np
import torch

from ..bus.schema import Channels, Keys, NNPrediction
from ..models.lstm import LSTMPricePredictor
from .base import Service

N_FEAT = 9


def build_features(candles: np.ndarray) -> np.ndarray:
    """(T, 4) [close, high, low, vol] -> (T, 9) feature matrix: the
    reference's 9-feature set (close/high/low/volume + rsi + macd + bb
    position + returns; neural_network_service.py:530-586)."""
    from ..ops.indicators import indicators_fast

    ind = indicators_fast(candles[None])[0]    # (T, 13), vectorized path
    close = candles[:, 0]
    ret1 = np.zeros_like(close)
    ret1[1:] = close[1:] / close[:-1] - 1.0
    bb_pos = (close - ind[:, 8]) / np.maximum(ind[:, 7] - ind[:, 8], 1e-9)
    feats = np.stack([
        close, candles[:, 1], candles[:, 2], candles[:, 3],
        ind[:, 5] / 100.0,          # rsi
        ind[:, 4],                  # macd_hist
        bb_pos,
        ret1 * 100.0,
        ind[:, 9],                  # atr
    ], axis=1).astype(np.float32)
    return feats


def make_windows(feats: np.ndarray, targets: np.ndarray, seq_len: int):
    """Sliding windows (B, seq, F) + next-step targets."""
    T = feats.shape[0]
    n = T - seq_len
    if n <= 0:
        return None, None
    idx = np.arange(n)[:, None] + np.arange(seq_len)[None]
    return feats[idx], targets[seq_len:]


class MinMaxScaler:
    """Per-feature min-max to [0,1] (reference uses sklearn MinMaxScaler)."""

    def fit(self, X: np.ndarray):
        flat = X.reshape(-1, X.shape[-1])
        self.lo = flat.min(axis=0)
        self.hi = flat.max(axis=0)
        self.span = np.maximum(self.hi - self.lo, 1e-9)
        return self

    def transform(self, X):
        return (X - self.lo) / self.span


class NeuralNetworkService(Service):
    name = "neural_network"

    def __init__(self, bus, config=None, device="cpu"):
        super().__init__(bus, config)
        self.device = torch.device(device)
        self.models: dict[str, LSTMPricePredictor] = {}
        self.scalers: dict[str, MinMaxScaler] = {}
        # (sym, regime) -> snapshot (reference :1445-1473)
        self.regime_models: dict[tuple, LSTMPricePredictor] = {}
        self.regime_scalers: dict[tuple, MinMaxScaler] = {}
        self.val_loss: dict[str, float] = {}
        self.candles: dict[str, list] = {}
        self.trained = 0
        self.predictions = 0

    def run_tasks(self):
        return [self._consume_market(), self._train_loop(),
                self._predict_loop()]

    async def _consume_market(self):
        sub = self.bus.subscribe(Channels.MARKET_UPDATES)

        def on_msg(_, m):
            sym = m.get("symbol")
            if sym:
                h = self.candles.setdefault(sym, [])
                h.append([m["current_price"], m["current_price"],
                          m["current_price"], m.get("avg_volume", 1.0)])
                if len(h) > 8192:
                    del h[:4096]

        await self.consume(sub, on_msg)

    # --- training --------------------------------------------------------
    def train(self, sym: str, candles: np.ndarray, epochs: int | None = None,
              batch: int | None = None) -> float:
        cfg = self.config.neural_network
        feats = build_features(candles)
        targets = np.zeros(len(feats), np.float32)
        close = candles[:, 0]
        targets[:-1] = (close[1:] / close[:-1] - 1.0) * 100.0
        scaler = MinMaxScaler().fit(feats)
        X, y = make_windows(scaler.transform(feats), targets, cfg.seq_len)
        if X is None or len(X) < 64:
            return float("nan")
        n_val = max(len(X) // 10, 1)
        Xt = torch.from_numpy(X[:-n_val]).to(self.device)
        yt = torch.from_numpy(y[:-n_val]).to(self.device)
        Xv = torch.from_numpy(X[-n_val:]).to(self.device)
        yv = torch.from_numpy(y[-n_val:]).to(self.device)

        model = self.models.get(sym)
        if model is None:
            model = LSTMPricePredictor(
                n_features=N_FEAT, seq_len=cfg.seq_len,
                hidden=tuple(cfg.hidden)).to(self.device)
            self.models[sym] = model
        opt = torch.optim.Adam(model.parameters(), lr=cfg.lr)
        bs = batch or cfg.batch_size
        best_val = float("inf")
        patience, bad = 5, 0
        for ep in range(epochs or min(cfg.epochs, 10)):
            perm = torch.randperm(len(Xt), device=self.device)
            for i in range(0, len(Xt), bs):
                j = perm[i:i + bs]
                opt.zero_grad()
                loss = ((model(Xt[j]) - yt[j]) ** 2).mean()
                loss.backward()
                opt.step()
            with torch.no_grad():
                vl = float(((model(Xv) - yv) ** 2).mean())
            if vl < best_val - 1e-6:
                best_val, bad = vl, 0
            else:
                bad += 1
                if bad >= patience:        # EarlyStopping (:805-1012)
                    break
        self.scalers[sym] = scaler
        self.val_loss[sym] = best_val
        self.trained += 1
        return best_val

    def snapshot_for_regime(self, sym: str, regime: str):
        """Regime-specific model snapshot (reference
        neural_network_service.py:1445-1473): keep a copy of the
        current trained model tagged with the regime it was trained
        under; predict() prefers the snapshot matching the current
        regime."""
        import copy

        model = self.models.get(sym)
        if model is None:
            return None
        key = (sym, regime)
        self.regime_models[key] = copy.deepcopy(model)
        self.regime_scalers[key] = self.scalers.get(sym)
        return key

    def model_for(self, sym: str, regime: str | None = None):
        """Current-regime snapshot if present, else the live model."""
        if regime is not None:
            key = (sym, regime)
            if key in self.regime_models:
                return self.regime_models[key], self.regime_scalers[key]
        return self.models.get(sym), self.scalers.get(sym)

    def save(self, directory: str | None = None):
        d = Path(directory or self.config.neural_network.model_dir)
        d.mkdir(parents=True, exist_ok=True)
        for sym, model in self.models.items():
            torch.save(model.state_dict(), d / f"nn_model_lstm_{sym}.pt")

    def load(self, directory: str | None = None) -> int:
        d = Path(directory or self.config.neural_network.model_dir)
        n = 0
        cfg = self.config.neural_network
        for p in d.glob("nn_model_lstm_*.pt"):
            sym = p.stem.replace("nn_model_lstm_", "")
            m = LSTMPricePredictor(N_FEAT, cfg.seq_len,
                                   tuple(cfg.hidden)).to(self.device)
            m.load_state_dict(torch.load(p, map_location=self.device))
            self.models[sym] = m
            n += 1
        return n

    # --- prediction ------------------------------------------------------
    def predict(self, sym: str, candles: np.ndarray,
                regime: str | None = None) -> dict | None:
        model, scaler = self.model_for(sym, regime)
        if model is None or scaler is None:
            return None
        cfg = self.config.neural_network
        # recurrent indicators (EMA/RSI/ATR) reconverge within 512 candles
        # (same argument as indicators.hip's chunk warmup: the forgetting
        # factor underflows f32), so serving only ever needs the tail —
        # not the whole history per request
        candles = candles[-(cfg.seq_len + 512):]
        feats = build_features(candles)[-cfg.seq_len:]
        if len(feats) < cfg.seq_len:
            return None
        x = torch.from_numpy(
            scaler.transform(feats)[None].astype(np.float32)
        ).to(self.device)
        with torch.no_grad():
            change_pct = float(model(x)[0])
        price = float(candles[-1, 0])
        # confidence from validation loss (reference :1090-1219)
        vl = self.val_loss.get(sym, 1.0)
        conf = float(np.clip(1.0 / (1.0 + 10.0 * vl), 0.05, 0.99))
        return NNPrediction(
            symbol=sym, interval="1m",
            predicted_price=price * (1 + change_pct / 100.0),
            current_price=price,
            predicted_change_pct=change_pct,
            confidence=conf,
            training_metrics={"val_loss": round(vl, 6)},
            features_used=list(self.FEATURE_NAMES),
        ).to_dict()

    FEATURE_NAMES = ["close", "high", "low", "volume", "rsi",
                     "macd_hist", "bb_position", "return_1", "atr"]

    def feature_importance(self, sym: str, candles: np.ndarray) -> dict | None:
        """Per-input-feature attribution on the fused-LSTM predictor —
        the offline counterpart of the reference's SHAP DeepExplainer
        (neural_network_service.py:957-1003). Two methods behind
        config.neural_network.attribution:
          grad_input            |d pred / d feature * feature|
          integrated_gradients  (x - x0) * mean_k grad(x0 + k/m (x-x0)),
                                baseline x0 = batch-mean window; IG's
                                completeness axiom makes attributions sum
                                to f(x) - f(x0), the SHAP-style property
        Both averaged over window steps and batch, normalized to sum 1."""
        model = self.models.get(sym)
        scaler = self.scalers.get(sym)
        if model is None or scaler is None:
            return None
        cfg = self.config.neural_network
        feats = build_features(candles)
        X, _ = make_windows(scaler.transform(feats).astype(np.float32),
                            np.zeros(len(feats), np.float32), cfg.seq_len)
        if X is None:
            return None
        x = torch.from_numpy(X[-64:]).to(self.device)
        if getattr(cfg, "attribution", "grad_input") == \
                "integrated_gradients":
            sal = self._integrated_gradients(model, x)
        else:
            x = x.requires_grad_(True)
            model(x).sum().backward()
            sal = (x.grad * x).abs().mean(dim=(0, 1))       # (F,)
        sal = (sal / sal.sum().clamp_min(1e-12)).detach().cpu().numpy()
        return {name: float(v)
                for name, v in zip(self.FEATURE_NAMES, sal)}

    @staticmethod
    def _integrated_gradients(model, x: torch.Tensor,
                              steps: int = 24) -> torch.Tensor:
        """Integrated gradients (Sundararajan et al.) along the straight
        path from the batch-mean baseline to each window."""
        x0 = x.mean(dim=0, keepdim=True).expand_as(x)
        acc = torch.zeros_like(x[0, 0])
        total = torch.zeros_like(x)
        for k in range(1, steps + 1):
            xi = (x0 + (k / steps) * (x - x0)).detach() \
                .requires_grad_(True)
            model(xi).sum().backward()
            total = total + xi.grad
        ig = (x - x0) * total / steps
        acc = ig.abs().mean(dim=(0, 1))                     # (F,)
        return acc

    def importance_report(self, sym: str,
                          candles: np.ndarray) -> dict | None:
        """The `feature_importance` report shape the reference publishes
        (README.md:420-468 / feature_importance_analyzer.py:610),
        sourced from the NN attribution instead of the signal RF."""
        imp = self.feature_importance(sym, candles)
        if imp is None:
            return None
        ranked = sorted(imp.items(), key=lambda kv: -kv[1])
        cfg = self.config.neural_network
        return {
            "symbol": sym,
            "model": "lstm_price_predictor",
            "method": getattr(cfg, "attribution", "grad_input"),
            "feature_importance": imp,
            "top_features": [k for k, _ in ranked[:5]],
            "recommendations": [
                f"feature '{ranked[-1][0]}' contributes "
                f"{ranked[-1][1]:.1%} — candidate to prune"
            ],
            "at": time.time(),
        }

    async def _train_loop(self):
        while self.running:
            for sym, h in list(self.candles.items()):
                if len(h) >= 256 and sym not in self.models:
                    try:
                        vl = self.train(sym, np.asarray(h, np.float32),
                                        epochs=2)
                        self.log.info("trained %s val_loss=%.5f", sym, vl)
                    except Exception as e:
                        self.log.warning("train %s failed: %s", sym, e)
            await self.sleep(10.0)

    async def _predict_loop(self):
        while self.running:
            for sym, h in list(self.candles.items()):
                if sym in self.models:
                    p = self.predict(sym, np.asarray(h, np.float32))
                    if p:
                        await self.bus.set(Keys.nn_prediction(sym, "1m"), p)
                        await self.bus.publish(Channels.NN_PREDICTIONS, p)
                        self.predictions += 1
            await self.sleep(5.0)

    async def run(self):
        pass
