"""Price-predictor model zoo (reference parity:
neural_network_service.py `create_model` :164-421 + ensemble :423-485):

  lstm           LSTM(64)->LSTM(32)->Dense(16)->Dense(1)   (:191-200)
  gru            GRU stack                                  (:202-211)
  bilstm         bidirectional LSTM                         (:213-222)
  cnn_lstm       Conv64/Conv32 + LSTM(32)                   (:224-234)
  attention      recurrent encoder + multi-head attention   (:236-245)
  transformer    2-block encoder, sinusoidal pos-encoding   (:247-306)
  multitask      3 horizon heads, loss weights 1.0/0.7/0.5  (:308-353)
  probabilistic  Normal head trained with NLL               (:355-395)
  ensemble       LSTM+GRU+CNN concat                        (:423-485)

Recurrent cells are the fused gfx950 kernels (models/lstm.py, models/
gru.py) on GPU and fp32 references on CPU. Hyperparameter search:
models/hpo.py (the reference used Optuna's 20-trial study :588-767)."""

from __future__ import annotations

import math

import torch
import torch.nn as nn

from .gru import FusedGRULayer
from .lstm import FusedLSTMLayer

MODEL_TYPES = ("lstm", "gru", "bilstm", "cnn_lstm", "attention",
               "transformer", "multitask", "probabilistic", "ensemble")


class _Head(nn.Sequential):
    def __init__(self, d_in: int, d_out: int = 1):
        super().__init__(nn.Linear(d_in, 16), nn.ReLU(),
                         nn.Linear(16, d_out))


def _last(h: torch.Tensor) -> torch.Tensor:     # (T,B,H) -> (B,H) f32
    return h[-1].float()


class RecurrentPredictor(nn.Module):
    """lstm / gru stacks."""

    def __init__(self, cell: str, n_features: int = 9,
                 hidden: tuple[int, int] = (64, 32)):
        super().__init__()
        Layer = FusedLSTMLayer if cell == "lstm" else FusedGRULayer
        self.l1 = Layer(n_features, hidden[0])
        self.l2 = Layer(hidden[0], hidden[1])
        self.head = _Head(hidden[1])

    def forward(self, x):                        # (B,T,F)
        h = self.l2(self.l1(x.transpose(0, 1)))
        return self.head(_last(h)).squeeze(-1)


class BiLSTMPredictor(nn.Module):
    def __init__(self, n_features: int = 9, hidden: int = 64):
        super().__init__()
        self.fwd = FusedLSTMLayer(n_features, hidden)
        self.bwd = FusedLSTMLayer(n_features, hidden)
        self.head = _Head(2 * hidden)

    def forward(self, x):
        xt = x.transpose(0, 1)
        hf = self.fwd(xt)
        hb = self.bwd(torch.flip(xt, dims=[0]))
        return self.head(
            torch.cat([_last(hf), _last(hb)], dim=1)).squeeze(-1)


class CNNLSTMPredictor(nn.Module):
    def __init__(self, n_features: int = 9, hidden: int = 32):
        super().__init__()
        self.conv = nn.Sequential(
            nn.Conv1d(n_features, 64, 3, padding=1), nn.ReLU(),
            nn.Conv1d(64, 32, 3, padding=1), nn.ReLU(),
        )
        self.rnn = FusedLSTMLayer(32, hidden)
        self.head = _Head(hidden)

    def forward(self, x):                        # (B,T,F)
        c = self.conv(x.transpose(1, 2)).transpose(1, 2)   # (B,T,32)
        h = self.rnn(c.transpose(0, 1))
        return self.head(_last(h)).squeeze(-1)


class AttentionPredictor(nn.Module):
    """Recurrent encoder + multi-head self-attention pooling (:236-245)."""

    def __init__(self, n_features: int = 9, hidden: int = 64,
                 n_heads: int = 4):
        super().__init__()
        self.enc = FusedLSTMLayer(n_features, hidden)
        self.attn = nn.MultiheadAttention(hidden, n_heads,
                                          batch_first=True)
        self.head = _Head(hidden)

    def forward(self, x):
        h = self.enc(x.transpose(0, 1)).float().transpose(0, 1)  # (B,T,H)
        a, _ = self.attn(h[:, -1:], h, h)
        return self.head(a[:, 0]).squeeze(-1)


class PositionalEncoding(nn.Module):
    """Sinusoidal (:247-306)."""

    def __init__(self, d_model: int, max_len: int = 512):
        super().__init__()
        pe = torch.zeros(max_len, d_model)
        pos = torch.arange(max_len).unsqueeze(1).float()
        div = torch.exp(torch.arange(0, d_model, 2).float()
                        * (-math.log(10000.0) / d_model))
        pe[:, 0::2] = torch.sin(pos * div)
        pe[:, 1::2] = torch.cos(pos * div)
        self.register_buffer("pe", pe)

    def forward(self, x):                        # (B,T,D)
        return x + self.pe[: x.shape[1]]


class TransformerPredictor(nn.Module):
    """2-block encoder on the fused gfx950 attention kernel
    (models/attention.py; torch SDPA on CPU)."""

    def __init__(self, n_features: int = 9, d_model: int = 64,
                 n_heads: int = 4, n_layers: int = 2):
        super().__init__()
        from .attention import FusedTransformerEncoderLayer

        self.proj = nn.Linear(n_features, d_model)
        self.pos = PositionalEncoding(d_model)
        self.layers = nn.ModuleList([
            FusedTransformerEncoderLayer(d_model, n_heads, 128)
            for _ in range(n_layers)
        ])
        self.head = _Head(d_model)

    def forward(self, x):
        h = self.pos(self.proj(x))
        for layer in self.layers:
            h = layer(h)
        return self.head(h[:, -1]).squeeze(-1)


class MultitaskPredictor(nn.Module):
    """3 horizon heads (1h/4h/24h), loss weights 1.0/0.7/0.5 (:308-353)."""

    LOSS_WEIGHTS = (1.0, 0.7, 0.5)

    def __init__(self, n_features: int = 9, hidden: int = 64):
        super().__init__()
        self.enc = FusedLSTMLayer(n_features, hidden)
        self.heads = nn.ModuleList([_Head(hidden) for _ in range(3)])

    def forward(self, x):                        # -> (B, 3)
        z = _last(self.enc(x.transpose(0, 1)))
        return torch.stack([h(z).squeeze(-1) for h in self.heads], dim=1)

    def loss(self, pred, targets):               # targets (B, 3)
        per = ((pred - targets) ** 2).mean(dim=0)
        w = pred.new_tensor(self.LOSS_WEIGHTS)
        return (per * w).sum() / w.sum()


class ProbabilisticPredictor(nn.Module):
    """mean + logvar head trained with Gaussian NLL (:355-395)."""

    def __init__(self, n_features: int = 9, hidden: int = 64):
        super().__init__()
        self.enc = FusedLSTMLayer(n_features, hidden)
        self.mu = _Head(hidden)
        self.logvar = _Head(hidden)

    def forward(self, x):
        z = _last(self.enc(x.transpose(0, 1)))
        return self.mu(z).squeeze(-1), self.logvar(z).squeeze(-1)

    @staticmethod
    def nll(mu, logvar, y):
        return 0.5 * (logvar + (y - mu) ** 2 / logvar.exp()).mean()


class EnsemblePredictor(nn.Module):
    """LSTM + GRU + CNN feature concat (:423-485)."""

    def __init__(self, n_features: int = 9, hidden: int = 32):
        super().__init__()
        self.lstm = FusedLSTMLayer(n_features, hidden)
        self.gru = FusedGRULayer(n_features, hidden)
        self.conv = nn.Sequential(
            nn.Conv1d(n_features, 32, 3, padding=1), nn.ReLU(),
            nn.AdaptiveAvgPool1d(1),
        )
        self.head = _Head(2 * hidden + 32)

    def forward(self, x):
        xt = x.transpose(0, 1)
        z = torch.cat([
            _last(self.lstm(xt)), _last(self.gru(xt)),
            self.conv(x.transpose(1, 2)).squeeze(-1),
        ], dim=1)
        return self.head(z).squeeze(-1)


def create_model(model_type: str, n_features: int = 9,
                 hidden: tuple[int, int] = (64, 32), **kw) -> nn.Module:
    """Factory (neural_network_service.py:164-421 contract)."""
    if model_type == "lstm":
        return RecurrentPredictor("lstm", n_features, hidden)
    if model_type == "gru":
        return RecurrentPredictor("gru", n_features, hidden)
    if model_type == "bilstm":
        return BiLSTMPredictor(n_features, hidden[0])
    if model_type == "cnn_lstm":
        return CNNLSTMPredictor(n_features, hidden[1])
    if model_type == "attention":
        return AttentionPredictor(n_features, hidden[0])
    if model_type == "transformer":
        return TransformerPredictor(n_features, hidden[0])
    if model_type == "multitask":
        return MultitaskPredictor(n_features, hidden[0])
    if model_type == "probabilistic":
        return ProbabilisticPredictor(n_features, hidden[0])
    if model_type == "ensemble":
        return EnsemblePredictor(n_features, hidden[1])
    raise ValueError(f"unknown model type '{model_type}' "
                     f"(known: {MODEL_TYPES})")
