"""Chart-pattern recognition (reference parity:
services/utils/pattern_recognition.py:26-1263 +
services/pattern_recognition_service.py:15-428).

A 1-D CNN classifier over normalized close windows for the reference's
14 chart patterns (config.json:516-531), trained on the synthetic pattern
generators (head&shoulders :863, double top :894, triangles :929,
rectangle :984, cup&handle :1014 — reimplemented parametrically here).
Torch model — trains on GPU when present (bf16-capable conv stack), CPU
otherwise."""

from __future__ import annotations

import numpy as np
import torch
import torch.nn as nn

PATTERNS = [
    "head_and_shoulders", "inverse_head_and_shoulders",
    "double_top", "double_bottom",
    "ascending_triangle", "descending_triangle", "symmetric_triangle",
    "rectangle", "cup_and_handle",
    "rising_wedge", "falling_wedge",
    "flag_bullish", "flag_bearish", "pennant",
    "none",
]
BULLISH = {"inverse_head_and_shoulders", "double_bottom",
           "ascending_triangle", "cup_and_handle", "falling_wedge",
           "flag_bullish"}
BEARISH = {"head_and_shoulders", "double_top", "descending_triangle",
           "rising_wedge", "flag_bearish"}

WIN = 64


def _noise(rng, n, scale=0.01):
    return rng.standard_normal(n) * scale


def generate_pattern(name: str, rng: np.random.Generator,
                     n: int = WIN) -> np.ndarray:
    """Synthetic normalized close window exhibiting `name`."""
    t = np.linspace(0, 1, n)
    if name == "head_and_shoulders":
        y = (0.5 * np.exp(-((t - 0.2) / 0.07) ** 2)
             + 1.0 * np.exp(-((t - 0.5) / 0.08) ** 2)
             + 0.5 * np.exp(-((t - 0.8) / 0.07) ** 2))
    elif name == "inverse_head_and_shoulders":
        return 2.0 - generate_pattern("head_and_shoulders", rng, n)
    elif name == "double_top":
        y = (np.exp(-((t - 0.3) / 0.08) ** 2)
             + np.exp(-((t - 0.7) / 0.08) ** 2))
    elif name == "double_bottom":
        return 2.0 - generate_pattern("double_top", rng, n)
    elif name == "ascending_triangle":
        y = 1.0 - (1 - t) * 0.5 * np.abs(np.sin(t * 12))
    elif name == "descending_triangle":
        y = (1 - t) * 0.5 * np.abs(np.sin(t * 12))
    elif name == "symmetric_triangle":
        y = 0.5 + (1 - t) * 0.4 * np.sin(t * 14)
    elif name == "rectangle":
        y = 0.5 + 0.2 * np.sign(np.sin(t * 16))
    elif name == "cup_and_handle":
        cup = 1.0 - 0.6 * np.sin(np.clip(t / 0.8, 0, 1) * np.pi)
        handle = np.where(t > 0.8, 0.9 - 0.1 * np.sin(
            (t - 0.8) / 0.2 * np.pi), 0)
        y = np.where(t <= 0.8, cup, handle + 0.1)
    elif name == "rising_wedge":
        y = t * 0.8 + 0.15 * (1 - t) * np.sin(t * 20)
    elif name == "falling_wedge":
        y = (1 - t) * 0.8 + 0.15 * t * np.sin(t * 20)
    elif name == "flag_bullish":
        y = np.where(t < 0.5, t * 1.6, 0.8 - (t - 0.5) * 0.2)
    elif name == "flag_bearish":
        y = np.where(t < 0.5, 1.0 - t * 1.6, 0.2 + (t - 0.5) * 0.2)
    elif name == "pennant":
        # sharp pole then a small converging (symmetric) consolidation
        y = np.where(t < 0.4, t * 2.0,
                     0.8 + 0.25 * (1 - t) * np.sin((t - 0.4) * 30))
    else:  # none: random walk
        y = np.cumsum(rng.standard_normal(n)) * 0.05
    y = y + _noise(rng, n, 0.04)
    y = (y - y.min()) / max(y.max() - y.min(), 1e-9)
    return y.astype(np.float32)


class PatternCNN(nn.Module):
    """Conv64/Conv32 + dense head (pattern_recognition.py:94-196 shape)."""

    def __init__(self, n_classes: int = len(PATTERNS), win: int = WIN):
        super().__init__()
        self.net = nn.Sequential(
            nn.Conv1d(1, 64, 5, padding=2), nn.ReLU(), nn.MaxPool1d(2),
            nn.Conv1d(64, 32, 3, padding=1), nn.ReLU(), nn.MaxPool1d(2),
            nn.Flatten(),
            nn.Linear(32 * (win // 4), 64), nn.ReLU(),
            nn.Linear(64, n_classes),
        )

    def forward(self, x):        # (B, WIN)
        return self.net(x[:, None, :])


class PatternLSTM(nn.Module):
    """LSTM(64)->LSTM(32)->dense classifier — the reference's LSTM
    pattern-model type (pattern_recognition.py:128-158)."""

    def __init__(self, n_classes: int = len(PATTERNS), win: int = WIN):
        super().__init__()
        self.l1 = nn.LSTM(1, 64, batch_first=True)
        self.l2 = nn.LSTM(64, 32, batch_first=True)
        self.head = nn.Sequential(nn.Linear(32, 64), nn.ReLU(),
                                  nn.Linear(64, n_classes))

    def forward(self, x):        # (B, WIN)
        h, _ = self.l1(x[:, :, None])
        h, _ = self.l2(h)
        return self.head(h[:, -1])


class PatternCNNLSTM(nn.Module):
    """Conv64/Conv32 front-end -> LSTM(32) -> dense — the reference's
    hybrid type (pattern_recognition.py:160-196)."""

    def __init__(self, n_classes: int = len(PATTERNS), win: int = WIN):
        super().__init__()
        self.conv = nn.Sequential(
            nn.Conv1d(1, 64, 5, padding=2), nn.ReLU(), nn.MaxPool1d(2),
            nn.Conv1d(64, 32, 3, padding=1), nn.ReLU(), nn.MaxPool1d(2),
        )
        self.lstm = nn.LSTM(32, 32, batch_first=True)
        self.head = nn.Linear(32, n_classes)

    def forward(self, x):        # (B, WIN)
        h = self.conv(x[:, None, :])            # (B, 32, WIN/4)
        h, _ = self.lstm(h.transpose(1, 2))     # (B, WIN/4, 32)
        return self.head(h[:, -1])


PATTERN_MODELS = {
    "cnn": PatternCNN, "lstm": PatternLSTM, "cnn_lstm": PatternCNNLSTM,
}


class PatternRecognitionModel:
    def __init__(self, device="cpu", seed: int = 0,
                 model_type: str = "cnn"):
        self.device = torch.device(device)
        torch.manual_seed(seed)
        self.model_type = model_type
        self.model = PATTERN_MODELS[model_type]().to(self.device)
        self.trained = False

    def make_dataset(self, n_per_class: int = 64, seed: int = 0):
        rng = np.random.default_rng(seed)
        X, y = [], []
        for ci, name in enumerate(PATTERNS):
            for _ in range(n_per_class):
                X.append(generate_pattern(name, rng))
                y.append(ci)
        return (torch.from_numpy(np.stack(X)),
                torch.tensor(y, dtype=torch.long))

    def train(self, epochs: int = 8, n_per_class: int = 64,
              lr: float = 2e-3, seed: int = 0) -> float:
        X, y = self.make_dataset(n_per_class, seed)
        X, y = X.to(self.device), y.to(self.device)
        opt = torch.optim.Adam(self.model.parameters(), lr=lr)
        lossf = nn.CrossEntropyLoss()
        n = len(X)
        for _ in range(epochs):
            perm = torch.randperm(n, device=self.device)
            for i in range(0, n, 128):
                j = perm[i:i + 128]
                opt.zero_grad()
                loss = lossf(self.model(X[j]), y[j])
                loss.backward()
                opt.step()
        with torch.no_grad():
            acc = float((self.model(X).argmax(1) == y).float().mean())
        self.trained = True
        return acc

    # typical pattern lengths in candles — the reference's table
    # (pattern_recognition.py:494-510)
    TYPICAL_LENGTH = {
        "head_and_shoulders": 30, "inverse_head_and_shoulders": 30,
        "double_top": 25, "double_bottom": 25,
        "ascending_triangle": 20, "descending_triangle": 20,
        "symmetric_triangle": 20, "rectangle": 20,
        "flag_bullish": 10, "flag_bearish": 10, "pennant": 15,
        "cup_and_handle": 40, "rising_wedge": 25, "falling_wedge": 25,
        "no_pattern": 1,
    }
    CONFIRMATION_PATTERNS = {
        "double_top", "double_bottom", "head_and_shoulders",
        "inverse_head_and_shoulders",
    }

    @classmethod
    def estimate_completion(cls, closes: np.ndarray,
                            pattern: str) -> float:
        """Completion %% with the reference estimator's exact semantics
        (pattern_recognition.py:476-530): data length over the typical
        pattern length, a confirmation-move boost for reversal patterns
        (mean |5-candle %%-change| / 10), rounded to the nearest 5%%."""
        if pattern in ("no_pattern", "none"):
            return 0.0
        typical = cls.TYPICAL_LENGTH.get(pattern, 20)
        completion = min(100.0, len(closes) / typical * 100.0)
        if pattern in cls.CONFIRMATION_PATTERNS and len(closes) >= 6:
            c = np.asarray(closes, np.float64)
            recent = np.abs(np.diff(c[-6:]) / c[-6:-1]).mean() * 100.0
            completion = min(100.0, completion * (1.0 + recent / 10.0))
        return round(completion / 5.0) * 5.0

    @torch.no_grad()
    def detect(self, closes: np.ndarray) -> dict:
        """Windowed detection + completion estimate
        (pattern_recognition.py:403-530)."""
        if len(closes) < WIN:
            return {"pattern": "none", "confidence": 0.0,
                    "signal": "neutral", "completion": 0.0}
        w = np.asarray(closes[-WIN:], np.float32)
        w = (w - w.min()) / max(w.max() - w.min(), 1e-9)
        x = torch.from_numpy(w[None]).to(self.device)
        probs = torch.softmax(self.model(x)[0], dim=0)
        ci = int(probs.argmax())
        name = PATTERNS[ci]
        conf = float(probs[ci])
        signal = ("bullish" if name in BULLISH else
                  "bearish" if name in BEARISH else "neutral")
        completion = self.estimate_completion(closes, name)
        return {"pattern": name, "confidence": conf, "signal": signal,
                "completion": completion}
