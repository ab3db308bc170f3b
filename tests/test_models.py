"""Model zoo (CPU paths), GRU reference, HPO."""

import numpy as np
import pytest
import torch

from ai_crypto_trader_amd.models.gru import FusedGRULayer
from ai_crypto_trader_amd.models.hpo import Pruned, RandomSearchStudy
from ai_crypto_trader_amd.models.zoo import (
    MODEL_TYPES, ProbabilisticPredictor, create_model,
)


def test_gru_reference_matches_torch_gru():
    """Our fp32 GRU reference must match torch.nn.GRU exactly (same
    formulation) — this is the oracle the HIP kernel is tested against."""
    torch.manual_seed(0)
    T, B, F, H = 12, 4, 9, 32
    ours = FusedGRULayer(F, H)
    ref = torch.nn.GRU(F, H)
    # copy weights: torch GRU uses (3H, F) weight_ih with gate order r,z,n
    with torch.no_grad():
        ref.weight_ih_l0.copy_(ours.w_ih.t())
        ref.weight_hh_l0.copy_(ours.w_hh.t())
        ref.bias_ih_l0.copy_(ours.b_ih)
        ref.bias_hh_l0.copy_(ours.b_hh)
    x = torch.randn(T, B, F)
    out_ours = ours._forward_reference(x)
    out_ref, _ = ref(x)
    torch.testing.assert_close(out_ours, out_ref, rtol=1e-5, atol=1e-5)


@pytest.mark.parametrize("mt", [m for m in MODEL_TYPES])
def test_zoo_forward_shapes(mt):
    torch.manual_seed(1)
    model = create_model(mt, n_features=9)
    x = torch.randn(8, 20, 9)
    out = model(x)
    if mt == "multitask":
        assert out.shape == (8, 3)
        loss = model.loss(out, torch.randn(8, 3))
        assert torch.isfinite(loss)
    elif mt == "probabilistic":
        mu, logvar = out
        assert mu.shape == (8,) and logvar.shape == (8,)
        assert torch.isfinite(
            ProbabilisticPredictor.nll(mu, logvar, torch.randn(8)))
    else:
        assert out.shape == (8,)
    # gradients flow
    if mt == "probabilistic":
        (mu.sum() + logvar.sum()).backward()
    elif mt == "multitask":
        out.sum().backward()
    else:
        out.sum().backward()
    grads = [p.grad for p in model.parameters() if p.requires_grad]
    assert any(g is not None and g.abs().sum() > 0 for g in grads)


def test_unknown_model_type():
    with pytest.raises(ValueError):
        create_model("nope")


def test_hpo_finds_good_region():
    study = RandomSearchStudy(
        {"x": ("uniform", -5.0, 5.0), "lr": ("log", 1e-4, 1e-1)}, seed=3)

    def objective(trial):
        return (trial.params["x"] - 2.0) ** 2

    best = study.optimize(objective, n_trials=30)
    assert abs(best.params["x"] - 2.0) < 1.0
    assert best.state == "complete"


def test_hpo_pruning():
    study = RandomSearchStudy({"x": ("uniform", 0.0, 1.0)}, seed=0,
                              prune_after=1, prune_quantile=0.5)

    def objective(trial):
        for step in range(4):
            v = trial.params["x"] + step * 0.01
            if study.should_prune(trial, step, v):
                raise Pruned()
        return trial.params["x"]

    best = study.optimize(objective, n_trials=20)
    states = {t.state for t in study.trials}
    assert "pruned" in states            # bad trials get cut
    assert best.value == min(t.value for t in study.trials
                             if t.state == "complete")


def test_tpe_study_beats_random():
    """TPE sampler (the reference's Optuna default) converges better than
    random search on a smooth objective."""
    from ai_crypto_trader_amd.models.hpo import RandomSearchStudy, TPEStudy

    space = {"x": ("uniform", -5.0, 5.0), "lr": ("log", 1e-4, 1e-1)}

    def obj(t):
        return ((t.params["x"] - 1.7) ** 2
                + (np.log10(t.params["lr"]) + 2) ** 2)

    rs = RandomSearchStudy(space, seed=1)
    rs.optimize(obj, 40)
    tpe = TPEStudy(space, seed=1)
    tpe.optimize(obj, 40)
    assert tpe.best_trial.value < 0.1
    assert tpe.best_trial.value <= rs.best_trial.value * 1.2


def test_nn_feature_saliency():
    """Gradient x input importance (SHAP counterpart): returns a
    normalized per-feature dict after training."""
    from ai_crypto_trader_amd.bus.message_bus import InProcessBus
    from ai_crypto_trader_amd.config import AppConfig
    from ai_crypto_trader_amd.data.synthetic import (
        candles_chl_v, generate_ohlcv,
    )
    from ai_crypto_trader_amd.services.neural_network import (
        NeuralNetworkService,
    )

    cfg = AppConfig()
    cfg.neural_network.epochs = 1
    svc = NeuralNetworkService(InProcessBus(), cfg, device="cpu")
    candles = candles_chl_v(generate_ohlcv(800, 1, seed=3))[0]
    svc.train("BTCUSDC", candles, epochs=1)
    imp = svc.feature_importance("BTCUSDC", candles)
    assert imp is not None and len(imp) == 9
    assert abs(sum(imp.values()) - 1.0) < 1e-4
    assert all(v >= 0 for v in imp.values())


def test_pattern_cnn_per_class_accuracy_threshold():
    """train-accuracy evidence across all 14 pattern
    classes — held-out accuracy must clear a real threshold overall AND
    no class may collapse (reference trains a CNN over the same 14
    synthetic generators, pattern_recognition.py:863-1041)."""
    import numpy as np
    import torch

    from ai_crypto_trader_amd.models.patterns import (
        PATTERNS, PatternRecognitionModel,
    )

    assert len(PATTERNS) == 15       # 14 chart patterns + "none"
    m = PatternRecognitionModel(seed=0)
    train_acc = m.train(epochs=10, n_per_class=48, seed=0)
    assert train_acc >= 0.9, f"train accuracy {train_acc:.2f}"

    # held-out set from a DIFFERENT seed
    Xh, yh = m.make_dataset(n_per_class=24, seed=99)
    with torch.no_grad():
        pred = m.model(Xh).argmax(1)
    acc = float((pred == yh).float().mean())
    assert acc >= 0.8, f"held-out accuracy {acc:.2f}"
    per_class = np.array([
        float((pred[yh == ci] == ci).float().mean())
        for ci in range(len(PATTERNS))
    ])
    assert (per_class >= 0.5).all(), (
        f"collapsed classes: "
        f"{[PATTERNS[i] for i in np.where(per_class < 0.5)[0]]} "
        f"({per_class.round(2).tolist()})"
    )


def test_pattern_completion_matches_reference_estimator():
    """Completion %% reproduces the reference estimator's semantics
    (pattern_recognition.py:476-530): typical-length table,
    confirmation boost for reversal patterns, round-to-5."""
    import numpy as np

    from ai_crypto_trader_amd.models.patterns import (
        PatternRecognitionModel as M,
    )

    flat = np.full(10, 100.0)
    # 10 candles of a 20-candle triangle -> 50%
    assert M.estimate_completion(flat, "ascending_triangle") == 50.0
    # 40+ candles saturate at 100%
    assert M.estimate_completion(np.full(40, 100.0),
                                 "ascending_triangle") == 100.0
    # flag_bull typical length 10 -> 10 candles = 100%
    assert M.estimate_completion(flat, "flag_bullish") == 100.0
    assert M.estimate_completion(flat, "no_pattern") == 0.0
    # confirmation boost: double_top with a sharp recent move exceeds
    # the unboosted value and rounds to a multiple of 5
    closes = np.concatenate([np.full(20, 100.0),
                             [100, 98, 96, 94, 92]])   # 25 candles
    c = M.estimate_completion(closes, "double_top")
    base = min(100.0, len(closes) / 25 * 100)          # 100 unboosted
    assert c % 5 == 0 and c >= min(base, 100.0) - 1e-9
    quiet = np.full(25, 100.0)
    c_quiet = M.estimate_completion(quiet, "double_top")
    assert c_quiet == 100.0                            # len == typical
    shorter = np.concatenate([np.full(10, 100.0), [99.9] * 2])
    c_short = M.estimate_completion(shorter, "double_top")
    assert c_short < 100.0 and c_short % 5 == 0
