"""GPU numerics tests: every HIP kernel vs its CPU/torch fp32 reference.

All tests here require an MI355X (run via `pytest -m gpu`). They fail
loudly (never skip) if the HIP extension is missing on a GPU machine.
"""

import numpy as np
import pytest

torch = pytest.importorskip("torch")

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    return torch.device("cuda:0")


def test_extension_loads_on_gpu():
    from ai_crypto_trader_amd.ops import require_hip_ops
    mod = require_hip_ops()
    assert mod.GFX_ARCH == "gfx950"


def test_mfma_f32_layout(dev):
    """Validates the mfma_f32_16x16x4f32 fragment mapping (covar.hip)
    against torch.matmul — asymmetric inputs so transposes can't hide."""
    from ai_crypto_trader_amd.ops import require_hip_ops
    ops = require_hip_ops()
    M, N, K = 32, 48, 64
    g = torch.Generator(device="cpu").manual_seed(0)
    A = torch.randn(M, K, generator=g).float()
    B = torch.randn(K, N, generator=g).float() + torch.arange(N).float() / N
    Ad, Bd = A.to(dev), B.to(dev)
    C = torch.empty(M, N, device=dev, dtype=torch.float32)
    stream = torch.cuda.current_stream(dev).cuda_stream
    ops.mfma_gemm_test(Ad.data_ptr(), Bd.data_ptr(), C.data_ptr(),
                       M, N, K, stream)
    torch.cuda.synchronize()
    ref = A @ B
    torch.testing.assert_close(C.cpu(), ref, rtol=1e-5, atol=1e-5)


def test_backtest_gpu_matches_cpu(dev, small_market):
    from ai_crypto_trader_amd.backtesting.engine_cpu import run_backtest_cpu
    from ai_crypto_trader_amd.backtesting.strategy import random_population
    from ai_crypto_trader_amd.ops.backtest import run_backtest_gpu

    pop = random_population(64, seed=21)
    ref = run_backtest_cpu(small_market, pop)

    c_t = torch.from_numpy(small_market).to(dev)
    p_t = torch.from_numpy(pop).to(dev)
    got = run_backtest_gpu(c_t, p_t)
    torch.cuda.synchronize()
    got = got.cpu().numpy()

    # integer-valued metrics must agree exactly (same decisions per candle)
    np.testing.assert_array_equal(got[..., 1], ref[..., 1])   # n_trades
    np.testing.assert_array_equal(got[..., 2], ref[..., 2])   # wins
    np.testing.assert_allclose(got[..., 0], ref[..., 0], rtol=2e-5)
    np.testing.assert_allclose(got[..., 5], ref[..., 5], rtol=1e-3,
                               atol=1e-6)
    np.testing.assert_allclose(got[..., 9], ref[..., 9], rtol=1e-3,
                               atol=1e-3)


def test_backtest_gpu_nondivisible_pop(dev, small_market):
    """P not a multiple of the 256-lane block must still work."""
    from ai_crypto_trader_amd.backtesting.engine_cpu import run_backtest_cpu
    from ai_crypto_trader_amd.backtesting.strategy import random_population
    from ai_crypto_trader_amd.ops.backtest import run_backtest_gpu

    pop = random_population(37, seed=5)
    ref = run_backtest_cpu(small_market, pop)
    got = run_backtest_gpu(
        torch.from_numpy(small_market).to(dev), torch.from_numpy(pop).to(dev)
    )
    torch.cuda.synchronize()
    np.testing.assert_array_equal(got.cpu().numpy()[..., 1], ref[..., 1])


def test_mc_gpu_matches_cpu_reference(dev):
    from ai_crypto_trader_amd.ops.montecarlo import mc_paths_cpu, mc_paths_gpu

    A, n_steps, n_paths = 8, 8, 4096
    rng = np.random.default_rng(3)
    corr = np.full((A, A), 0.3) + 0.7 * np.eye(A)
    chol = np.linalg.cholesky(corr)
    mu = rng.uniform(0.0, 0.2, A)
    sigma = rng.uniform(0.2, 0.8, A)
    w = np.full(A, 1.0 / A)
    ref_fv, ref_dd = mc_paths_cpu(
        chol, mu, sigma, w, n_steps=n_steps, n_paths=n_paths,
        dt=1 / 252, seed=9,
    )
    fv, dd = mc_paths_gpu(
        chol, mu, sigma, w, n_steps=n_steps, n_paths=n_paths,
        dt=1 / 252, seed=9, device=dev,
    )
    torch.cuda.synchronize()
    np.testing.assert_allclose(fv.cpu().numpy(), ref_fv, rtol=5e-4)
    np.testing.assert_allclose(dd.cpu().numpy(), ref_dd, rtol=5e-3,
                               atol=1e-5)


def test_mc_gpu_analytic_moments(dev):
    from ai_crypto_trader_amd.ops.montecarlo import mc_paths_gpu, risk_stats

    A, n_steps, n_paths = 64, 32, 400_000
    rho = 0.4
    corr = np.full((A, A), rho) + (1 - rho) * np.eye(A)
    chol = np.linalg.cholesky(corr)
    mu = np.full(A, 0.1)
    sigma = np.full(A, 0.5)
    w = np.full(A, 1.0 / A)
    dt = 1 / 252
    fv, dd = mc_paths_gpu(
        chol, mu, sigma, w, n_steps=n_steps, n_paths=n_paths, dt=dt,
        seed=11, device=dev,
    )
    torch.cuda.synchronize()
    horizon = n_steps * dt
    assert abs(float(fv.mean()) - np.exp(0.1 * horizon)) < 0.005
    sig_p = 0.5 * np.sqrt((1 + (A - 1) * rho) / A)
    assert abs(float(torch.log(fv).std()) - sig_p * np.sqrt(horizon)) < 0.005
    s = risk_stats(fv, v0=1.0)
    assert s["cvar_95"] > s["var_95"] > 0
    assert (dd >= 0).all()


def test_cov_gpu_matches_cpu(dev):
    from ai_crypto_trader_amd.ops.covar import cov_cpu, cov_gpu

    rng = np.random.default_rng(4)
    X = (rng.standard_normal((20_000, 64)) * 0.01).astype(np.float32)
    ref = cov_cpu(X)
    got = cov_gpu(torch.from_numpy(X).to(dev))
    torch.cuda.synchronize()
    np.testing.assert_allclose(got.cpu().numpy(), ref, rtol=1e-3,
                               atol=1e-10)


def test_cov_gpu_odd_T_and_small_N(dev):
    from ai_crypto_trader_amd.ops.covar import cov_cpu, cov_gpu

    rng = np.random.default_rng(6)
    X = (rng.standard_normal((1037, 16)) * 0.02).astype(np.float32)
    got = cov_gpu(torch.from_numpy(X).to(dev))
    torch.cuda.synchronize()
    np.testing.assert_allclose(got.cpu().numpy(), cov_cpu(X), rtol=1e-3,
                               atol=1e-10)


def test_indicators_gpu_matches_cpu(dev, small_market):
    from ai_crypto_trader_amd.ops.indicators import (
        indicators_cpu, indicators_gpu,
    )

    ref = indicators_cpu(small_market)
    got = indicators_gpu(torch.from_numpy(small_market).to(dev))
    torch.cuda.synchronize()
    got = got.cpu().numpy()
    # chunk-warmup reconvergence: equal to sequential within f32 noise
    np.testing.assert_allclose(got, ref, rtol=1e-3, atol=2e-4)


def test_ga_gpu_invariants(dev):
    from ai_crypto_trader_amd.backtesting.strategy import (
        PARAM_BOUNDS, random_population,
    )
    from ai_crypto_trader_amd.ops.ga import ga_evolve_gpu, population_diversity

    P = 512
    pop = torch.from_numpy(random_population(P, seed=1)).to(dev)
    g = torch.Generator(device="cpu").manual_seed(0)
    fitness = torch.randn(P, generator=g).float().to(dev)
    child = ga_evolve_gpu(pop, fitness, elite_k=8, seed=3, gen=0)
    torch.cuda.synchronize()
    child_np = child.cpu().numpy()
    order = torch.argsort(fitness, descending=True)[:8]
    np.testing.assert_array_equal(
        child_np[:8], pop[order].cpu().numpy())        # elitism
    lo, hi = PARAM_BOUNDS[:, 0], PARAM_BOUNDS[:, 1]
    assert (child_np >= lo - 1e-5).all()
    assert (child_np <= hi + 1e-5).all()
    assert (child_np[:, 4] > child_np[:, 3]).all()
    assert population_diversity(child_np) > 0.001
    # determinism
    child2 = ga_evolve_gpu(pop, fitness, elite_k=8, seed=3, gen=0)
    torch.cuda.synchronize()
    np.testing.assert_array_equal(child_np, child2.cpu().numpy())


def test_mfma_bf16_layout(dev):
    """Validates the 16x16x32 bf16 fragment mapping (lstm.hip) against
    torch.matmul with asymmetric inputs."""
    from ai_crypto_trader_amd.ops import require_hip_ops
    ops = require_hip_ops()
    M, N, K = 32, 48, 64
    g = torch.Generator(device="cpu").manual_seed(1)
    A = torch.randn(M, K, generator=g).bfloat16()
    B = (torch.randn(K, N, generator=g) + torch.arange(N) / N).bfloat16()
    Ad, Bd = A.to(dev), B.to(dev)
    C = torch.empty(M, N, device=dev, dtype=torch.float32)
    stream = torch.cuda.current_stream(dev).cuda_stream
    ops.mfma_gemm_test_bf16(Ad.data_ptr(), Bd.data_ptr(), C.data_ptr(),
                            M, N, K, stream)
    torch.cuda.synchronize()
    ref = (A.float() @ B.float())
    torch.testing.assert_close(C.cpu(), ref, rtol=2e-2, atol=2e-2)


@pytest.mark.parametrize("H", [64, 32])
def test_lstm_fwd_matches_reference(dev, H):
    from ai_crypto_trader_amd.models.lstm import FusedLSTMLayer

    torch.manual_seed(0)
    T, B, F = 20, 128, 9
    layer = FusedLSTMLayer(F, H)
    x = torch.randn(T, B, F)
    ref = layer._forward_reference(x)           # fp32 CPU reference
    layer_g = layer.to(dev)
    out = layer_g(x.to(dev))
    torch.cuda.synchronize()
    assert out.dtype == torch.bfloat16
    torch.testing.assert_close(
        out.float().cpu(), ref, rtol=5e-2, atol=5e-2
    )


def test_lstm_bwd_matches_reference(dev):
    from ai_crypto_trader_amd.models.lstm import FusedLSTMLayer

    torch.manual_seed(1)
    T, B, F, H = 12, 64, 9, 64
    layer_ref = FusedLSTMLayer(F, H)
    layer_gpu = FusedLSTMLayer(F, H)
    layer_gpu.load_state_dict(layer_ref.state_dict())
    layer_gpu = layer_gpu.to(dev)

    x = torch.randn(T, B, F, requires_grad=True)
    out_ref = layer_ref._forward_reference(x)
    loss_ref = (out_ref ** 2).mean()
    loss_ref.backward()

    xg = x.detach().clone().to(dev).requires_grad_(True)
    out_g = layer_gpu(xg)
    loss_g = (out_g.float() ** 2).mean()
    loss_g.backward()
    torch.cuda.synchronize()

    def rel(a, b):
        return (a - b).abs().max() / (b.abs().max() + 1e-8)

    assert rel(xg.grad.cpu().float(), x.grad) < 0.08
    assert rel(layer_gpu.w_hh.grad.cpu().float(), layer_ref.w_hh.grad) < 0.08
    assert rel(layer_gpu.w_ih.grad.cpu().float(), layer_ref.w_ih.grad) < 0.08
    assert rel(layer_gpu.b_hh.grad.cpu().float(), layer_ref.b_hh.grad) < 0.08


def test_lstm_predictor_trains(dev):
    """Full model: one forward+backward+opt step reduces loss on a toy
    regression (flagship model of neural_network_service parity)."""
    from ai_crypto_trader_amd.models.lstm import LSTMPricePredictor

    torch.manual_seed(2)
    model = LSTMPricePredictor(n_features=9, seq_len=30).to(dev)
    opt = torch.optim.Adam(model.parameters(), lr=3e-3)
    B = 256
    x = torch.randn(B, 30, 9, device=dev)
    y = x[:, -5:, 0].mean(dim=1)
    losses = []
    for _ in range(30):
        opt.zero_grad()
        pred = model(x)
        loss = ((pred - y) ** 2).mean()
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    assert losses[-1] < losses[0] * 0.7, losses[::10]


def test_env_gpu_matches_cpu(dev):
    """HIP env kernel vs the numpy reference: same Philox resets, same
    per-step state machine."""
    from ai_crypto_trader_amd.data.synthetic import (
        candles_chl_v, generate_ohlcv,
    )
    from ai_crypto_trader_amd.models.rl import TradingVecEnv, TradingVecEnvCPU

    market = candles_chl_v(generate_ohlcv(3000, 4, seed=17))
    env_g = TradingVecEnv(torch.from_numpy(market).to(dev), n_envs=8,
                          ep_len=128, seed=5)
    env_c = TradingVecEnvCPU(market, n_envs=8, ep_len=128, seed=5)
    obs_g = env_g.reset().cpu().numpy()
    obs_c = env_c.reset()
    np.testing.assert_allclose(obs_g, obs_c, rtol=1e-4, atol=1e-5)
    rng = np.random.default_rng(1)
    for i in range(150):
        a = rng.integers(0, 3, 8)
        og, rg, dg = env_g.step(torch.from_numpy(a).to(dev))
        oc, rc, dc = env_c.step(a)
        np.testing.assert_allclose(rg.cpu().numpy(), rc, rtol=1e-3,
                                   atol=1e-5, err_msg=f"step {i}")
        np.testing.assert_array_equal(dg.cpu().numpy(), dc)
        np.testing.assert_allclose(og.cpu().numpy(), oc, rtol=2e-3,
                                   atol=1e-3, err_msg=f"step {i}")


def test_gae_gpu_matches_reference(dev):
    from ai_crypto_trader_amd.models.rl import gae_gpu, gae_reference

    torch.manual_seed(3)
    T, E = 128, 256
    rew = torch.randn(T, E, device=dev)
    val = torch.randn(T + 1, E, device=dev)
    dones = (torch.rand(T, E, device=dev) < 0.05).float()
    adv, ret = gae_gpu(rew, val, dones)
    torch.cuda.synchronize()
    adv_r, ret_r = gae_reference(rew, val, dones)
    torch.testing.assert_close(adv, adv_r, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(ret, ret_r, rtol=1e-4, atol=1e-5)


def test_ppo_gpu_trains(dev):
    """PPO over 256 HIP envs: one full train_step runs and produces finite
    losses (flagship RL path, BASELINE config #4)."""
    from ai_crypto_trader_amd.data.synthetic import (
        candles_chl_v, generate_ohlcv,
    )
    from ai_crypto_trader_amd.models.rl import PPOAgent, TradingVecEnv

    market = candles_chl_v(generate_ohlcv(20_000, 8, seed=19))
    env = TradingVecEnv(torch.from_numpy(market).to(dev), n_envs=256,
                        ep_len=1024, seed=7)
    env.reset()
    agent = PPOAgent(dev, seed=1)
    stats = agent.train_step(env, horizon=64)
    torch.cuda.synchronize()
    assert np.isfinite(stats["pi_loss"]) and np.isfinite(stats["v_loss"])


@pytest.mark.parametrize("H", [64, 32])
def test_gru_fwd_matches_reference(dev, H):
    from ai_crypto_trader_amd.models.gru import FusedGRULayer

    torch.manual_seed(4)
    T, B, F = 20, 128, 9
    layer = FusedGRULayer(F, H)
    x = torch.randn(T, B, F)
    ref = layer._forward_reference(x)
    out = layer.to(dev)(x.to(dev))
    torch.cuda.synchronize()
    torch.testing.assert_close(out.float().cpu(), ref, rtol=5e-2,
                               atol=5e-2)


def test_gru_bwd_matches_reference(dev):
    from ai_crypto_trader_amd.models.gru import FusedGRULayer

    torch.manual_seed(5)
    T, B, F, H = 12, 64, 9, 64
    ref_l = FusedGRULayer(F, H)
    gpu_l = FusedGRULayer(F, H)
    gpu_l.load_state_dict(ref_l.state_dict())
    gpu_l = gpu_l.to(dev)
    x = torch.randn(T, B, F, requires_grad=True)
    loss_ref = (ref_l._forward_reference(x) ** 2).mean()
    loss_ref.backward()
    xg = x.detach().clone().to(dev).requires_grad_(True)
    loss_g = (gpu_l(xg).float() ** 2).mean()
    loss_g.backward()
    torch.cuda.synchronize()

    def rel(a, b):
        return (a - b).abs().max() / (b.abs().max() + 1e-8)

    assert rel(xg.grad.cpu().float(), x.grad) < 0.08
    assert rel(gpu_l.w_hh.grad.cpu().float(), ref_l.w_hh.grad) < 0.08
    assert rel(gpu_l.w_ih.grad.cpu().float(), ref_l.w_ih.grad) < 0.08


def test_model_zoo_trains_on_gpu(dev):
    """Every zoo model does a forward+backward step on cuda."""
    from ai_crypto_trader_amd.models.zoo import MODEL_TYPES, create_model

    torch.manual_seed(6)
    x = torch.randn(64, 20, 9, device=dev)
    for mt in MODEL_TYPES:
        model = create_model(mt).to(dev)
        out = model(x)
        if mt == "probabilistic":
            loss = (out[0].sum() + out[1].sum())
        elif mt == "multitask":
            loss = out.sum()
        else:
            loss = out.sum()
        loss.backward()
        torch.cuda.synchronize()
        assert torch.isfinite(loss), mt


def test_mc_mfma_matches_valu_kernel(dev):
    """bf16 MFMA pathgen vs the exact f32 VALU kernel on identical Philox
    draws: per-path values within bf16 quantization, aggregate statistics
    much tighter."""
    from ai_crypto_trader_amd.ops.montecarlo import mc_paths_gpu

    A, n_steps, n_paths = 64, 30, 256 * 64
    rho = 0.4
    corr = np.full((A, A), rho) + (1 - rho) * np.eye(A)
    chol = np.linalg.cholesky(corr)
    mu = np.full(A, 0.1)
    sigma = np.full(A, 0.5)
    w = np.full(A, 1.0 / A)
    kw = dict(n_steps=n_steps, n_paths=n_paths, dt=1 / 252, seed=31,
              device=dev)
    fv_m, dd_m = mc_paths_gpu(chol, mu, sigma, w, use_mfma=True, **kw)
    fv_v, dd_v = mc_paths_gpu(chol, mu, sigma, w, use_mfma=False, **kw)
    torch.cuda.synchronize()
    rel = (fv_m - fv_v).abs() / fv_v.abs()
    assert float(rel.median()) < 2e-2          # per-path bf16-level
    assert abs(float(fv_m.mean() - fv_v.mean())) < 2e-3
    assert abs(float(fv_m.std() - fv_v.std())) < 2e-3
    assert abs(float(dd_m.mean() - dd_v.mean())) < 5e-3


def test_attn_fwd_matches_sdpa(dev):
    """Fused attention kernel vs torch SDPA fp32 (asymmetric inputs)."""
    from ai_crypto_trader_amd.models.attention import attn_fwd_hip

    torch.manual_seed(7)
    for S, D in ((60, 16), (64, 32), (33, 64)):
        BH = 32
        q = torch.randn(BH, S, D, device=dev)
        k = torch.randn(BH, S, D, device=dev) * 0.7 + 0.1
        v = torch.randn(BH, S, D, device=dev)
        o = attn_fwd_hip(q, k, v)
        torch.cuda.synchronize()
        ref = torch.nn.functional.scaled_dot_product_attention(
            q.float(), k.float(), v.float())
        torch.testing.assert_close(o.float(), ref, rtol=3e-2, atol=3e-2)


def test_fused_attention_backward(dev):
    """Recompute backward: grads match torch autograd on the same math."""
    from ai_crypto_trader_amd.models.attention import fused_attention

    torch.manual_seed(8)
    B, H, S, D = 4, 4, 60, 16
    q = torch.randn(B, H, S, D, device=dev, requires_grad=True)
    k = torch.randn(B, H, S, D, device=dev, requires_grad=True)
    v = torch.randn(B, H, S, D, device=dev, requires_grad=True)
    o = fused_attention(q, k, v)
    o.float().pow(2).mean().backward()
    torch.cuda.synchronize()

    q2 = q.detach().clone().requires_grad_(True)
    k2 = k.detach().clone().requires_grad_(True)
    v2 = v.detach().clone().requires_grad_(True)
    ref = torch.nn.functional.scaled_dot_product_attention(
        q2.reshape(B * H, S, D), k2.reshape(B * H, S, D),
        v2.reshape(B * H, S, D)).reshape(B, H, S, D)
    ref.float().pow(2).mean().backward()

    def rel(a, b):
        return (a - b).abs().max() / (b.abs().max() + 1e-9)

    # forward value is bf16-kernel; grads are fp32 recompute -> tight-ish
    assert rel(q.grad, q2.grad) < 0.05
    assert rel(k.grad, k2.grad) < 0.05
    assert rel(v.grad, v2.grad) < 0.05


def test_ppo_graph_rollout(dev):
    """hipGraph-captured rollout: trains, produces finite losses, and
    replays (2nd step uses graph.replay)."""
    from ai_crypto_trader_amd.data.synthetic import (
        candles_chl_v, generate_ohlcv,
    )
    from ai_crypto_trader_amd.models.rl import PPOAgent, TradingVecEnv

    market = candles_chl_v(generate_ohlcv(30_000, 4, seed=41))
    env = TradingVecEnv(torch.from_numpy(market).to(dev), n_envs=256,
                        ep_len=512, seed=9)
    env.reset()
    agent = PPOAgent(dev, seed=2, use_graph=True)
    s1 = agent.train_step(env, horizon=32)
    assert agent.use_graph, "graph capture fell back to eager"
    assert agent._graph is not None
    s2 = agent.train_step(env, horizon=32)     # replay path
    torch.cuda.synchronize()
    for s in (s1, s2):
        assert np.isfinite(s["pi_loss"]) and np.isfinite(s["v_loss"])
    # buffers actually advance between replays (env state moves)
    assert float(env.state[:, 1].min()) > 0


def test_lagged_corr_gpu(dev):
    from ai_crypto_trader_amd.ops.social_corr import (
        lagged_corr_gpu, lagged_pearson_cpu, rank_transform_gpu,
    )

    rng = np.random.default_rng(51)
    n, L, lead = 4000, 24, 5
    b = rng.standard_normal(n).astype(np.float32)
    a = np.roll(b, lead) * 0.8 + rng.standard_normal(n).astype(
        np.float32) * 0.3
    # a[i] ~ b[i-lead]: pairing a[i]~b[i+k] peaks at k = -lead
    ref = lagged_pearson_cpu(a, b, L)
    got = lagged_corr_gpu(torch.from_numpy(a).to(dev),
                          torch.from_numpy(b).to(dev), L)
    torch.cuda.synchronize()
    np.testing.assert_allclose(got.cpu().numpy(), ref, rtol=1e-3,
                               atol=1e-4)
    assert int(got.argmax()) == L - lead       # peak at the known lag
    # spearman path via rank transform
    ra = rank_transform_gpu(torch.from_numpy(a).to(dev))
    rb = rank_transform_gpu(torch.from_numpy(b).to(dev))
    sp = lagged_corr_gpu(ra, rb, L)
    torch.cuda.synchronize()
    assert int(sp.argmax()) == L - lead


def test_vp_hist_gpu(dev):
    """Volume-profile histogram kernel vs numpy histogram."""
    from ai_crypto_trader_amd.data.synthetic import (
        candles_chl_v, generate_ohlcv,
    )
    from ai_crypto_trader_amd.utils.volume_profile import vp_hist_gpu

    market = candles_chl_v(generate_ohlcv(50_000, 4, seed=61))
    c = torch.from_numpy(market).to(dev)
    hist, updown, lo, hi = vp_hist_gpu(c, n_bins=32)
    torch.cuda.synchronize()
    for s in range(4):
        ref, _ = np.histogram(
            market[s, :, 0], bins=32,
            range=(float(lo[s]), float(hi[s])),
            weights=market[s, :, 3])
        # bin-EDGE float rounding differs between the kernel's
        # (x-lo)*inv_w and numpy's edge array: a candle exactly on an
        # edge may land one bin over (<0.1% of a bin's volume)
        np.testing.assert_allclose(hist[s].cpu().numpy(), ref, rtol=2e-3,
                                   atol=1.0)
        close = market[s, :, 0]
        vol = market[s, :, 3]
        up = vol[1:][close[1:] >= close[:-1]].sum()
        dn = vol[1:][close[1:] < close[:-1]].sum()
        np.testing.assert_allclose(updown[s].cpu().numpy(), [up, dn],
                                   rtol=1e-4)


def test_ga_engine_memory_stable(dev):
    """50 GA generations allocate no net GPU memory after warmup — the
    driver's scaling runs and the evolution service loop depend on a flat
    allocator profile."""
    from ai_crypto_trader_amd.backtesting.ga_engine import GAEngine
    from ai_crypto_trader_amd.data.synthetic import (
        candles_chl_v, generate_ohlcv,
    )

    candles = candles_chl_v(generate_ohlcv(100_000, 4, seed=1))
    eng = GAEngine(candles, pop_per_rank=256, device=dev, seed=2,
                   segments=4)
    for _ in range(5):
        eng.step()
    torch.cuda.synchronize()
    base = torch.cuda.memory_allocated()
    for _ in range(45):
        eng.step()
    torch.cuda.synchronize()
    assert torch.cuda.memory_allocated() <= base + (1 << 20)


def test_ppo_graphed_memory_stable(dev):
    from ai_crypto_trader_amd.data.synthetic import (
        candles_chl_v, generate_ohlcv,
    )
    from ai_crypto_trader_amd.models.rl import PPOAgent, TradingVecEnv

    market = torch.from_numpy(
        candles_chl_v(generate_ohlcv(100_000, 4, seed=2))).to(dev)
    env = TradingVecEnv(market, n_envs=128, ep_len=512, seed=0)
    env.reset()
    agent = PPOAgent(dev, seed=0, use_graph=True)
    for _ in range(3):
        agent.train_step(env, horizon=32)
    torch.cuda.synchronize()
    base = torch.cuda.memory_allocated()
    for _ in range(10):
        agent.train_step(env, horizon=32)
    torch.cuda.synchronize()
    assert agent.use_graph                      # graphs actually engaged
    assert torch.cuda.memory_allocated() <= base + (1 << 20)


@pytest.mark.parametrize("seed", [101, 202, 303, 404, 505])
def test_backtest_bitwise_parity_fuzz(dev, seed):
    """Multi-seed stress of the bitwise CPU/GPU decision parity: fresh
    random market + population per seed, trade/win counts must agree
    EXACTLY on every (strategy, symbol)."""
    from ai_crypto_trader_amd.backtesting.engine_cpu import run_backtest_cpu
    from ai_crypto_trader_amd.backtesting.strategy import random_population
    from ai_crypto_trader_amd.data.synthetic import (
        candles_chl_v, generate_ohlcv,
    )
    from ai_crypto_trader_amd.ops.backtest import run_backtest_gpu

    market = candles_chl_v(
        generate_ohlcv(3000, 3, seed=seed, sigma=1.0 + (seed % 3)))
    pop = random_population(96, seed=seed + 7)
    ref = run_backtest_cpu(market, pop)
    got = run_backtest_gpu(
        torch.from_numpy(market).to(dev), torch.from_numpy(pop).to(dev))
    torch.cuda.synchronize()
    got = got.cpu().numpy()
    np.testing.assert_array_equal(got[..., 1], ref[..., 1])
    np.testing.assert_array_equal(got[..., 2], ref[..., 2])
    np.testing.assert_allclose(got[..., 0], ref[..., 0], rtol=2e-5)


def test_mc_antithetic_pairing(dev):
    """antithetic=True: the top half of the path range mirrors the bottom
    half's normals negated — pairs are strongly anticorrelated, the mean
    estimate matches iid, and the mean-estimator variance shrinks."""
    from ai_crypto_trader_amd.ops.montecarlo import mc_paths_gpu

    A, n_steps, n_paths = 8, 16, 40_000
    rng = np.random.default_rng(5)
    corr = np.full((A, A), 0.3) + 0.7 * np.eye(A)
    chol = np.linalg.cholesky(corr)
    mu = np.full(A, 0.2)
    sigma = np.full(A, 0.6)
    w = np.full(A, 1.0 / A)
    kw = dict(n_steps=n_steps, n_paths=n_paths, dt=1 / 252, seed=17,
              device=dev)
    fv_iid, _ = mc_paths_gpu(chol, mu, sigma, w, **kw)
    fv_anti, _ = mc_paths_gpu(chol, mu, sigma, w, antithetic=True, **kw)
    torch.cuda.synchronize()
    half = n_paths // 2
    a = fv_anti[:half].double()
    b = fv_anti[half:].double()
    # bottom half identical to the iid run (same Philox counters)
    np.testing.assert_array_equal(fv_anti[:half].cpu().numpy(),
                                  fv_iid[:half].cpu().numpy())
    # mirrored pairs anticorrelated
    rho = float(((a - a.mean()) * (b - b.mean())).mean()
                / (a.std() * b.std()))
    assert rho < -0.5, rho
    # unbiased: means agree within a few std errors
    se = float(fv_iid.std() / np.sqrt(n_paths))
    assert abs(float(fv_anti.mean()) - float(fv_iid.mean())) < 5 * se
    # the paired-mean variance is lower than the iid pair variance
    var_anti = float(((a + b) / 2).var())
    pair_iid = (fv_iid[:half].double() + fv_iid[half:].double()) / 2
    assert var_anti < float(pair_iid.var())


def test_mc_bootstrap_matches_cpu(dev):
    """Historical-bootstrap kernel vs the numpy twin: identical Philox
    index stream, same resampled rows, equal values within f32/exp2
    rounding; empirical moments track the resampled distribution."""
    from ai_crypto_trader_amd.ops.montecarlo import (
        mc_bootstrap_cpu, mc_bootstrap_gpu,
    )

    rng = np.random.default_rng(9)
    T_hist, A = 300, 8
    lr = (rng.standard_normal((T_hist, A)) * 0.01 + 0.0004).astype(
        np.float32)
    w = np.full(A, 1.0 / A)
    n_steps, n_paths = 20, 8192
    ref_fv, ref_dd = mc_bootstrap_cpu(lr, w, n_steps=n_steps,
                                      n_paths=n_paths, seed=4)
    fv, dd = mc_bootstrap_gpu(lr, w, n_steps=n_steps, n_paths=n_paths,
                              seed=4, device=dev)
    torch.cuda.synchronize()
    np.testing.assert_allclose(fv.cpu().numpy(), ref_fv, rtol=5e-4)
    np.testing.assert_allclose(dd.cpu().numpy(), ref_dd, rtol=5e-3,
                               atol=1e-5)
    # sanity: mean final value ~ exp of n_steps * mean log-return
    expect = float(np.exp(n_steps * lr.mean() * A / A))
    assert abs(float(fv.mean()) - expect) < 0.05


def test_gru_infer_matches_training_forward(dev):
    """The no-save GRU inference forward produces the same hidden states
    as the training forward (mirrors the LSTM serving path)."""
    from ai_crypto_trader_amd.models.gru import FusedGRULayer

    torch.manual_seed(3)
    layer = FusedGRULayer(9, 64).to(dev)
    x = torch.randn(24, 128, 9, device=dev)
    with torch.enable_grad():
        h_train = layer(x)
    with torch.no_grad():
        h_infer = layer(x)
    torch.cuda.synchronize()
    np.testing.assert_array_equal(h_train.detach().cpu().float().numpy(),
                                  h_infer.cpu().float().numpy())
