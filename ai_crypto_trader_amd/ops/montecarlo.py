"""Monte-Carlo correlated-GBM ops: GPU wrapper + CPU golden reference.

Replaces monte_carlo_service.py:197-336: GBM paths, percentiles, VaR/CVaR,
probability of profit, per-path max drawdown. The CPU reference
reimplements the exact Philox4x32-7 + Box-Muller stream of the kernel in
vectorized numpy so small-scale tests are bit-comparable (modulo
transcendental rounding), and an analytic-moment test covers the large-N
statistics.
"""

from __future__ import annotations

import numpy as np

from . import require_hip_ops

# --- Philox4x32-7 in numpy (matches ops/hip/common.hpp) --------------------
# 7 rounds = the BigCrush-passing count from the Random123 paper; must stay
# equal to PHILOX_ROUNDS in ops/hip/common.hpp (bit-reproducibility tests
# compare the two streams directly).

_PHILOX_ROUNDS = 7
_M0 = np.uint64(0xD2511F53)
_M1 = np.uint64(0xCD9E8D57)
_W0 = np.uint32(0x9E3779B9)
_W1 = np.uint32(0xBB67AE85)


def philox4x32_np(seed: int, ctr_lo: np.ndarray, ctr_hi: np.ndarray):
    """Vectorized Philox4x32-7. ctr_lo/ctr_hi: uint64 arrays."""
    u32 = np.uint32
    u64 = np.uint64
    k0 = u32(seed & 0xFFFFFFFF)
    k1 = u32((seed >> 32) & 0xFFFFFFFF)
    ctr_lo = ctr_lo.astype(u64)
    ctr_hi = ctr_hi.astype(u64)
    c0 = ctr_lo.astype(u32)
    c1 = (ctr_lo >> u64(32)).astype(u32)
    c2 = ctr_hi.astype(u32)
    c3 = (ctr_hi >> u64(32)).astype(u32)
    k0 = np.full_like(c0, k0)
    k1 = np.full_like(c0, k1)
    for _ in range(_PHILOX_ROUNDS):
        p0 = _M0 * c0.astype(u64)
        p1 = _M1 * c2.astype(u64)
        hi0 = (p0 >> u64(32)).astype(u32)
        lo0 = p0.astype(u32)
        hi1 = (p1 >> u64(32)).astype(u32)
        lo1 = p1.astype(u32)
        n0 = hi1 ^ c1 ^ k0
        n1 = lo1
        n2 = hi0 ^ c3 ^ k1
        n3 = lo0
        c0, c1, c2, c3 = n0, n1, n2, n3
        k0 = k0 + _W0
        k1 = k1 + _W1
    return c0, c1, c2, c3


def _u32_to_unit(v: np.ndarray) -> np.ndarray:
    # matches device: ((float)v + 1.0f) * 2.3283064e-10f  (f32 throughout)
    return (v.astype(np.float32) + np.float32(1.0)) * np.float32(2.3283064e-10)


def philox_normal4_np(seed: int, ctr_lo: np.ndarray, ctr_hi: np.ndarray):
    """Matches device philox_normal4: returns (..., 4) float32 normals."""
    a, b, c, d = philox4x32_np(seed, ctr_lo, ctr_hi)

    def bm(x, y):
        u1 = _u32_to_unit(x).astype(np.float32)
        u2 = _u32_to_unit(y).astype(np.float32)
        r = np.sqrt(np.float32(-2.0) * np.log(u1))
        ang = np.float32(2.0 * np.pi) * u2
        return r * np.cos(ang), r * np.sin(ang)

    z0, z1 = bm(a, b)
    z2, z3 = bm(c, d)
    return np.stack([z0, z1, z2, z3], axis=-1).astype(np.float32)


# --- shared GBM term builder (CPU reference AND GPU wrapper) ---------------

_LOG2E = np.log2(np.e)     # f64; kernels work in base-2 log space


def _gbm_terms(chol, mu, sigma, dt):
    """(cvol_k_major, drift, both f32 and pre-scaled by log2(e)).

    The kernels accumulate logS in BASE-2 so the per-asset exponential is
    a single native v_exp_f32 (exp2) — the log2(e) factor is folded into
    drift and cvol here, identically for the CPU reference and the GPU
    wrapper so the two streams stay comparable."""
    f32 = np.float32
    chol = np.asarray(chol)
    mu = np.asarray(mu)
    sigma = np.asarray(sigma)
    cvol = sigma[:, None] * chol * np.sqrt(dt) * _LOG2E          # (a, k)
    cvol_k_major = np.ascontiguousarray(cvol.T, dtype=f32)       # (k, a)
    drift = ((mu - 0.5 * sigma**2) * dt * _LOG2E).astype(f32)
    return cvol_k_major, drift


# --- CPU reference path generator -----------------------------------------

def mc_paths_cpu(
    chol: np.ndarray,        # (A, A) Cholesky factor of the correlation
    mu: np.ndarray,          # (A,) annualized drift
    sigma: np.ndarray,       # (A,) annualized vol
    weights: np.ndarray,     # (A,) portfolio weights (sum to 1)
    *,
    n_steps: int,
    n_paths: int,
    dt: float,
    s0: float = 1.0,
    seed: int = 0,
):
    """Reference for mc_paths_kernel: same Philox stream, same update
    order. Returns (final_value, max_dd) each (n_paths,) f32."""
    A = chol.shape[0]
    f32 = np.float32
    cvol_k_major, drift = _gbm_terms(chol, mu, sigma, dt)      # base-2 log
    wS0 = (weights * s0).astype(f32)

    logS = np.zeros((n_paths, A), f32)
    v0 = f32(weights.sum() * s0)
    vmax = np.full(n_paths, v0, f32)
    mdd = np.zeros(n_paths, f32)
    V = np.full(n_paths, v0, f32)
    paths = np.arange(n_paths, dtype=np.uint64)
    for step in range(n_steps):
        for k4 in range(A // 4):
            ctr_hi = np.full(
                n_paths, (np.uint64(step) << np.uint64(32)) | np.uint64(k4),
                dtype=np.uint64,
            )
            z4 = philox_normal4_np(seed, paths, ctr_hi)        # (P, 4)
            for dz in range(4):
                k = k4 * 4 + dz
                logS += np.outer(z4[:, dz], cvol_k_major[k])
        logS += drift
        V = (wS0 * np.exp2(logS)).sum(axis=1).astype(f32)
        vmax = np.maximum(vmax, V)
        mdd = np.maximum(mdd, (vmax - V) / vmax)
    return V, mdd


def mc_paths_gpu(
    chol, mu, sigma, weights, *, n_steps: int, n_paths: int, dt: float,
    s0: float = 1.0, seed: int = 0, device="cuda", use_mfma: bool | None = None,
    path_base: int = 0, antithetic: bool = False,
):
    """GPU path generation; returns (final_value, max_dd) torch tensors.

    use_mfma=None (auto): the bf16 MFMA pathgen kernel when A==64 and
    n_paths is a multiple of 256, else the exact f32 VALU kernel. The two
    draw identical Philox normals; MFMA quantizes Z/CVOL to bf16
    (statistics agree to ~1e-2, see tests).

    antithetic=True pairs the top half of the path range with the bottom
    half's normals negated (antithetic variates): an unbiased estimator
    with reduced variance for the monotone risk statistics, at half the
    RNG cost per effective path. Off for benchmarks (iid paths)."""
    import torch

    ops = require_hip_ops()
    A = int(chol.shape[0])
    if use_mfma is None:
        use_mfma = (A == 64)
    if use_mfma and A != 64:
        use_mfma = False
    pad = ((-n_paths) % 256) if use_mfma else 0   # MFMA tile = 256 paths
    f32 = np.float32
    cvol_k_major, drift = _gbm_terms(chol, mu, sigma, dt)      # base-2 log
    wS0 = (np.asarray(weights) * s0).astype(f32)
    v0 = float(wS0.sum())

    t_cvol = torch.from_numpy(cvol_k_major).to(device)
    t_drift = torch.from_numpy(drift).to(device)
    t_w = torch.from_numpy(wS0).to(device)
    fv = torch.empty(n_paths + pad, dtype=torch.float32, device=device)
    dd = torch.empty(n_paths + pad, dtype=torch.float32, device=device)
    stream = torch.cuda.current_stream(fv.device).cuda_stream
    if use_mfma:
        ops.mc_paths_mfma(
            t_cvol.data_ptr(), t_drift.data_ptr(), t_w.data_ptr(),
            fv.data_ptr(), dd.data_ptr(), A, n_steps, n_paths + pad, v0,
            seed, path_base, int(antithetic), stream,
        )
        if pad:
            fv, dd = fv[:n_paths], dd[:n_paths]
    else:
        ops.mc_paths(
            t_cvol.data_ptr(), t_drift.data_ptr(), t_w.data_ptr(), 0,
            fv.data_ptr(), dd.data_ptr(), A, n_steps, n_paths, v0, seed,
            path_base, int(antithetic), stream,
        )
    return fv, dd


def mc_bootstrap_cpu(log_returns, weights, *, n_steps: int, n_paths: int,
                     s0: float = 1.0, seed: int = 0):
    """CPU twin of mc_bootstrap_kernel (reference 'historical' method,
    monte_carlo_service.py:275-298): paths resample historical log-return
    ROWS with replacement (joint resampling preserves the empirical
    cross-asset correlation). Same Philox index stream as the kernel."""
    f32 = np.float32
    lr = np.asarray(log_returns, f32)           # (T_hist, A) natural logs
    lr2 = (lr * np.float32(_LOG2E)).astype(f32)  # base-2 like the kernel
    T_hist, A = lr2.shape
    wS0 = (np.asarray(weights) * s0).astype(f32)
    v0 = f32(wS0.sum())
    logS = np.zeros((n_paths, A), f32)
    vmax = np.full(n_paths, v0, f32)
    mdd = np.zeros(n_paths, f32)
    V = np.full(n_paths, v0, f32)
    paths = np.arange(n_paths, dtype=np.uint64)
    for s4 in range((n_steps + 3) // 4):
        ctr_hi = np.full(n_paths,
                         np.uint64(0x8000000000000000) | np.uint64(s4),
                         np.uint64)
        u = philox4x32_np(seed, paths, ctr_hi)       # 4 x (P,) uint32
        for j in range(4):
            step = s4 * 4 + j
            if step >= n_steps:
                break
            t = (u[j] % np.uint32(T_hist)).astype(np.int64)
            logS += lr2[t]
            V = (wS0 * np.exp2(logS)).sum(axis=1).astype(f32)
            vmax = np.maximum(vmax, V)
            mdd = np.maximum(mdd, (vmax - V) / vmax)
    return V, mdd


def mc_bootstrap_gpu(log_returns, weights, *, n_steps: int, n_paths: int,
                     s0: float = 1.0, seed: int = 0, device="cuda",
                     path_base: int = 0):
    """GPU historical-bootstrap paths; returns (final_value, max_dd)."""
    import torch

    ops = require_hip_ops()
    f32 = np.float32
    lr = np.asarray(log_returns, f32)
    lr2 = np.ascontiguousarray(lr * np.float32(_LOG2E), dtype=f32)
    T_hist, A = lr2.shape
    wS0 = (np.asarray(weights) * s0).astype(f32)
    v0 = float(wS0.sum())
    t_lr = torch.from_numpy(lr2).to(device)
    t_w = torch.from_numpy(wS0).to(device)
    fv = torch.empty(n_paths, dtype=torch.float32, device=device)
    dd = torch.empty(n_paths, dtype=torch.float32, device=device)
    stream = torch.cuda.current_stream(fv.device).cuda_stream
    ops.mc_bootstrap(t_lr.data_ptr(), t_w.data_ptr(), fv.data_ptr(),
                     dd.data_ptr(), A, T_hist, n_steps, n_paths, v0,
                     seed, path_base, stream)
    return fv, dd


def risk_stats(final_values, v0: float, confidences=(0.95, 0.99)):
    """VaR/CVaR/percentiles/prob-profit from per-path final values
    (monte_carlo_service.py:304-325 semantics). Works on numpy arrays or
    torch tensors (stays on device for torch)."""
    import torch

    is_torch = isinstance(final_values, torch.Tensor)
    fv = final_values
    n = fv.shape[0]
    sorted_fv = torch.sort(fv).values if is_torch else np.sort(fv)
    out = {}
    for conf in confidences:
        idx = int((1.0 - conf) * n)
        var_level = sorted_fv[idx]
        tail = sorted_fv[: max(idx, 1)]
        cvar_level = tail.mean()
        out[f"var_{int(conf * 100)}"] = float(v0 - var_level)
        out[f"cvar_{int(conf * 100)}"] = float(v0 - cvar_level)
    for pct in (5, 25, 50, 75, 95):
        idx = min(n - 1, int(pct / 100.0 * n))
        out[f"p{pct}"] = float(sorted_fv[idx])
    out["mean"] = float(fv.mean())
    out["prob_profit"] = float((fv > v0).sum() / n)
    return out


def mc_paths_sharded(
    chol, mu, sigma, weights, *, n_paths_total: int, n_steps: int,
    dt: float, rank: int = 0, world: int = 1, seed: int = 0, s0: float = 1.0,
    device="cuda", n_bins: int = 2048, antithetic: bool = False,
):
    """Monte-Carlo sharded across ranks (path batches DP
    across GPUs, stats via RCCL all-reduce).

    Each rank generates its contiguous PATH-INDEX slice — the Philox
    counter is the global path id, so the union over ranks is bit-identical
    to a single-GPU run of n_paths_total. Global statistics come from
    collectives only (no gather of the 10M-path vectors):
      - mean/std: all-reduced sums
      - VaR/CVaR/percentiles: an all-reduced fixed-range histogram
        (n_bins bins => quantile resolution ~ range/n_bins)
    Returns (local_fv, local_dd, global_stats dict).
    """
    import torch

    from ..parallel import dist as pdist

    per = (n_paths_total + world - 1) // world
    lo = rank * per
    hi = min(lo + per, n_paths_total)
    n_local = hi - lo
    # the Philox counter is the GLOBAL path id: each rank passes its slice
    # start as path_base, so the union over ranks is bit-identical to a
    # single-GPU run and no rank generates another rank's paths.
    if device == "cpu":
        fv_np, dd_np = mc_paths_cpu(
            chol, mu, sigma, weights, n_steps=n_steps,
            n_paths=n_paths_total, dt=dt, s0=s0, seed=seed)
        fv = torch.from_numpy(fv_np[lo:hi])
        dd = torch.from_numpy(dd_np[lo:hi])
    else:
        fv, dd = mc_paths_gpu(
            chol, mu, sigma, weights, n_steps=n_steps, n_paths=n_local,
            dt=dt, s0=s0, seed=seed, device=device, path_base=lo,
            antithetic=antithetic)    # pairs mirror within each shard

    v0 = float(np.sum(np.asarray(weights) * s0))
    # all-reduced moments
    sums = torch.stack([
        fv.sum(), (fv.double() ** 2).sum().float(), dd.sum(),
        torch.tensor(float(n_local), device=fv.device),
    ])
    pdist.all_reduce_sum_(sums)
    n = float(sums[3])
    mean = float(sums[0]) / n
    var = max(float(sums[1]) / n - mean * mean, 0.0)
    # all-reduced histogram for quantiles
    fmin = torch.tensor(float(fv.min()), device=fv.device)
    fmax = torch.tensor(float(fv.max()), device=fv.device)
    if pdist.is_dist():
        import torch.distributed as tdist
        tdist.all_reduce(fmin, op=tdist.ReduceOp.MIN)
        tdist.all_reduce(fmax, op=tdist.ReduceOp.MAX)
    lo_v, hi_v = float(fmin), float(fmax) + 1e-9
    hist = torch.histc(fv, bins=n_bins, min=lo_v, max=hi_v)
    pdist.all_reduce_sum_(hist)
    cum = torch.cumsum(hist, 0)
    edges = torch.linspace(lo_v, hi_v, n_bins + 1, device=fv.device)

    def quantile(q):
        idx = int(torch.searchsorted(cum, torch.tensor(
            q * n, device=fv.device)))
        return float(edges[min(idx + 1, n_bins)])

    def cvar(q):
        """E[v0 - V | V <= quantile(q)] from the all-reduced histogram
        (bin midpoints weighted by counts below the q-quantile)."""
        idx = int(torch.searchsorted(cum, torch.tensor(
            q * n, device=fv.device)))
        idx = max(idx, 1)
        mids = (edges[:-1] + edges[1:]) / 2
        w_tail = hist[:idx]
        m = float(w_tail.sum())
        if m <= 0:
            return v0 - quantile(q)
        return v0 - float((mids[:idx] * w_tail).sum()) / m

    stats = {
        "mean": mean, "std": var ** 0.5,
        "var_95": v0 - quantile(0.05), "var_99": v0 - quantile(0.01),
        "cvar_95": cvar(0.05), "cvar_99": cvar(0.01),
        "p5": quantile(0.05), "p50": quantile(0.5), "p95": quantile(0.95),
        "max_drawdown_mean": float(sums[2]) / n,
        "n_paths": int(n), "world": world,
    }
    return fv, dd, stats
