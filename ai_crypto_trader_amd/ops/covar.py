"""Portfolio covariance / VaR ops: MFMA GPU kernel wrapper + CPU reference.

Replaces portfolio_risk_service.py:
  :217-247  per-asset historical VaR (percentile of daily returns x value)
  :249-284  CVaR (mean of tail beyond VaR)
  :286-326  correlation matrix
  :328-398  portfolio VaR = sqrt(w^T (vv^T o C) w) with PD check
The covariance itself is the MFMA GEMM kernel (ops/hip/covar.hip); the
O(N^2) quadratic form and Cholesky PD check run on torch (device-resident).
"""

from __future__ import annotations

import numpy as np

from . import require_hip_ops


def cov_cpu(returns: np.ndarray) -> np.ndarray:
    """Sample covariance of (T, N) returns, f32. Golden reference for the
    MFMA kernel: one-pass E[xy] - E[x]E[y] with (T-1) normalization, the
    same formula the kernel's finalize step applies."""
    X = np.asarray(returns, dtype=np.float64)
    T = X.shape[0]
    sx = X.sum(axis=0)
    sxy = X.T @ X
    cov = (sxy - np.outer(sx, sx) / T) / (T - 1)
    return cov.astype(np.float32)


def cov_gpu(returns) -> "torch.Tensor":
    """(T, N) f32 cuda tensor -> (N, N) covariance on device."""
    import torch

    ops = require_hip_ops()
    assert returns.is_cuda and returns.dtype == torch.float32
    returns = returns.contiguous()
    T, N = returns.shape
    cov = torch.empty((N, N), dtype=torch.float32, device=returns.device)
    stream = torch.cuda.current_stream(returns.device).cuda_stream
    ops.cov(returns.data_ptr(), cov.data_ptr(), T, N, stream)
    return cov


def corr_from_cov(cov):
    """Correlation matrix from covariance (numpy or torch)."""
    import torch

    if isinstance(cov, torch.Tensor):
        d = torch.sqrt(torch.clamp(torch.diagonal(cov), min=1e-18))
        return cov / d[:, None] / d[None, :]
    d = np.sqrt(np.clip(np.diagonal(cov), 1e-18, None))
    return cov / d[:, None] / d[None, :]


def portfolio_var(
    values, vols, corr, *, confidence_z: float = 1.645
) -> float:
    """sqrt(w^T (vv^T o C) w) * z — portfolio_risk_service.py:328-398:
    v = per-position VaR vector (value_i * vol_i * z), C = correlation."""
    import torch

    if isinstance(corr, torch.Tensor):
        v = (values * vols * confidence_z).to(corr.dtype)
        q = torch.einsum("i,ij,j->", v, corr, v)
        return float(torch.sqrt(torch.clamp(q, min=0.0)))
    v = np.asarray(values) * np.asarray(vols) * confidence_z
    q = v @ np.asarray(corr, dtype=np.float64) @ v
    return float(np.sqrt(max(q, 0.0)))


def is_positive_definite(corr) -> bool:
    """PD check via Cholesky (portfolio_risk_service.py:378 used eigvals;
    Cholesky is the O(N^3/3) device-friendly equivalent)."""
    import torch

    if isinstance(corr, torch.Tensor):
        try:
            torch.linalg.cholesky(corr.double())
            return True
        except Exception:
            return False
    try:
        np.linalg.cholesky(np.asarray(corr, dtype=np.float64))
        return True
    except np.linalg.LinAlgError:
        return False


def historical_var_cvar(returns_1d, value: float, confidence: float = 0.95):
    """Per-asset historical VaR/CVaR (portfolio_risk_service.py:217-284)."""
    r = np.sort(np.asarray(returns_1d, dtype=np.float64))
    idx = max(int((1.0 - confidence) * len(r)), 1)
    var = -r[idx - 1] * value
    cvar = -r[:idx].mean() * value
    return float(var), float(cvar)
