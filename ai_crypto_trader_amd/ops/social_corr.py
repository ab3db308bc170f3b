"""Lead/lag lagged-correlation op: GPU kernel wrapper + numpy reference
(the math of services/social.py::SocialMetricsAnalyzer.lead_lag, SURVEY.md
§2.9 'batched lagged-correlation kernel')."""

from __future__ import annotations

import numpy as np

from . import require_hip_ops


def lagged_pearson_cpu(a: np.ndarray, b: np.ndarray,
                       max_lag: int) -> np.ndarray:
    """Pearson for lags -max_lag..max_lag; lag k>=0 pairs a[i], b[i+k]."""
    out = np.zeros(2 * max_lag + 1, np.float32)
    for j, lag in enumerate(range(-max_lag, max_lag + 1)):
        if lag >= 0:
            x, y = a[: len(a) - lag or None], b[lag:]
        else:
            x, y = a[-lag:], b[: len(b) + lag]
        m = min(len(x), len(y))
        if m < 3:
            continue
        x, y = x[:m].astype(np.float64), y[:m].astype(np.float64)
        den = x.std() * y.std()
        if den > 1e-12:
            out[j] = ((x * y).mean() - x.mean() * y.mean()) / den
    return out


def lagged_corr_gpu(a, b, max_lag: int):
    """a, b: 1-D f32 cuda tensors -> (2*max_lag+1,) pearson tensor.
    Spearman: rank-transform first (ranks = argsort of argsort) and call
    this on the ranks."""
    import torch

    ops = require_hip_ops()
    assert a.is_cuda and b.is_cuda and a.dtype == torch.float32
    n = int(min(a.numel(), b.numel()))
    a = a[:n].contiguous()
    b = b[:n].contiguous()
    out = torch.empty(2 * max_lag + 1, dtype=torch.float32,
                      device=a.device)
    stream = torch.cuda.current_stream(a.device).cuda_stream
    ops.lagged_corr(a.data_ptr(), b.data_ptr(), out.data_ptr(), n,
                    max_lag, stream)
    return out


def rank_transform_gpu(x):
    import torch

    order = torch.argsort(x)
    ranks = torch.empty_like(x)
    ranks[order] = torch.arange(len(x), dtype=x.dtype, device=x.device)
    return ranks
