"""Fused attention op + transformer encoder layer on the gfx950 kernel
(ops/hip/attention.hip).

fused_attention(): multi-head attention on the HIP kernels for S <= 64
(the predictors' shape). The forward saves the softmax matrix P when
grads are enabled; the fused backward kernel (attn_bwd) computes
dV = P^T dO, the softmax-jacobian correction, and dQ/dK on the same MFMA
fragments — no torch recompute on the training path.
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn

from ..ops import require_hip_ops


def attn_fwd_hip(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                 scale: float | None = None, save_p: bool = False):
    """q/k/v: (BH, S, D) -> (BH, S, D). bf16 in/out, S <= 64.
    save_p=True additionally returns the (BH, 64, 64) softmax matrix for
    the fused backward."""
    ops = require_hip_ops()
    bh, s, d = q.shape
    if scale is None:
        scale = 1.0 / math.sqrt(d)
    q = q.to(torch.bfloat16).contiguous()
    k = k.to(torch.bfloat16).contiguous()
    v = v.to(torch.bfloat16).contiguous()
    o = torch.empty_like(q)
    p = torch.empty((bh, 64, 64), dtype=torch.bfloat16, device=q.device) \
        if save_p else None
    stream = torch.cuda.current_stream(q.device).cuda_stream
    ops.attn_fwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                 p.data_ptr() if save_p else 0, bh, s, d, float(scale),
                 stream)
    return (o, p) if save_p else o


def attn_bwd_hip(q, k, v, p, grad_o, scale):
    """Fused backward: returns (dQ, dK, dV) bf16."""
    ops = require_hip_ops()
    bh, s, d = q.shape
    go = grad_o.to(torch.bfloat16).contiguous()
    dq = torch.empty_like(q)
    dk = torch.empty_like(k)
    dv = torch.empty_like(v)
    stream = torch.cuda.current_stream(q.device).cuda_stream
    ops.attn_bwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), p.data_ptr(),
                 go.data_ptr(), dq.data_ptr(), dk.data_ptr(),
                 dv.data_ptr(), bh, s, d, float(scale), stream)
    return dq, dk, dv


class _FusedAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale):
        need_grad = any(t.requires_grad for t in (q, k, v))
        qb = q.detach().to(torch.bfloat16).contiguous()
        kb = k.detach().to(torch.bfloat16).contiguous()
        vb = v.detach().to(torch.bfloat16).contiguous()
        if need_grad:
            o, p = attn_fwd_hip(qb, kb, vb, scale, save_p=True)
            ctx.save_for_backward(qb, kb, vb, p)
        else:
            o = attn_fwd_hip(qb, kb, vb, scale)
        ctx.scale = scale
        ctx.dtypes = (q.dtype, k.dtype, v.dtype)
        return o

    @staticmethod
    def backward(ctx, grad_o):
        q, k, v, p = ctx.saved_tensors
        dq, dk, dv = attn_bwd_hip(q, k, v, p, grad_o, ctx.scale)
        qt, kt, vt = ctx.dtypes
        return dq.to(qt), dk.to(kt), dv.to(vt), None


def fused_attention(q, k, v, scale=None):
    """(B, H, S, D) or (BH, S, D); fused HIP forward AND backward."""
    shape4 = q.dim() == 4
    if shape4:
        B, H, S, D = q.shape
        q, k, v = (t.reshape(B * H, S, D) for t in (q, k, v))
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    o = _FusedAttention.apply(q, k, v, scale)
    return o.reshape(B, H, S, D) if shape4 else o


class FusedMultiheadSelfAttention(nn.Module):
    """Self-attention block using the fused kernel on GPU (torch SDPA on
    CPU), for d_model/n_heads with d_head in {16, 32, 64} and S <= 64."""

    def __init__(self, d_model: int, n_heads: int):
        super().__init__()
        assert d_model % n_heads == 0
        self.h = n_heads
        self.dh = d_model // n_heads
        assert self.dh in (16, 32, 64)
        self.qkv = nn.Linear(d_model, 3 * d_model)
        self.proj = nn.Linear(d_model, d_model)

    def forward(self, x: torch.Tensor) -> torch.Tensor:   # (B, S, Dm)
        B, S, Dm = x.shape
        qkv = self.qkv(x).reshape(B, S, 3, self.h, self.dh)
        q, k, v = (qkv[:, :, i].transpose(1, 2) for i in range(3))
        if x.is_cuda:
            o = fused_attention(q, k, v).float()
        else:
            o = torch.nn.functional.scaled_dot_product_attention(q, k, v)
        o = o.transpose(1, 2).reshape(B, S, Dm)
        return self.proj(o.to(x.dtype))


class FusedTransformerEncoderLayer(nn.Module):
    def __init__(self, d_model: int, n_heads: int, dim_ff: int = 128):
        super().__init__()
        self.attn = FusedMultiheadSelfAttention(d_model, n_heads)
        self.ff = nn.Sequential(nn.Linear(d_model, dim_ff), nn.ReLU(),
                                nn.Linear(dim_ff, d_model))
        self.n1 = nn.LayerNorm(d_model)
        self.n2 = nn.LayerNorm(d_model)

    def forward(self, x):
        x = self.n1(x + self.attn(x))
        return self.n2(x + self.ff(x))
