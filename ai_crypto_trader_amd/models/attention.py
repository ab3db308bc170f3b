"""Fused attention op + transformer encoder layer on the gfx950 kernel
(ops/hip/attention.hip).

fused_attention(): multi-head attention forward on the HIP kernel for
S <= 64 (the predictors' shape). Training gradients come from a
differentiable torch recompute (the kernel forward defines the value;
backward rebuilds the same math with torch ops — bf16-level agreement is
covered by tests), so the fused path serves both inference and training.
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn

from ..ops import require_hip_ops


def attn_fwd_hip(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                 scale: float | None = None) -> torch.Tensor:
    """q/k/v: (BH, S, D) -> (BH, S, D). bf16 in/out, S <= 64."""
    ops = require_hip_ops()
    bh, s, d = q.shape
    if scale is None:
        scale = 1.0 / math.sqrt(d)
    q = q.to(torch.bfloat16).contiguous()
    k = k.to(torch.bfloat16).contiguous()
    v = v.to(torch.bfloat16).contiguous()
    o = torch.empty_like(q)
    stream = torch.cuda.current_stream(q.device).cuda_stream
    ops.attn_fwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
                 bh, s, d, float(scale), stream)
    return o


class _FusedAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, scale):
        ctx.save_for_backward(q, k, v)
        ctx.scale = scale
        return attn_fwd_hip(q, k, v, scale)

    @staticmethod
    def backward(ctx, grad_o):
        q, k, v = ctx.saved_tensors
        with torch.enable_grad():
            q2 = q.detach().float().requires_grad_(True)
            k2 = k.detach().float().requires_grad_(True)
            v2 = v.detach().float().requires_grad_(True)
            s = torch.softmax(
                (q2 @ k2.transpose(1, 2)) * ctx.scale, dim=-1)
            o = s @ v2
            gq, gk, gv = torch.autograd.grad(o, (q2, k2, v2),
                                             grad_o.float())
        return gq.to(q.dtype), gk.to(k.dtype), gv.to(v.dtype), None


def fused_attention(q, k, v, scale=None):
    """(B, H, S, D) or (BH, S, D); HIP forward + recompute backward."""
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
