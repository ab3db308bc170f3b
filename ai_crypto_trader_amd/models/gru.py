"""GRU layer on the fused gfx950 sequence kernels (ops/hip/gru.hip) —
sibling of models/lstm.py (neural_network_service.py:202-211 GRU model)."""

from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import require_hip_ops


class _FusedGRUSeq(torch.autograd.Function):
    @staticmethod
    def forward(ctx, xproj: torch.Tensor, w_hh: torch.Tensor,
                b_hh: torch.Tensor):
        ops = require_hip_ops()
        T, B, three_h = xproj.shape
        H = three_h // 3
        dev = xproj.device
        w_bf = w_hh.to(torch.bfloat16).contiguous()          # (H, 3H)
        wt_bf = w_bf.t().contiguous()                        # (3H, H)
        bias = b_hh.float().contiguous()
        h_out = torch.empty((T, B, H), dtype=torch.bfloat16, device=dev)
        gates = torch.empty((T, B, three_h), dtype=torch.bfloat16,
                            device=dev)
        hpn = torch.empty((T, B, H), dtype=torch.float32, device=dev)
        stream = torch.cuda.current_stream(dev).cuda_stream
        ops.gru_seq_fwd(
            xproj.contiguous().data_ptr(), wt_bf.data_ptr(),
            bias.data_ptr(), h_out.data_ptr(), gates.data_ptr(),
            hpn.data_ptr(), B, T, H, 1, stream,
        )
        ctx.save_for_backward(gates, hpn, h_out, w_bf)
        ctx.dims = (T, B, H)
        return h_out

    @staticmethod
    def backward(ctx, grad_h: torch.Tensor):
        ops = require_hip_ops()
        gates, hpn, h_out, w_bf = ctx.saved_tensors
        T, B, H = ctx.dims
        dev = grad_h.device
        dgates_x = torch.empty((T, B, 3 * H), dtype=torch.bfloat16,
                               device=dev)
        dh_up = grad_h.float().contiguous()
        stream = torch.cuda.current_stream(dev).cuda_stream
        ops.gru_seq_bwd(
            dh_up.data_ptr(), gates.data_ptr(), hpn.data_ptr(),
            h_out.data_ptr(), w_bf.contiguous().data_ptr(),
            dgates_x.data_ptr(), B, T, H, stream,
        )
        # h-side dgates: the n-column is da_n * r (gru.hip header).
        # All bf16: the weight-grad GEMM accumulates f32 inside hipBLASLt
        # (autocast semantics), no f32 materialization of dgates
        dg_h = dgates_x.clone()
        dg_h[..., 2 * H:] = dg_h[..., 2 * H:] * gates[..., :H]
        h_prev = torch.cat(
            [torch.zeros((1, B, H), dtype=h_out.dtype, device=dev),
             h_out[:-1]], dim=0,
        )
        dg = dg_h.reshape(T * B, 3 * H)
        dw_hh = (h_prev.reshape(T * B, H).t() @ dg).float()
        db_hh = dg.sum(dim=0, dtype=torch.float32)
        return dgates_x, dw_hh, db_hh


def gru_seq_infer(xproj: torch.Tensor, w_hh: torch.Tensor,
                  b_hh: torch.Tensor) -> torch.Tensor:
    """Inference-only fused forward: skips the backward-save stores
    (gates/hpn), ~4x fewer global writes per cell — the serving path
    (mirrors lstm_seq_infer)."""
    ops = require_hip_ops()
    T, B, three_h = xproj.shape
    H = three_h // 3
    dev = xproj.device
    wt_bf = w_hh.detach().to(torch.bfloat16).t().contiguous()
    bias = b_hh.detach().float().contiguous()
    h_out = torch.empty((T, B, H), dtype=torch.bfloat16, device=dev)
    stream = torch.cuda.current_stream(dev).cuda_stream
    ops.gru_seq_fwd(xproj.contiguous().data_ptr(), wt_bf.data_ptr(),
                    bias.data_ptr(), h_out.data_ptr(), 0, 0, B, T, H, 0,
                    stream)
    return h_out


class FusedGRULayer(nn.Module):
    """One GRU layer (T, B, F) -> (T, B, H); HIP path on GPU, plain fp32
    reference on CPU (also the numerics oracle for GPU tests)."""

    def __init__(self, input_size: int, hidden_size: int):
        super().__init__()
        assert hidden_size in (32, 64)
        self.input_size = input_size
        self.hidden_size = hidden_size
        self.w_ih = nn.Parameter(torch.empty(input_size, 3 * hidden_size))
        self.b_ih = nn.Parameter(torch.zeros(3 * hidden_size))
        self.w_hh = nn.Parameter(torch.empty(hidden_size, 3 * hidden_size))
        self.b_hh = nn.Parameter(torch.zeros(3 * hidden_size))
        k = hidden_size ** -0.5
        nn.init.uniform_(self.w_ih, -k, k)
        nn.init.uniform_(self.w_hh, -k, k)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        if x.is_cuda:
            T, B, _ = x.shape
            pad = (-B) % 64
            if pad:
                x = torch.cat([x, x.new_zeros(T, pad, x.shape[2])], dim=1)
            bf = torch.bfloat16    # f32 master weights, bf16 GEMMs
            xproj = (x.to(bf) @ self.w_ih.to(bf) + self.b_ih.to(bf))
            if torch.is_grad_enabled():
                h = _FusedGRUSeq.apply(xproj, self.w_hh, self.b_hh)
            else:    # serving path: no backward saves
                h = gru_seq_infer(xproj, self.w_hh, self.b_hh)
            return h[:, :B] if pad else h
        return self._forward_reference(x)

    def _forward_reference(self, x: torch.Tensor) -> torch.Tensor:
        T, B, _ = x.shape
        H = self.hidden_size
        h = x.new_zeros(B, H)
        outs = []
        for t in range(T):
            xp = x[t] @ self.w_ih + self.b_ih
            hp = h @ self.w_hh + self.b_hh
            xr, xz, xn = xp.split(H, dim=1)
            hr, hz, hn = hp.split(H, dim=1)
            r = torch.sigmoid(xr + hr)
            z = torch.sigmoid(xz + hz)
            n = torch.tanh(xn + r * hn)
            h = (1 - z) * n + z * h
            outs.append(h)
        return torch.stack(outs, dim=0)
