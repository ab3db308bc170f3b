"""LSTM price predictor with hand-written gfx950 recurrent-cell kernels.

Replaces neural_network_service.py's Keras model zoo entry points
(create_model :164-421 — LSTM 64->32 + Dense16 + Dense1,
config.json:454-476) with PyTorch-ROCm modules whose recurrent step is the
fused HIP kernel ops/hip/lstm.hip (bf16 MFMA gate GEMM, W_hh LDS-resident
across the sequence). The input projection x @ W_ih and the backward
weight reductions are plain hipBLASLt GEMMs via torch.matmul.

On CPU (no GPU) the same module runs on torch.nn.LSTM so the control-plane
services and tests work everywhere; on a GPU machine the HIP path is
mandatory (ops.require_hip_ops) — no silent eager fallback.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from ..ops import require_hip_ops


class _FusedLSTMSeq(torch.autograd.Function):
    """Recurrent part only: xproj (T,B,4H) bf16 -> h_out (T,B,H) bf16.

    W_hh enters as float32 master weights; dW_hh/db_hh are computed with
    one big GEMM over the saved sequence (no per-step reduction kernel).
    """

    @staticmethod
    def forward(ctx, xproj: torch.Tensor, w_hh: torch.Tensor,
                b_hh: torch.Tensor):
        ops = require_hip_ops()
        T, B, four_h = xproj.shape
        H = four_h // 4
        assert xproj.dtype == torch.bfloat16 and xproj.is_cuda
        dev = xproj.device
        w_bf = w_hh.to(torch.bfloat16).contiguous()           # (H, 4H)
        wt_bf = w_bf.t().contiguous()                         # (4H, H)
        bias = b_hh.float().contiguous()
        h_out = torch.empty((T, B, H), dtype=torch.bfloat16, device=dev)
        gates = torch.empty((T, B, four_h), dtype=torch.bfloat16, device=dev)
        c_sav = torch.empty((T, B, H), dtype=torch.float32, device=dev)
        stream = torch.cuda.current_stream(dev).cuda_stream
        ops.lstm_seq_fwd(
            xproj.contiguous().data_ptr(), wt_bf.data_ptr(), bias.data_ptr(),
            h_out.data_ptr(), gates.data_ptr(), c_sav.data_ptr(), B, T, H,
            1, stream,
        )
        ctx.save_for_backward(gates, c_sav, h_out, w_bf)
        ctx.dims = (T, B, H)
        return h_out


    @staticmethod
    def backward(ctx, grad_h: torch.Tensor):
        ops = require_hip_ops()
        gates, c_sav, h_out, w_bf = ctx.saved_tensors
        T, B, H = ctx.dims
        dev = grad_h.device
        dgates = torch.empty((T, B, 4 * H), dtype=torch.bfloat16, device=dev)
        dh_up = grad_h.float().contiguous()
        stream = torch.cuda.current_stream(dev).cuda_stream
        ops.lstm_seq_bwd(
            dh_up.data_ptr(), gates.data_ptr(), c_sav.data_ptr(),
            w_bf.contiguous().data_ptr(), dgates.data_ptr(), B, T, H, stream,
        )
        # dW_hh = sum_t h_{t-1}^T dgates_t — one hipBLASLt bf16 GEMM
        # (f32 accumulation inside, bf16 grad out = torch.autocast
        # semantics; the f32 materialization of dgates was 1 GB/step)
        h_prev = torch.cat(
            [torch.zeros((1, B, H), dtype=h_out.dtype, device=dev),
             h_out[:-1]], dim=0,
        )
        dg = dgates.reshape(T * B, 4 * H)
        dw_hh = (h_prev.reshape(T * B, H).t() @ dg).float()
        db_hh = dg.sum(dim=0, dtype=torch.float32)
        return dgates, dw_hh, db_hh



def lstm_seq_infer(xproj: torch.Tensor, w_hh: torch.Tensor,
                   b_hh: torch.Tensor) -> torch.Tensor:
    """Inference-only fused forward: skips the backward-save stores
    (gates/c), ~5x fewer global writes per cell — the serving path."""
    ops = require_hip_ops()
    T, B, four_h = xproj.shape
    H = four_h // 4
    dev = xproj.device
    w_bf = w_hh.detach().to(torch.bfloat16).t().contiguous()
    bias = b_hh.detach().float().contiguous()
    h_out = torch.empty((T, B, H), dtype=torch.bfloat16, device=dev)
    stream = torch.cuda.current_stream(dev).cuda_stream
    ops.lstm_seq_fwd(
        xproj.contiguous().data_ptr(), w_bf.data_ptr(), bias.data_ptr(),
        h_out.data_ptr(), 0, 0, B, T, H, 0, stream,
    )
    return h_out

class FusedLSTMLayer(nn.Module):
    """One LSTM layer: hipBLASLt input projection + fused HIP recurrence.

    Input/output layout (T, B, F) time-major (matches the kernels).
    """

    def __init__(self, input_size: int, hidden_size: int):
        super().__init__()
        assert hidden_size in (32, 64), "HIP kernel supports H in {32, 64}"
        self.input_size = input_size
        self.hidden_size = hidden_size
        self.w_ih = nn.Parameter(torch.empty(input_size, 4 * hidden_size))
        self.b_ih = nn.Parameter(torch.zeros(4 * hidden_size))
        self.w_hh = nn.Parameter(torch.empty(hidden_size, 4 * hidden_size))
        self.b_hh = nn.Parameter(torch.zeros(4 * hidden_size))
        k = hidden_size ** -0.5
        nn.init.uniform_(self.w_ih, -k, k)
        nn.init.uniform_(self.w_hh, -k, k)

    def forward(self, x: torch.Tensor) -> torch.Tensor:   # (T, B, F)
        if x.is_cuda:
            T, B, _ = x.shape
            pad = (-B) % 64              # kernel batch tile is 64 rows
            if pad:
                x = torch.cat([x, x.new_zeros(T, pad, x.shape[2])], dim=1)
            # bf16 input projection with f32 master weights (autocast
            # style): halves xproj traffic and makes dW_ih a bf16 GEMM
            bf = torch.bfloat16
            xproj = (x.to(bf) @ self.w_ih.to(bf) + self.b_ih.to(bf))
            if torch.is_grad_enabled():
                h = _FusedLSTMSeq.apply(xproj, self.w_hh, self.b_hh)
            else:   # serving path: no backward saves
                h = lstm_seq_infer(xproj, self.w_hh, self.b_hh)
            return h[:, :B] if pad else h
        return self._forward_reference(x)

    def _forward_reference(self, x: torch.Tensor) -> torch.Tensor:
        """Plain fp32 reference (CPU path and GPU numerics tests)."""
        T, B, _ = x.shape
        H = self.hidden_size
        h = x.new_zeros(B, H)
        c = x.new_zeros(B, H)
        outs = []
        for t in range(T):
            g = x[t] @ self.w_ih + self.b_ih + h @ self.w_hh + self.b_hh
            i, f, gg, o = g.split(H, dim=1)
            i, f, o = torch.sigmoid(i), torch.sigmoid(f), torch.sigmoid(o)
            gg = torch.tanh(gg)
            c = f * c + i * gg
            h = o * torch.tanh(c)
            outs.append(h)
        return torch.stack(outs, dim=0)


class LSTMPricePredictor(nn.Module):
    """LSTM(64) -> LSTM(32) -> Dense(16) -> Dense(1), predicting the next
    close return from a (B, T, F) feature window — the reference's default
    architecture (neural_network_service.py:191-200, config.json:454-476).
    """

    def __init__(self, n_features: int = 9, seq_len: int = 60,
                 hidden: tuple[int, int] = (64, 32)):
        super().__init__()
        self.seq_len = seq_len
        self.l1 = FusedLSTMLayer(n_features, hidden[0])
        self.l2 = FusedLSTMLayer(hidden[0], hidden[1])
        self.head = nn.Sequential(
            nn.Linear(hidden[1], 16), nn.ReLU(), nn.Linear(16, 1),
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:   # (B, T, F)
        x = x.transpose(0, 1)                             # (T, B, F)
        h = self.l1(x)
        h = self.l2(h)
        last = h[-1].float()                              # (B, H2)
        return self.head(last).squeeze(-1)
