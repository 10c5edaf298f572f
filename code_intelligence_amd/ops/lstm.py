"""Fused LSTM layer op.

Forward over a whole (B, T, In) sequence for ONE layer:

  * the input-side projection ``x @ W_ih^T + b`` has no time dependency and
    runs as one large plain GEMM (hipBLASLt via torch.matmul — per the
    MI355X design rules, library GEMMs are allowed for plain GEMMs);
  * the recurrent part is the hot sequential loop: per timestep a
    (B,H)x(H,4H) GEMM + gate activations + c/h update. On ROCm this is a
    hand-written CDNA4 kernel — either the fully fused MFMA cell kernel
    (lstm_gemm.hip, gates interleaved per hidden unit so the epilogue can
    finish c/h locally) or a hipBLASLt GEMM + pointwise HIP kernel
    (lstm_pointwise.hip), selected by CI_LSTM_MODE=fused|lib (default fused).

Backward mirrors it: a sequential per-timestep pointwise+GEMM loop for
dh/dc/dgates, then batched plain GEMMs for dW_ih, dW_hh, db, dx.

Gate order follows PyTorch/cuDNN (i, f, g, o) for checkpoint parity.
The CPU path is a straightforward PyTorch reference (tests compare the HIP
kernels against it in fp32).
"""
from __future__ import annotations

import os
from typing import Tuple

import torch
from torch import Tensor

from . import extension as ext

__all__ = ["lstm_forward"]


def _cpu_lstm_loop(x: Tensor, h0: Tensor, c0: Tensor, w_ih: Tensor, w_hh: Tensor,
                   b_ih: Tensor, b_hh: Tensor) -> Tuple[Tensor, Tensor, Tensor]:
    """Pure-PyTorch reference (autograd-capable). x: (B,T,In)."""
    B, T, _ = x.shape
    H = w_hh.shape[1]
    xp = torch.addmm((b_ih + b_hh), x.reshape(B * T, -1), w_ih.t()).view(B, T, 4 * H)
    h, c = h0, c0
    outs = []
    for t in range(T):
        gates = xp[:, t] + h @ w_hh.t()
        i, f, g, o = gates.chunk(4, dim=1)
        i, f, g, o = i.sigmoid(), f.sigmoid(), g.tanh(), o.sigmoid()
        c = f * c + i * g
        h = o * torch.tanh(c)
        outs.append(h)
    return torch.stack(outs, dim=1), h, c


class _FusedLSTMFunction(torch.autograd.Function):
    """GPU path: HIP cell kernels, explicit backward."""

    @staticmethod
    def forward(ctx, x, h0, c0, w_ih, w_hh, b_ih, b_hh):
        lib = ext.require()
        B, T, In = x.shape
        H = w_hh.shape[1]
        dt = x.dtype
        # input projection: one plain GEMM over all timesteps
        bias = (b_ih + b_hh).to(torch.float32)
        xp = torch.matmul(x.reshape(B * T, In), w_ih.t()).view(B, T, 4 * H)
        hs = torch.empty(B, T, H, dtype=dt, device=x.device)
        cs = torch.empty(B, T, H, dtype=torch.float32, device=x.device)
        gates = torch.empty(B, T, 4 * H, dtype=dt, device=x.device)
        mode = os.environ.get("CI_LSTM_MODE", "fused")
        if mode == "fused":
            lib.lstm_seq_forward_fused(xp, bias, h0, c0.to(torch.float32), w_hh,
                                       hs, cs, gates)
        else:
            lib.lstm_seq_forward_lib(xp, bias, h0, c0.to(torch.float32), w_hh,
                                     hs, cs, gates)
        ctx.save_for_backward(x, h0, c0, w_ih, w_hh, gates, hs, cs)
        hT = hs[:, -1].clone()
        cT = cs[:, -1].to(dt).clone()
        return hs, hT, cT

    @staticmethod
    def backward(ctx, dhs, dhT, dcT):
        lib = ext.require()
        x, h0, c0, w_ih, w_hh, gates, hs, cs = ctx.saved_tensors
        B, T, In = x.shape
        H = w_hh.shape[1]
        dt = x.dtype
        dgates = torch.empty(B, T, 4 * H, dtype=dt, device=x.device)
        dh0 = torch.empty(B, H, dtype=torch.float32, device=x.device)
        dc0 = torch.empty(B, H, dtype=torch.float32, device=x.device)
        # sequential reverse loop: pointwise cell backward + per-step dh GEMM
        lib.lstm_seq_backward(dhs.contiguous(), dhT.contiguous(),
                              dcT.to(torch.float32).contiguous(),
                              gates, hs, cs, c0.to(torch.float32), w_hh,
                              dgates, dh0, dc0)
        dg2 = dgates.reshape(B * T, 4 * H)
        # batched weight/input grads: plain GEMMs
        dx = torch.matmul(dg2, w_ih).view(B, T, In)
        dw_ih = torch.matmul(dg2.t(), x.reshape(B * T, In))
        hprev = torch.cat([h0.unsqueeze(1), hs[:, :-1]], dim=1).reshape(B * T, H)
        dw_hh = torch.matmul(dg2.t(), hprev)
        db = dg2.sum(dim=0).to(dt)
        return (dx, dh0.to(dt), dc0.to(dt), dw_ih.to(w_ih.dtype),
                dw_hh.to(w_hh.dtype), db, db.clone())


def lstm_forward(x: Tensor, h0: Tensor, c0: Tensor, w_ih: Tensor, w_hh: Tensor,
                 b_ih: Tensor, b_hh: Tensor):
    """Run one LSTM layer over (B,T,In). Returns (out (B,T,H), (hT, cT))."""
    if x.is_cuda:
        hs, hT, cT = _FusedLSTMFunction.apply(x, h0, c0, w_ih, w_hh, b_ih, b_hh)
        return hs, (hT, cT)
    out, h, c = _cpu_lstm_loop(x, h0, c0, w_ih, w_hh, b_ih, b_hh)
    return out, (h, c)
