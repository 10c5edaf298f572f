"""QRNN forward: T-parallel gate GEMM + CDNA4 fo-pooling scan.

The reference exposes ``--qrnn`` on its training CLIs
(Issue_Embeddings/train.py:43, hyperparam_sweep/lm_tune.py:43 — fastai's
QRNN needs cuDNN there). MI355X design: the z|f|o projection over ALL
timesteps is one hipBLASLt GEMM of shape (B·T, win·E)×(win·E, 3H) — that is
the entire point of QRNN vs LSTM, the recurrent GEMM disappears — and the
only sequential work is the elementwise fo-pool scan
``c_t = f_t·c_{t-1} + (1-f_t)·z_t, h_t = o_t·c_t`` which runs as a
bandwidth-bound HIP kernel (one lane per (b,h), ops/csrc/qrnn_pool.hip).

fastai QRNNLayer semantics kept: window=2 on the first layer (gate input is
``[x_t, x_{t-1}]``, with ``x_{-1}`` carried across BPTT windows via
save_prev_x), window=1 on the rest; tanh/sigmoid/sigmoid activations;
output gate applied to c.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch
from torch import Tensor

from . import extension


def _fo_pool_torch(gates: Tensor, c0: Tensor) -> Tuple[Tensor, Tensor]:
    """Plain-PyTorch fo-pool (CPU path / numerics reference). gates is the
    PRE-activation (B,T,3H) buffer; autograd handles backward."""
    H = gates.size(-1) // 3
    z = torch.tanh(gates[..., :H])
    f = torch.sigmoid(gates[..., H:2 * H])
    o = torch.sigmoid(gates[..., 2 * H:])
    cs = []
    c = c0
    for t in range(gates.size(1)):
        c = f[:, t] * c + (1 - f[:, t]) * z[:, t]
        cs.append(c)
    c_all = torch.stack(cs, dim=1)
    return o * c_all, c


class _FoPoolFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gates: Tensor, c0: Tensor):
        ext = extension.require()
        gates = gates.contiguous()
        c0 = c0.contiguous()
        h, c = ext.qrnn_fo_pool_fwd(gates, c0)  # activates gates in place
        ctx.save_for_backward(gates, c, c0)
        return h, c[:, -1].clone()

    @staticmethod
    def backward(ctx, dh: Tensor, dcT: Optional[Tensor]):
        gates, c, c0 = ctx.saved_tensors
        if dcT is None:
            dcT = torch.zeros_like(c0)
        ext = extension.require()
        dgates, dc0 = ext.qrnn_fo_pool_bwd(
            gates, c, c0, dh.contiguous(), dcT.contiguous())
        return dgates, dc0


def fo_pool(gates: Tensor, c0: Tensor) -> Tuple[Tensor, Tensor]:
    """(h (B,T,H), c_T (B,H)) from pre-activation gates (B,T,3H).

    On GPU ``gates`` is CONSUMED: the kernel overwrites it with the
    activated gate values (saved for backward). Pass a fresh tensor —
    ``qrnn_forward`` always does."""
    if gates.is_cuda:  # require() inside raises loudly if the .so is missing
        return _FoPoolFn.apply(gates, c0)
    return _fo_pool_torch(gates, c0)


def qrnn_forward(x: Tensor, c0: Tensor, weight: Tensor, bias: Tensor,
                 window: int = 1, prev_x: Optional[Tensor] = None
                 ) -> Tuple[Tensor, Tensor]:
    """One QRNN layer. x (B,T,E); weight (3H, window*E); returns (h, c_T)."""
    B, T, E = x.shape
    if window == 2:
        if prev_x is None:
            prev_x = x.new_zeros(B, 1, E)
        shifted = torch.cat([prev_x, x[:, :-1]], dim=1)
        inp = torch.cat([x, shifted], dim=-1)
    elif window == 1:
        inp = x
    else:
        raise ValueError(f"window must be 1 or 2 (got {window})")
    gates = _gates_gemm(inp.reshape(B * T, -1), weight, bias)
    return fo_pool(gates.view(B, T, -1), c0)


def _gates_gemm(flat: Tensor, weight: Tensor, bias: Tensor,
                max_rows: Optional[int] = None) -> Tensor:
    """bias + flat @ weight.T, chunked over rows when the output would be
    huge. Large serve batches (e.g. B=200, T=1600) push the fused gate
    GEMM's output past 2^31 elements, which memory-faults in the GEMM
    library (observed on ROCm 7.2 at 320000x7200 out); chunking keeps
    every call well under while leaving each chunk GEMM-saturating."""
    n_out = weight.size(0)
    if max_rows is None:
        max_rows = max(1, ((1 << 31) - (1 << 27)) // n_out)
    if flat.size(0) <= max_rows:
        return torch.addmm(bias, flat, weight.t())
    wt = weight.t()
    chunks = range(0, flat.size(0), max_rows)
    if torch.is_grad_enabled() and (flat.requires_grad
                                    or weight.requires_grad
                                    or bias.requires_grad):
        return torch.cat([torch.addmm(bias, flat[s:s + max_rows], wt)
                          for s in chunks])
    gates = flat.new_empty((flat.size(0), n_out))
    for s in chunks:  # serve path: write into one buffer, no autograd
        torch.addmm(bias, flat[s:s + max_rows], wt, out=gates[s:s + max_rows])
    return gates
