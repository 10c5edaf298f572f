"""Fused AR/TAR activation regularization (fastai RNNTrainer alpha=2,
beta=1 — reference fit-loop semantics).

reg = alpha * mean(out^2) + beta * mean((r[:, 1:] - r[:, :-1])^2)

On CUDA one HIP kernel reads each (B, T, H) activation once (vs ~6 eager
kernels re-streaming them, ~13 ms/step at the bench shape); backward is a
single kernel writing both grad contributions. The raw output ``r`` is
consumed through its TIME-MAJOR base storage (the LSTM layer's native
(T, B, H) layout) so no transpose copy happens. CPU / odd layouts: plain
torch (numerics reference for tests)."""
from __future__ import annotations

import torch
from torch import Tensor

from . import extension as ext

__all__ = ["artar_loss"]


def _flat_base(t: Tensor):
    """The AR term is layout-agnostic: any contiguous covering of the
    elements works. TensorIterator keeps the dropout output in the LSTM's
    time-major layout, so accept either orientation's contiguous base.
    Returns (base, swapped) or (None, False)."""
    if t.is_contiguous():
        return t, False
    tt = t.transpose(0, 1)
    if tt.is_contiguous():
        return tt, True
    return None, False


class _ARTARFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, out: Tensor, r: Tensor, alpha: float, beta: float):
        # r: (B, T, H) transpose view of contiguous (T, B, H) storage
        lib = ext.require()
        r_tm = r.transpose(0, 1)
        out, ctx.out_swapped = _flat_base(out)
        acc = lib.artar_forward(out, r_tm)  # [sum sq, sum diff sq] fp32
        n = max(out.numel(), 1)
        m = max(r.shape[0] * (r.shape[1] - 1) * r.shape[2], 1)
        ctx.save_for_backward(out, r)
        ctx.coeffs = (2.0 * alpha / n, 2.0 * beta / m)
        w = torch.stack([torch.full((), alpha / n, device=out.device),
                         torch.full((), beta / m, device=out.device)])
        return (acc * w).sum()

    @staticmethod
    def backward(ctx, dloss: Tensor):
        lib = ext.require()
        out, r = ctx.saved_tensors
        ca, cb = ctx.coeffs
        d32 = dloss.detach().to(torch.float32).reshape(1).contiguous()
        ob, _ = _flat_base(out)
        dout, dr_tm = lib.artar_backward(ob, r.transpose(0, 1), d32, ca, cb)
        if ctx.out_swapped:
            dout = dout.transpose(0, 1)
        return dout, dr_tm.transpose(0, 1), None, None


def artar_loss(out: Tensor, r: Tensor, alpha: float, beta: float) -> Tensor:
    """out: output-dropped activations (B, T, H); r: raw last-layer
    output (B, T, H, typically a transpose view of time-major storage).
    Returns the scalar regularization term (0-dim fp32 tensor)."""
    if (out.is_cuda and out.dim() == 3 and _flat_base(out)[0] is not None
            and r.dim() == 3 and r.transpose(0, 1).is_contiguous()
            and r.shape[1] > 1
            and (r.shape[0] * r.shape[2]) % 8 == 0
            and out.dtype == r.dtype):
        return _ARTARFunction.apply(out, r, alpha, beta)
    reg = out.new_zeros((), dtype=torch.float32)
    if alpha:
        reg = reg + alpha * out.pow(2).mean(dtype=torch.float32)
    if beta and r.shape[1] > 1:
        reg = reg + beta * (r[:, 1:] - r[:, :-1]).pow(2) \
            .mean(dtype=torch.float32)
    return reg
