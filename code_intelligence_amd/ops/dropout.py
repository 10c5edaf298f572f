"""Variational (locked) dropout op — SURVEY.md §2.4 K4.

One (B, 1, H) mask broadcast across all timesteps (fastai RNNDropout
semantics: hidden_p between LSTM layers, input_p on embeddings, output_p on
the decoder input). The broadcast multiply is a single memory-bound
elementwise kernel either way; PyTorch-ROCm emits one fused broadcast-mul,
so a custom kernel buys nothing until it is fused into the LSTM layer
epilogue (tracked as the K4-fusion item in SURVEY.md §2.4).
"""
from __future__ import annotations

import torch
from torch import Tensor

from . import extension as ext

__all__ = ["variational_dropout", "dropconnect"]


def variational_dropout(x: Tensor, p: float, training: bool) -> Tensor:
    if not training or p == 0.0:
        return x
    # mask over (B, 1, H) — same mask for every timestep
    if x.dim() == 3:
        size = (x.size(0), 1, x.size(2))
    else:
        size = x.shape
    mask = x.new_empty(size).bernoulli_(1 - p).div_(1 - p)
    return x * mask


class _DropConnectFunction(torch.autograd.Function):
    """Seeded DropConnect (K3, fastai WeightDropout weight_p) for the
    recurrent weights. The mask is a hash of (seed, index) regenerated in
    backward (dropconnect.hip), so no mask tensor is ever materialized and
    the grad pass masks the incoming (4H, H) gradient IN PLACE — the
    masked weight's only consumer is the LSTM layer, whose backward
    returns a freshly-allocated dW, so in-place is safe."""

    @staticmethod
    def forward(ctx, w: Tensor, p: float, seed: int):
        out = ext.require().dropconnect_apply(w.contiguous(), seed, p)
        ctx.p, ctx.seed = p, seed
        return out

    @staticmethod
    def backward(ctx, g: Tensor):
        g = g.contiguous()
        ext.require().dropconnect_grad_(g, ctx.seed, ctx.p)
        return g, None, None


def dropconnect(w: Tensor, p: float, training: bool) -> Tensor:
    """Mask-scale ``w`` by a fresh per-forward DropConnect mask.

    CUDA: seeded-mask HIP kernel (no mask tensor). CPU: F.dropout
    reference (numerics tests compare the two distributionally; exact
    per-element equality is not required — masks are random either way).
    """
    if not training or p == 0.0:
        return w
    if w.is_cuda:  # ext.require() inside raises if the .so is missing
        seed = int(torch.randint(0, 2 ** 62, (1,)).item())
        return _DropConnectFunction.apply(w, p, seed)
    return torch.nn.functional.dropout(w, p=p, training=True)
