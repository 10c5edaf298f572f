"""Variational (locked) dropout op — SURVEY.md §2.4 K4.

One (B, 1, H) mask broadcast across all timesteps (fastai RNNDropout
semantics: hidden_p between LSTM layers, input_p on embeddings, output_p on
the decoder input). The broadcast multiply is a single memory-bound
elementwise kernel either way; PyTorch-ROCm emits one fused broadcast-mul,
so a custom kernel buys nothing until it is fused into the LSTM layer
epilogue (tracked as the K4-fusion item in SURVEY.md §2.4).
"""
from __future__ import annotations

from torch import Tensor

__all__ = ["variational_dropout"]


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
