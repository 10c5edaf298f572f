"""Masked concat-pool (K5): cat([mean, max, last], dim=-1) with true lengths.

Reference semantics: py/code_intelligence/inference.py:93 (single sequence)
and inference.py:232-263 ``batch_seq_pool`` (batched, padding masked per true
length; "last" is the hidden state at position length-1, NOT the padded tail).

On ROCm this is a single-pass HIP reduction kernel (the serve hot path);
the CPU path is the PyTorch composition used as the numerics reference.
"""
from __future__ import annotations

import torch
from torch import Tensor

from . import extension as ext

__all__ = ["concat_pool"]


def _cpu_concat_pool(hidden: Tensor, lengths: Tensor) -> Tensor:
    """hidden: (B, T, H) final-layer hidden states; lengths: (B,) int64."""
    B, T, H = hidden.shape
    ar = torch.arange(T, device=hidden.device).unsqueeze(0)          # (1,T)
    mask = (ar < lengths.unsqueeze(1)).unsqueeze(-1)                 # (B,T,1)
    hf = hidden.float()
    summed = (hf * mask).sum(dim=1)
    mean = summed / lengths.clamp_min(1).unsqueeze(1).float()
    neg = torch.finfo(torch.float32).min
    maxed = hf.masked_fill(~mask, neg).max(dim=1).values
    last = hf[torch.arange(B, device=hidden.device), (lengths - 1).clamp_min(0)]
    return torch.cat([mean, maxed, last], dim=1).to(hidden.dtype)


def concat_pool(hidden: Tensor, lengths: Tensor) -> Tensor:
    """Returns (B, 3H): [mean, max, last] pooled over true lengths."""
    if hidden.is_cuda:
        lib = ext.require()
        return lib.concat_pool(hidden.contiguous(), lengths.to(torch.int32).contiguous())
    return _cpu_concat_pool(hidden, lengths)
