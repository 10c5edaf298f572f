"""K1: embedding gather with fused word-level row dropout.

fastai ``EmbeddingDropout`` (reference train.py:69-70, embed_p=0.02)
multiplies the whole (vocab, emb) table by a (vocab, 1) Bernoulli mask and
then gathers — materializing a masked 60k x 800 copy in HBM every training
forward. The MI355X path fuses the row mask into the gather
(embedding.hip): only the (B, T) looked-up rows are read, each scaled by
its row's mask value on the fly; backward scatter-adds the masked output
grads straight into an fp32 grad buffer (one atomicAdd kernel, pad row
skipped — matching F.embedding padding_idx semantics).
"""
from __future__ import annotations

from typing import Optional

import torch
from torch import Tensor

from . import extension as ext

__all__ = ["embedding_row_dropout"]

_EMPTY_F32 = {}


def _empty(device) -> Tensor:
    key = str(device)
    if key not in _EMPTY_F32:
        _EMPTY_F32[key] = torch.empty(0, dtype=torch.float32, device=device)
    return _EMPTY_F32[key]


class _EmbGatherFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, weight: Tensor, ids: Tensor, rowmask: Tensor,
                pad_idx: int):
        lib = ext.require()
        ids = ids.contiguous()
        out = lib.emb_gather(weight, ids, rowmask)
        ctx.save_for_backward(ids, rowmask)
        ctx.V = weight.shape[0]
        ctx.pad = pad_idx
        ctx.wdtype = weight.dtype
        return out

    @staticmethod
    def backward(ctx, gout: Tensor):
        lib = ext.require()
        ids, rowmask = ctx.saved_tensors
        dw = lib.emb_scatter(gout, ids, rowmask, ctx.V, ctx.pad)
        return dw.to(ctx.wdtype), None, None, None


def embedding_row_dropout(weight: Tensor, ids: Tensor,
                          embed_p: float, training: bool,
                          pad_idx: Optional[int],
                          scale: Optional[float] = None) -> Tensor:
    """CUDA path of EmbeddingDropout. Returns (B, T, E) embeddings with
    whole-word rows dropped (train) — no masked table materialization."""
    if training and embed_p != 0.0:
        rowmask = torch.empty(weight.shape[0], dtype=torch.float32,
                              device=weight.device)
        rowmask.bernoulli_(1 - embed_p).div_(1 - embed_p)
        if scale is not None:
            rowmask.mul_(scale)
    elif scale is not None:
        rowmask = torch.full((weight.shape[0],), float(scale),
                             dtype=torch.float32, device=weight.device)
    else:
        rowmask = _empty(weight.device)
    pad = -1 if pad_idx is None else pad_idx
    return _EmbGatherFunction.apply(weight, ids, rowmask, pad)
