"""Fused tied-decoder softmax + cross-entropy (K6) — the FLOPs king.

Reference semantics: fastai LinearDecoder + FlattenedLoss(CrossEntropy) over
a 60k vocab (train.py:70, tie_weights/out_bias). Materializing (B·T, V)
logits for the whole batch costs ~31 GB at the bench shape — instead the op
chunks over rows: per chunk a plain hipBLASLt GEMM produces a logits tile
that never leaves HBM-resident scratch, a HIP kernel reduces it to
(logsumexp, target-logit) in one pass, and backward RECOMPUTES the chunk's
logits and transforms them in place to dlogits = (softmax - onehot)/N with a
second HIP kernel, feeding the dh/dE GEMMs. Only O(B·T) state (the lse
vector) is saved between forward and backward.

CPU path: plain F.cross_entropy composition (numerics reference).
"""
from __future__ import annotations

import torch
from torch import Tensor

from . import extension as ext

__all__ = ["tied_decoder_ce", "TiedDecoderCE"]


class _FusedCEFunction(torch.autograd.Function):
    CHUNK = 8192

    @staticmethod
    def forward(ctx, h: Tensor, weight: Tensor, bias: Tensor, targets: Tensor):
        lib = ext.require()
        N, H = h.shape
        lse = torch.empty(N, dtype=torch.float32, device=h.device)
        tgt_logit = torch.empty(N, dtype=torch.float32, device=h.device)
        for s in range(0, N, _FusedCEFunction.CHUNK):
            e = min(N, s + _FusedCEFunction.CHUNK)
            logits = torch.matmul(h[s:e], weight.t())
            if bias is not None:
                logits += bias
            lib.ce_rowstats(logits, targets[s:e].to(torch.int64),
                            lse[s:e], tgt_logit[s:e])
        loss = (lse - tgt_logit).mean()
        ctx.save_for_backward(h, weight, bias if bias is not None else torch.empty(0), targets, lse)
        ctx.has_bias = bias is not None
        return loss

    @staticmethod
    def backward(ctx, dloss: Tensor):
        lib = ext.require()
        h, weight, bias, targets, lse = ctx.saved_tensors
        has_bias = ctx.has_bias
        N, H = h.shape
        dh = torch.empty_like(h)
        dw = torch.zeros_like(weight, dtype=torch.float32)
        db = torch.zeros(weight.shape[0], dtype=torch.float32, device=h.device) if has_bias else None
        scale = (dloss / N).to(torch.float32)
        for s in range(0, N, _FusedCEFunction.CHUNK):
            e = min(N, s + _FusedCEFunction.CHUNK)
            logits = torch.matmul(h[s:e], weight.t())
            if has_bias:
                logits += bias
            # in-place: logits <- (softmax(logits) - onehot(target)) * scale
            lib.ce_dlogits(logits, targets[s:e].to(torch.int64), lse[s:e], scale)
            dlog = logits.to(h.dtype)
            dh[s:e] = torch.matmul(dlog, weight)
            dw += torch.matmul(dlog.t(), h[s:e]).to(torch.float32)
            if has_bias:
                db += dlog.sum(dim=0).to(torch.float32)
        return (dh, dw.to(weight.dtype),
                db.to(weight.dtype) if has_bias else None, None)


def tied_decoder_ce(h: Tensor, weight: Tensor, bias: Tensor | None,
                    targets: Tensor) -> Tensor:
    """h: (N, H) decoder input (already output-dropped); weight: (V, H) tied
    embedding; targets: (N,) int64. Returns scalar mean CE loss."""
    if h.is_cuda:
        return _FusedCEFunction.apply(h, weight, bias, targets)
    logits = torch.nn.functional.linear(h.float(), weight.float(),
                                        bias.float() if bias is not None else None)
    return torch.nn.functional.cross_entropy(logits, targets)


class TiedDecoderCE(torch.nn.Module):
    """Loss module bound to a LinearDecoder's weight/bias."""

    def __init__(self, decoder: torch.nn.Linear):
        super().__init__()
        self.decoder = decoder

    def forward(self, h: Tensor, targets: Tensor) -> Tensor:
        return tied_decoder_ce(h.reshape(-1, h.shape[-1]), self.decoder.weight,
                               self.decoder.bias, targets.reshape(-1))
