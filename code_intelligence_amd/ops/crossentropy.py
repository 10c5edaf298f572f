"""Fused tied-decoder softmax + cross-entropy (K6) — the FLOPs king.

Reference semantics: fastai LinearDecoder + FlattenedLoss(CrossEntropy)
over a 60k vocab (train.py:70, tie_weights/out_bias). Materializing
(B*T, V) logits for the whole batch costs ~31 GB at the bench shape —
instead the op chunks over rows (CHUNK=16384 measured fastest on MI355X,
scripts/gemm_probe.py: 1.05 PF for the chunk GEMM): per chunk a plain
hipBLASLt GEMM fills a preallocated logits tile, a HIP kernel reduces it
to (logsumexp, target-logit) in one pass — the decoder BIAS is folded
into that kernel so the (chunk, 60k) bias broadcast-add never
materializes — and backward RECOMPUTES the chunk's logits, transforming
them in place to dlogits = (softmax - onehot)/N. Only O(B*T) state (lse)
is saved between forward and backward.

CPU path: plain F.cross_entropy composition (numerics reference)."""
from __future__ import annotations

import torch
from torch import Tensor

from . import extension as ext

__all__ = ["tied_decoder_ce", "TiedDecoderCE"]

_EMPTY = {}


def _empty_f32(device) -> Tensor:
    key = str(device)
    if key not in _EMPTY:
        _EMPTY[key] = torch.empty(0, dtype=torch.float32, device=device)
    return _EMPTY[key]


class _FusedCEFunction(torch.autograd.Function):
    CHUNK = 16384

    @staticmethod
    def forward(ctx, h: Tensor, weight: Tensor, bias: Tensor, targets: Tensor):
        lib = ext.require()
        N, H = h.shape
        V = weight.shape[0]
        C = _FusedCEFunction.CHUNK
        lse = torch.empty(N, dtype=torch.float32, device=h.device)
        tgt_logit = torch.empty(N, dtype=torch.float32, device=h.device)
        b32 = bias.to(torch.float32) if bias is not None else _empty_f32(h.device)
        tgt64 = targets.to(torch.int64)
        logits_buf = torch.empty(min(C, N), V, dtype=h.dtype, device=h.device)
        w_t = weight.t()
        for s in range(0, N, C):
            e = min(N, s + C)
            logits = logits_buf[: e - s]
            torch.mm(h[s:e], w_t, out=logits)
            lib.ce_rowstats(logits, tgt64[s:e], b32, lse[s:e], tgt_logit[s:e])
        loss = (lse - tgt_logit).mean()
        ctx.save_for_backward(h, weight, b32, tgt64, lse)
        ctx.has_bias = bias is not None
        return loss

    @staticmethod
    def backward(ctx, dloss: Tensor):
        lib = ext.require()
        h, weight, b32, targets, lse = ctx.saved_tensors
        has_bias = ctx.has_bias
        N, H = h.shape
        V = weight.shape[0]
        C = _FusedCEFunction.CHUNK
        dh = torch.empty_like(h)
        dw = torch.zeros_like(weight, dtype=torch.float32)
        db = torch.zeros(V, dtype=torch.float32, device=h.device) if has_bias else None
        scale = (dloss / N).to(torch.float32).reshape(1)
        logits_buf = torch.empty(min(C, N), V, dtype=h.dtype, device=h.device)
        w_t = weight.t()
        bias_arg = b32 if has_bias else _empty_f32(h.device)
        for s in range(0, N, C):
            e = min(N, s + C)
            dlog = logits_buf[: e - s]
            torch.mm(h[s:e], w_t, out=dlog)
            # in-place: dlog <- (softmax(dlog + bias) - onehot) * scale
            lib.ce_dlogits(dlog, targets[s:e], bias_arg, lse[s:e], scale)
            torch.mm(dlog, weight, out=dh[s:e])
            dw += torch.mm(dlog.t(), h[s:e])
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
