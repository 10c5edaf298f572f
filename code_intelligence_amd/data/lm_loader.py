"""Concat-stream BPTT language-model loader (fastai LMDataLoader equivalent).

Reference semantics: documents are concatenated into one
token stream, split into ``bs`` parallel streams, and yielded as
(x (bs, bptt), y = x shifted by one) windows in order — hidden state is
carried across windows (train.py:63-64: bptt 63-70 truncated BPTT).

MI355X adaptations:
* windows come from one pinned int64 epoch matrix (real async H2D);
* the corpus may be COMPACT — ``{"flat": int32 (total,), "offsets":
  int64 (n_docs+1,)}`` (what scripts/prepare_data.py writes). At the
  reference's 16.7M-issue scale a Python list-of-lists costs ~8x the RAM
  of the flat tensor, and the per-epoch reshuffle here is vectorized
  gather over doc chunks instead of millions of per-doc tensor ops.
"""
from __future__ import annotations

from typing import Iterator, List, Optional, Tuple, Union

import torch
from torch import Tensor

CompactCorpus = dict  # {"flat": int tensor, "offsets": int64 tensor}


def docs_to_compact(docs: List[List[int]]) -> CompactCorpus:
    """List-of-lists -> {flat int32, offsets int64} corpus."""
    lengths = torch.tensor([len(d) for d in docs], dtype=torch.int64)
    offsets = torch.zeros(len(docs) + 1, dtype=torch.int64)
    torch.cumsum(lengths, 0, out=offsets[1:])
    flat = torch.empty(int(offsets[-1]), dtype=torch.int32)
    for i, d in enumerate(docs):
        if d:
            flat[offsets[i]: offsets[i + 1]] = torch.as_tensor(
                d, dtype=torch.int32)
    return {"flat": flat, "offsets": offsets}


def split_compact(c: CompactCorpus, n_first: int
                  ) -> Tuple[CompactCorpus, CompactCorpus]:
    """Split into (first n_first docs, rest) without copying flat data."""
    off = c["offsets"]
    cut = int(off[n_first])
    a = {"flat": c["flat"][:cut], "offsets": off[: n_first + 1].clone()}
    b = {"flat": c["flat"][cut:],
         "offsets": off[n_first:].clone() - cut}
    return a, b


class LMStreamLoader:
    def __init__(self, docs: Union[List[List[int]], CompactCorpus],
                 bs: int, bptt: int,
                 bos_idx: Optional[int] = 2, shuffle: bool = True,
                 seed: int = 0, device: Optional[torch.device] = None):
        if bs < 1 or bptt < 1:
            raise ValueError(f"bs and bptt must be >= 1 (got {bs}, {bptt})")
        if isinstance(docs, dict):
            self._flat = docs["flat"]
            self._offsets = docs["offsets"].to(torch.int64)
        else:
            c = docs_to_compact(docs)
            self._flat, self._offsets = c["flat"], c["offsets"]
        self.bs, self.bptt = bs, bptt
        self.bos_idx, self.shuffle, self.seed = bos_idx, shuffle, seed
        self.device = torch.device(device) if device is not None else None
        self.epoch = 0

    @property
    def n_docs(self) -> int:
        return self._offsets.numel() - 1

    @property
    def total_tokens(self) -> int:
        return int(self._offsets[-1]) + \
            (self.n_docs if self.bos_idx is not None else 0)

    def _stream(self) -> Tensor:
        n = self.n_docs
        if n == 0:
            return torch.empty(0, dtype=torch.int32)
        if self.shuffle:
            g = torch.Generator().manual_seed(self.seed + self.epoch)
            order = torch.randperm(n, generator=g)
        else:
            order = torch.arange(n)
        lengths = self._offsets[1:] - self._offsets[:-1]
        L = lengths[order]
        bos = 1 if self.bos_idx is not None else 0
        seg = L + bos
        dst_start = torch.zeros(n + 1, dtype=torch.int64)
        torch.cumsum(seg, 0, out=dst_start[1:])
        # int32 epoch stream (windows cast to int64 at yield): halves the
        # resident stream at reference scale (~9 vs ~19 GB for 2.3B tokens)
        out = torch.empty(int(dst_start[-1]), dtype=torch.int32)
        src_start = self._offsets[:-1][order]
        # vectorized permuted-concat in doc chunks (bounds index-tensor RAM)
        CH = 262144
        for s in range(0, n, CH):
            e = min(n, s + CH)
            seg_c = seg[s:e]
            base = int(dst_start[s])
            pos = torch.arange(int(dst_start[e]) - base, dtype=torch.int64)
            doc = torch.repeat_interleave(
                torch.arange(e - s, dtype=torch.int64), seg_c)
            local = pos - (dst_start[s:e][doc] - base)
            src = src_start[s:e][doc] + (local - bos)
            vals = self._flat[src.clamp_min_(0)].to(torch.int32)
            if bos:
                vals[local == 0] = self.bos_idx
            out[base: base + pos.numel()] = vals
        return out

    def __len__(self) -> int:
        per_stream = self.total_tokens // self.bs
        return max(0, (per_stream - 1) // self.bptt)

    def __iter__(self) -> Iterator[Tuple[Tensor, Tensor]]:
        stream = self._stream()
        self.epoch += 1
        per = stream.numel() // self.bs
        if per < 2:
            return
        mat = stream[: per * self.bs].view(self.bs, per)
        if self.device is not None and self.device.type == "cuda" \
                and torch.cuda.is_available():
            # one pinned epoch matrix => every window's .to(non_blocking=True)
            # is a real async H2D copy instead of a silent sync one
            mat = mat.contiguous().pin_memory()
        n_batches = (per - 1) // self.bptt
        for k in range(n_batches):
            s = k * self.bptt
            x = mat[:, s: s + self.bptt]
            y = mat[:, s + 1: s + self.bptt + 1]
            if self.device is not None:
                # copy the int32 window first (async from the pinned mat),
                # widen to int64 on-device
                x = x.to(self.device, non_blocking=True)
                y = y.to(self.device, non_blocking=True)
            yield x.contiguous().to(torch.int64), \
                y.contiguous().to(torch.int64)
