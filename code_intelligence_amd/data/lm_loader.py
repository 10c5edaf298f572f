"""Concat-stream BPTT language-model loader (fastai LMDataLoader equivalent).

Reference semantics: documents are concatenated into one
token stream, split into ``bs`` parallel streams, and yielded as
(x (bs, bptt), y = x shifted by one) windows in order — hidden state is
carried across windows (train.py:63-64: bptt 63-70 truncated BPTT).

MI355X adaptation: windows are materialized as pinned int64 tensors and the
stream layout is computed once per epoch (cheap reshuffle of document
order); no per-batch tokenization.
"""
from __future__ import annotations

import math
from typing import Iterator, List, Optional, Tuple

import torch
from torch import Tensor


class LMStreamLoader:
    def __init__(self, docs: List[List[int]], bs: int, bptt: int,
                 bos_idx: Optional[int] = 2, shuffle: bool = True,
                 seed: int = 0, device: Optional[torch.device] = None):
        if bs < 1 or bptt < 1:
            raise ValueError(f"bs and bptt must be >= 1 (got {bs}, {bptt})")
        self.docs, self.bs, self.bptt = docs, bs, bptt
        self.bos_idx, self.shuffle, self.seed = bos_idx, shuffle, seed
        self.device = torch.device(device) if device is not None else None
        self.epoch = 0

    def _stream(self) -> Tensor:
        order = list(range(len(self.docs)))
        if self.shuffle:
            g = torch.Generator().manual_seed(self.seed + self.epoch)
            order = torch.randperm(len(self.docs), generator=g).tolist()
        parts = []
        for i in order:
            if self.bos_idx is not None:
                parts.append(torch.tensor([self.bos_idx], dtype=torch.int64))
            parts.append(torch.as_tensor(self.docs[i], dtype=torch.int64))
        return torch.cat(parts) if parts else torch.empty(0, dtype=torch.int64)

    def __len__(self) -> int:
        total = sum(len(d) for d in self.docs) + \
            (len(self.docs) if self.bos_idx is not None else 0)
        per_stream = total // self.bs
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
                x = x.to(self.device, non_blocking=True)
                y = y.to(self.device, non_blocking=True)
            yield x.contiguous(), y.contiguous()
