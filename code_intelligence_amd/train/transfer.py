"""Transfer-learning fine-tune: frozen AWD-LSTM encoder + MLP head
(BASELINE.json config 5; reference semantics: repo_mlp training on
pooled-embedding features with the encoder frozen).

MI355X design: the encoder forward runs the fused CDNA4 kernels under
no_grad; pooled features (concat-pool -> first 1600 dims) feed the torch
MLP head whose step runs FusedAdamW; with DP the head's gradients are
all-reduced by the same bucketer as LM pretraining (they're tiny — one
bucket)."""
from __future__ import annotations

from typing import Optional

import torch
from torch import nn

from ..engine.embeddings import CLASSIFIER_DIMS
from ..label.mlp import MLPHead
from ..models.awd_lstm import AWDLSTMEncoder
from ..ops.adam import FusedAdamW
from ..ops.pool import concat_pool
from ..parallel.ddp import DistributedGrads, broadcast_parameters


class TransferTrainer:
    def __init__(self, encoder: AWDLSTMEncoder, n_labels: int,
                 hidden=(600, 600), lr: float = 1e-3, wd: float = 0.01,
                 distributed: bool = False, head_dtype: torch.dtype = torch.float32):
        self.encoder = encoder.eval()
        for p in self.encoder.parameters():
            p.requires_grad_(False)
        dev = next(encoder.parameters()).device
        in_dim = min(CLASSIFIER_DIMS, 2 * encoder.emb_sz)
        self.in_dim = in_dim
        self.head = MLPHead(in_dim, hidden, n_labels).to(dev, head_dtype)
        self.opt = FusedAdamW(self.head.parameters(), lr=lr, weight_decay=wd)
        self.lossf = nn.BCEWithLogitsLoss()
        if distributed:
            broadcast_parameters(self.head)
            self.dist = DistributedGrads(self.head, bucket_mb=8)
        else:
            self.dist = None

    @torch.no_grad()
    def embed(self, ids: torch.Tensor, lengths: torch.Tensor) -> torch.Tensor:
        """(B, T) tokens -> (B, in_dim) frozen pooled features."""
        self.encoder.reset(ids.shape[0])
        _, outputs = self.encoder(ids)
        pooled = concat_pool(outputs[-1], lengths)
        return pooled[:, : self.in_dim].float()

    def train_step(self, ids: torch.Tensor, lengths: torch.Tensor,
                   targets: torch.Tensor, lr: Optional[float] = None) -> float:
        if lr is not None:
            for g in self.opt.param_groups:
                g["lr"] = lr
        feats = self.embed(ids, lengths).to(next(self.head.parameters()).dtype)
        if self.dist is not None:
            self.dist.prepare()
        self.opt.zero_grad(set_to_none=True)
        logits = self.head(feats)
        loss = self.lossf(logits, targets.to(logits.dtype))
        loss.backward()
        if self.dist is not None:
            self.dist.finalize()
        self.opt.step()
        return float(loss.detach())
