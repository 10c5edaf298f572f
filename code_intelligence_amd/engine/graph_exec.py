"""hipGraph capture of the encoder forward for fixed (B, T) buckets.

The serve path's cost at small batch is launch overhead: T timesteps x
n_layers fused-cell launches (plus projections) per request. Capturing one
(B, T) bucket as a hipGraph (torch.cuda.CUDAGraph == hipGraph on ROCm)
replays the whole sequence loop as one submission (BASELINE.json config 4).

Hidden state is reset to zeros inside the captured region's static buffers
before each replay, matching the per-request ``encoder.reset()`` contract
(reference inference.py:60,70).
"""
from __future__ import annotations

import torch

from ..models.awd_lstm import AWDLSTMEncoder


class GraphedEncoder:
    def __init__(self, encoder: AWDLSTMEncoder, B: int, T: int, device):
        self.encoder = encoder
        self.B, self.T = B, T
        self.static_ids = torch.zeros(B, T, dtype=torch.int64, device=device)
        was_training = encoder.training
        encoder.eval()
        # warmup on a side stream (required before capture)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s), torch.no_grad():
            for _ in range(2):
                encoder.reset(B)
                encoder(self.static_ids)
        torch.cuda.current_stream().wait_stream(s)
        self.graph = torch.cuda.CUDAGraph()
        with torch.no_grad():
            encoder.reset(B)
            # zero hidden INSIDE capture so replays start from clean state
            with torch.cuda.graph(self.graph):
                for h, c in encoder.hidden:
                    h.zero_()
                    c.zero_()
                _, outputs = encoder(self.static_ids)
                self.static_out = outputs[-1]
        if was_training:
            encoder.train()

    @torch.no_grad()
    def run(self, ids: torch.Tensor) -> torch.Tensor:
        assert ids.shape == (self.B, self.T), (ids.shape, self.B, self.T)
        self.static_ids.copy_(ids)
        self.graph.replay()
        return self.static_out
