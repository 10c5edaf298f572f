"""Universal 3-kind model: bug / feature / question.

Reference: py/label_microservice/universal_kind_label_model.py — a Keras
CNN over ktext-preprocessed title+body with thresholds 0.52 (bug),
0.52 (feature), 0.60 (question) and optional 'kind/' prefixing.

MI355X re-design: a compact torch text classifier (embedding -> 1D convs
-> max-pool -> linear) over this framework's own tokenizer — no TF, no
per-call graph reload (the reference reloads the Keras model inside a
fresh tf.Graph on EVERY predict as a thread-affinity workaround,
universal_kind_label_model.py:86-92; torch needs no such hack). Artifacts
are a plain state-dict + vocab."""
from __future__ import annotations

import json
from pathlib import Path
from typing import Dict, List, Optional

import torch
from torch import nn

from ..text.tokenizer import Tokenizer, Vocab, defaults_specials
from .models import IssueLabelModel


class UniversalKindNet(nn.Module):
    def __init__(self, vocab_sz: int, emb_dim: int = 64, n_classes: int = 3,
                 channels: int = 128):
        super().__init__()
        self.emb = nn.Embedding(vocab_sz, emb_dim, padding_idx=1)
        self.convs = nn.ModuleList([
            nn.Conv1d(emb_dim, channels, k, padding=k // 2) for k in (3, 5, 7)])
        self.out = nn.Linear(3 * channels, n_classes)

    def forward(self, ids: torch.Tensor) -> torch.Tensor:
        e = self.emb(ids).transpose(1, 2)           # (B, E, T)
        feats = [torch.relu(c(e)).amax(dim=2) for c in self.convs]
        return self.out(torch.cat(feats, dim=1))    # logits (B, 3)


class UniversalKindLabelModel(IssueLabelModel):
    CLASS_NAMES = ["bug", "feature", "question"]

    def __init__(self, net: Optional[UniversalKindNet] = None,
                 vocab: Optional[Vocab] = None, max_len: int = 512,
                 prefix: str = "", device: str = "cpu"):
        # thresholds: reference universal_kind_label_model.py:50-51
        self.thresholds = {"bug": 0.52, "feature": 0.52, "question": 0.60}
        self.vocab = vocab or Vocab(defaults_specials)
        self.net = net or UniversalKindNet(max(len(self.vocab), 16))
        self.device = torch.device(device)
        self.net = self.net.to(self.device).eval()
        self.tokenizer = Tokenizer()
        self.max_len = max_len
        self.prefix = prefix  # e.g. 'kind/' for kubeflow-style labels

    @classmethod
    def load(cls, path, device: str = "cpu") -> "UniversalKindLabelModel":
        root = Path(path)
        vocab = Vocab.load(root / "vocab.json")
        cfg = json.loads((root / "config.json").read_text())
        net = UniversalKindNet(cfg.get("vocab_sz", len(vocab)),
                               cfg.get("emb_dim", 64),
                               cfg.get("n_classes", 3), cfg.get("channels", 128))
        net.load_state_dict(torch.load(root / "model.pth", map_location="cpu",
                                       weights_only=True))
        return cls(net, vocab, prefix=cfg.get("prefix", ""), device=device)

    def save(self, path) -> None:
        root = Path(path)
        root.mkdir(parents=True, exist_ok=True)
        self.vocab.save(root / "vocab.json")
        (root / "config.json").write_text(json.dumps({
            "vocab_sz": self.net.emb.num_embeddings,
            "emb_dim": self.net.emb.embedding_dim,
            "n_classes": self.net.out.out_features,
            "channels": self.net.convs[0].out_channels,
            "prefix": self.prefix}))
        torch.save(self.net.state_dict(), root / "model.pth")

    def _encode(self, title: str, text: List[str]) -> torch.Tensor:
        body = "\n".join(text or [])
        toks = self.tokenizer.process_text(f"{title}\n{body}")[: self.max_len]
        ids = self.vocab.numericalize(toks) or [0]
        return torch.tensor([ids], dtype=torch.int64, device=self.device)

    @torch.no_grad()
    def predict_issue_labels(self, org: str, repo: str, title: str,
                             text: List[str], context: Optional[dict] = None
                             ) -> Dict[str, float]:
        probs = torch.sigmoid(self.net(self._encode(title, text)))[0]
        out: Dict[str, float] = {}
        for i, name in enumerate(self.CLASS_NAMES):
            p = float(probs[i])
            if p >= self.thresholds[name]:
                out[self.prefix + name] = p
        return out
