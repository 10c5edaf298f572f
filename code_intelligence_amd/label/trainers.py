"""Repo-specific transfer-learning pipeline (L2/L4 glue).

Re-creates the reference's two-step KFP pipeline
(Label_Microservice/notebooks/Training_Pipeline.ipynb: 'scrape issues'
(issues_loader.py save_issue_embeddings) -> 'train' (repo_mlp.py train))
and the repo_mlp notebook semantics:

* label filter: count >= 30, excluding 'lifecycle'/'status' prefixes
  (repo_mlp.ipynb cell 21)
* one-hot targets (cells 22-24)
* MLP (600, 600) with early stopping (cell 28) — here the torch MLPHead
  trained on GPU (frozen encoder), DP-capable (BASELINE.json config 5)
* per-label threshold search + weighted AUC report (mlp.py semantics)
* artifacts at the RepoConfig object-store paths the serving worker loads
"""
from __future__ import annotations

import io
import os
import logging
import tempfile
from typing import Dict, List, Optional

import numpy as np
import yaml

from ..engine.embeddings import get_all_issue_text
from ..engine.inference import InferenceWrapper
from ..gh.gcs_util import ObjectStore, default_store
from .mlp import MLPWrapper
from .repo_config import RepoConfig

log = logging.getLogger(__name__)

MIN_LABEL_COUNT = 30
EXCLUDED_PREFIXES = ("lifecycle", "status")


def filter_labels(label_lists: List[List[str]],
                  min_count: int = MIN_LABEL_COUNT) -> List[str]:
    """Labels with >= min_count occurrences, excluding lifecycle/status."""
    from collections import Counter
    counts = Counter(l for labels in label_lists for l in labels)
    keep = [l for l, c in counts.items()
            if c >= min_count and not l.startswith(EXCLUDED_PREFIXES)]
    return sorted(keep)


def one_hot(label_lists: List[List[str]], names: List[str]) -> np.ndarray:
    idx = {n: i for i, n in enumerate(names)}
    y = np.zeros((len(label_lists), len(names)), dtype=np.float32)
    for r, labels in enumerate(label_lists):
        for l in labels:
            if l in idx:
                y[r, idx[l]] = 1.0
    return y


def save_issue_embeddings(org: str, repo: str, wrapper: InferenceWrapper,
                          store: Optional[ObjectStore] = None,
                          archive_root=None, bs: int = 100) -> str:
    """Pipeline step 1: embed all issues, store npz at RepoConfig path."""
    store = store or default_store()
    df, feats = get_all_issue_text(org, repo, wrapper, archive_root, bs=bs)
    cfg = RepoConfig(org, repo)
    buf = io.BytesIO()
    np.savez_compressed(
        buf, features=feats,
        labels=np.array([",".join(l) for l in df.get("labels", [])], dtype=object),
        issue_nums=df.get("issue_num", []).to_numpy() if not df.empty else np.array([]))
    store.write_bytes(cfg.embeddings_gcs_uri, buf.getvalue())
    log.info("wrote %d embeddings for %s/%s", len(feats), org, repo)
    return cfg.embeddings_gcs_uri


def train_repo_mlp(org: str, repo: str, store: Optional[ObjectStore] = None,
                   min_label_count: int = MIN_LABEL_COUNT,
                   device: str = "cpu",
                   hidden=(600, 600)) -> Dict:
    """Pipeline step 2: train the per-repo MLP head + thresholds, publish
    artifacts at the RepoConfig paths the worker loads."""
    store = store or default_store()
    cfg = RepoConfig(org, repo)
    with np.load(io.BytesIO(store.read_bytes(cfg.embeddings_gcs_uri)),
                 allow_pickle=True) as z:
        X = z["features"]
        label_lists = [s.split(",") if s else [] for s in z["labels"].tolist()]
    names = filter_labels(label_lists, min_label_count)
    if not names:
        raise ValueError(f"no labels with >= {min_label_count} examples")
    y = one_hot(label_lists, names)
    mlp = MLPWrapper(in_dim=X.shape[1], hidden=hidden, n_labels=len(names),
                     device=device)
    mlp.fit(X, y)
    thresholds = mlp.find_probability_thresholds(X, y)
    try:
        auc = mlp.calculate_auc(X, y)
    except ValueError:
        auc = float("nan")
    with tempfile.NamedTemporaryFile(suffix=".dpkl", delete=False) as tmp:
        mlp.save_model(tmp.name)
        store.upload(tmp.name, cfg.model_gcs_uri)
    os.unlink(tmp.name)
    store.write_bytes(cfg.labels_gcs_uri, yaml.safe_dump({
        "labels": names,
        "probability_thresholds": {int(k): v for k, v in thresholds.items()},
    }).encode())
    log.info("trained %s/%s MLP: %d labels, weighted AUC %.3f",
             org, repo, len(names), auc)
    return {"labels": names, "auc": auc, "thresholds": thresholds,
            "model_uri": cfg.model_gcs_uri}


def run_training_pipeline(org: str, repo: str, wrapper: InferenceWrapper,
                          store: Optional[ObjectStore] = None,
                          archive_root=None, device: str = "cpu") -> Dict:
    """Both steps (Training_Pipeline.ipynb @dsl.pipeline equivalent)."""
    save_issue_embeddings(org, repo, wrapper, store, archive_root)
    return train_repo_mlp(org, repo, store, device=device)


# --- universal kind model training (reference: the Keras universal model
# trained on kind labels; here the torch CNN of universal_kind_label_model) --
def kind_targets(label_lists: List[List[str]]) -> np.ndarray:
    """3-class targets (bug/feature/question) from raw labels; accepts both
    bare and 'kind/'-prefixed forms."""
    classes = ("bug", "feature", "question")
    y = np.zeros((len(label_lists), 3), dtype=np.float32)
    for r, labels in enumerate(label_lists):
        for l in labels:
            name = l.split("/", 1)[-1].lower()
            if name in classes:
                y[r, classes.index(name)] = 1.0
    return y


def train_universal_model(org: str, archive_root=None, epochs: int = 5,
                          lr: float = 1e-3, max_vocab: int = 20000,
                          device: str = "cpu", batch_size: int = 64,
                          prefix: str = ""):
    """Train the universal 3-kind classifier from the issue archive.
    Returns a ready UniversalKindLabelModel."""
    import torch
    from ..gh import bigquery
    from ..text.tokenizer import Tokenizer, Vocab
    from .universal_kind_label_model import (UniversalKindLabelModel,
                                             UniversalKindNet)

    df = bigquery.get_issues(org, archive_root=archive_root)
    if df.empty:
        raise ValueError(f"no archived issues for org {org}")
    tok = Tokenizer()
    docs = [tok.process_text(f"{t}\n{b}")[:256]
            for t, b in zip(df["title"], df["body"])]
    vocab = Vocab.create(docs, max_vocab=max_vocab, min_freq=1)
    y = torch.tensor(kind_targets(df["labels"].tolist()))
    ids = [torch.tensor(vocab.numericalize(d) or [0]) for d in docs]
    maxlen = max(len(i) for i in ids)
    X = torch.full((len(ids), maxlen), 1, dtype=torch.int64)  # pad=1
    for r, seq in enumerate(ids):
        X[r, :len(seq)] = seq
    net = UniversalKindNet(len(vocab)).to(device)
    opt = torch.optim.AdamW(net.parameters(), lr=lr)
    lossf = torch.nn.BCEWithLogitsLoss()
    net.train()
    X, y = X.to(device), y.to(device)
    for _ in range(epochs):
        perm = torch.randperm(len(X), device=device)
        for s in range(0, len(X), batch_size):
            b = perm[s:s + batch_size]
            opt.zero_grad()
            loss = lossf(net(X[b]), y[b])
            loss.backward()
            opt.step()
    net.eval()
    model = UniversalKindLabelModel(net.cpu(), vocab, prefix=prefix)
    log.info("trained universal model on %d issues (final loss %.4f)",
             len(X), float(loss.detach()))
    return model
