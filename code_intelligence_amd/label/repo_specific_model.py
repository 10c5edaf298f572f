"""Repo-specific transfer model (reference: repo_specific_model.py).

Loads the per-repo MLP + label/threshold yaml from the object store
(RepoConfig naming), fetches the issue embedding from the embedding REST
service (POST {endpoint}/text -> np.frombuffer(content, '<f4')[:1600] —
repo_specific_model.py:154-183; the [:1600] truncation keeps only the
mean+max pools, embeddings.py:116), applies per-label probability
thresholds (None => never predict)."""
from __future__ import annotations

import logging
import os
from typing import Dict, List, Optional

import numpy as np
import yaml

from ..gh.gcs_util import ObjectStore, default_store
from .mlp import MLPWrapper
from .models import IssueLabelModel
from .repo_config import RepoConfig

log = logging.getLogger(__name__)

DEFAULT_EMBEDDING_ENDPOINT = "http://issue-embedding-server"  # reference :16
EMBEDDING_DIM = 1600


class RepoSpecificLabelModel(IssueLabelModel):
    def __init__(self, mlp: MLPWrapper, label_names: List[str],
                 thresholds: Dict[int, Optional[float]],
                 embedding_api_endpoint: Optional[str] = None,
                 session=None):
        self.mlp = mlp
        self.label_names = label_names
        self.thresholds = thresholds
        if embedding_api_endpoint is None:
            # deploy-time wiring like the reference's k8s service DNS name
            embedding_api_endpoint = os.environ.get(
                "ISSUE_EMBEDDING_SERVICE", DEFAULT_EMBEDDING_ENDPOINT)
        self.endpoint = embedding_api_endpoint.rstrip("/")
        if session is None:
            import requests
            session = requests.Session()
        self.session = session

    @classmethod
    def from_repo(cls, repo_owner: str, repo_name: str,
                  embedding_api_endpoint: Optional[str] = None,
                  store: Optional[ObjectStore] = None, session=None
                  ) -> "RepoSpecificLabelModel":
        """reference repo_specific_model.py:32-88."""
        store = store or default_store()
        cfg = RepoConfig(repo_owner, repo_name)
        import tempfile
        with tempfile.NamedTemporaryFile(suffix=".dpkl") as tmp:
            store.download(cfg.model_gcs_uri, tmp.name)
            mlp = MLPWrapper.load_model(tmp.name)
        meta = yaml.safe_load(store.read_bytes(cfg.labels_gcs_uri))
        labels = meta["labels"]
        thresholds = {int(k): v for k, v in
                      (meta.get("probability_thresholds") or {}).items()}
        return cls(mlp, labels, thresholds, embedding_api_endpoint, session)

    def _get_issue_embedding(self, title: str, text: List[str]) -> Optional[np.ndarray]:
        """POST /text; None on non-200 (reference returns None on 404 so the
        worker degrades gracefully — repo_specific_model_test.py:15-47)."""
        try:
            r = self.session.post(f"{self.endpoint}/text",
                                  json={"title": title,
                                        "body": "\n".join(text or [])})
        except Exception:
            log.exception("embedding service unreachable")
            return None
        if r.status_code != 200:
            log.warning("embedding service returned %s", r.status_code)
            return None
        # keep the mean+max pools the classifier was trained on
        # (1600 at the deployed config — embeddings.py:116)
        keep = getattr(self.mlp, "in_dim", EMBEDDING_DIM)
        return np.frombuffer(r.content, dtype="<f4")[:keep]

    def predict_issue_labels(self, org: str, repo: str, title: str,
                             text: List[str], context: Optional[dict] = None
                             ) -> Dict[str, float]:
        emb = self._get_issue_embedding(title, text)
        if emb is None:
            return {}
        probs = self.mlp.predict_probabilities(emb[None, :])[0]
        out: Dict[str, float] = {}
        for i, name in enumerate(self.label_names):
            thr = self.thresholds.get(i)
            if thr is None:
                continue  # never predict (no satisfactory P/R point)
            if probs[i] >= thr:
                out[name] = float(probs[i])
        return out
