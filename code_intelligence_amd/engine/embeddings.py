"""Repo bulk-embedding helpers (reference: py/code_intelligence/embeddings.py).

The reference scrapes github.com HTML (deprecated even there) and
bulk-embeds a repo's issues, returning ``features[:, :1600]`` — the
repo-specific classifiers consume only the first 1600 dims (mean+max
pools; the truncation at embeddings.py:116 drops the "last" pool).

Offline redesign: issues come from the local archive (gh/bigquery.py)
instead of HTML scraping; the embedding path is the MI355X engine."""
from __future__ import annotations

import logging
from typing import Optional, Tuple

import numpy as np
import pandas as pd

from ..gh import bigquery
from .inference import InferenceWrapper

log = logging.getLogger(__name__)

CLASSIFIER_DIMS = 1600  # 2 * emb_sz pools (mean+max) at the deployed config


def find_max_issue_num(df: pd.DataFrame) -> int:
    """Largest issue number present (reference embeddings.py:14-32 probes
    the website; here the archive is the source of truth)."""
    if df.empty:
        return 0
    return int(df["issue_num"].max())


def get_issue_text(org: str, repo: str, issue_num: int,
                   archive_root=None) -> Optional[dict]:
    """{title, body} for one issue from the archive (embeddings.py:36-75)."""
    df = bigquery.get_issues(org, archive_root=archive_root)
    row = df[(df["repo"] == repo) & (df["issue_num"] == issue_num)]
    if row.empty:
        return None
    r = row.iloc[0]
    return {"title": r["title"], "body": r["body"]}


def get_all_issue_text(org: str, repo: str, inf_wrapper: InferenceWrapper,
                       archive_root=None, bs: int = 100
                       ) -> Tuple[pd.DataFrame, np.ndarray]:
    """Embed every archived issue of org/repo; returns (issues_df,
    (N, 1600) features) — truncation semantics of embeddings.py:116."""
    df = bigquery.get_issues(org, archive_root=archive_root)
    df = df[df["repo"] == repo].reset_index(drop=True)
    keep = 2 * inf_wrapper.emb_sz  # mean+max pools = 1600 at emb_sz=800
    if df.empty:
        return df, np.zeros((0, keep), dtype=np.float32)
    feats = inf_wrapper.df_to_embedding(df, bs=bs)
    return df, feats[:, :keep]


def load_model_artifact(model_path: str, **kw) -> InferenceWrapper:
    """reference embeddings.py:126-150: construct the inference wrapper
    from a model artifact directory."""
    return InferenceWrapper(model_path=model_path, **kw)
