"""Synthetic GitHub-issue-shaped data (there is no network for the real
GHArchive corpus — BASELINE.md row 'LM training corpus'). Token streams are
Zipf-distributed over the vocab (natural-language-like rank-frequency) with
lognormal document lengths around the reference's issue-length profile."""
from __future__ import annotations

from typing import List

import numpy as np


def synthetic_issue_tokens(n_docs: int, vocab_sz: int, seed: int = 0,
                           mean_len: float = 120.0, sigma: float = 0.8,
                           n_special: int = 9) -> List[List[int]]:
    rng = np.random.default_rng(seed)
    lens = np.clip(rng.lognormal(np.log(mean_len), sigma, n_docs), 8, 2048).astype(int)
    docs = []
    # Zipf over the non-special vocab ids
    ranks = np.arange(1, vocab_sz - n_special + 1)
    probs = 1.0 / ranks ** 1.05
    probs /= probs.sum()
    for L in lens:
        ids = rng.choice(len(ranks), size=int(L), p=probs) + n_special
        docs.append(ids.tolist())
    return docs


_WORDS = ("the fix bug error crash when run build test model train gpu issue "
          "label feature request install version update doc link code python "
          "fails kubeflow pipeline deploy k8s container image notebook data").split()


def synthetic_issue_texts(n: int, seed: int = 0) -> List[dict]:
    """(title, body) dicts for serve-path benches/tests."""
    rng = np.random.default_rng(seed)
    out = []
    for _ in range(n):
        title = " ".join(rng.choice(_WORDS, size=int(rng.integers(3, 10))))
        body = " ".join(rng.choice(_WORDS, size=int(rng.integers(20, 200))))
        out.append({"title": title, "body": body})
    return out
