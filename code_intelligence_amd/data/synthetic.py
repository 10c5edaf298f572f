"""Synthetic GitHub-issue-shaped data (there is no network for the real
GHArchive corpus — BASELINE.md row 'LM training corpus'). Token streams are
Zipf-distributed over the vocab (natural-language-like rank-frequency) with
lognormal document lengths around the reference's issue-length profile."""
from __future__ import annotations

from typing import List

import numpy as np


def synthetic_issue_tokens(n_docs: int, vocab_sz: int, seed: int = 0,
                           mean_len: float = 120.0, sigma: float = 0.8,
                           n_special: int = 9,
                           markov: bool = False) -> List[List[int]]:
    """markov=False: i.i.d. Zipf tokens (throughput benches — no learnable
    sequence structure by construction). markov=True: a sparse first-order
    Markov chain over the vocab (each token has 8 Zipf-weighted successors)
    — learnable, so convergence runs can drive perplexity well below the
    unigram entropy."""
    rng = np.random.default_rng(seed)
    lens = np.clip(rng.lognormal(np.log(mean_len), sigma, n_docs), 8, 2048).astype(int)
    V = vocab_sz - n_special
    ranks = np.arange(1, V + 1)
    probs = 1.0 / ranks ** 1.05
    probs /= probs.sum()
    docs = []
    if not markov:
        for L in lens:
            ids = rng.choice(V, size=int(L), p=probs) + n_special
            docs.append(ids.tolist())
        return docs
    # sparse transition structure: token v -> one of 8 fixed successors
    branch = 8
    succ = rng.integers(0, V, size=(V, branch))
    w = 1.0 / np.arange(1, branch + 1) ** 1.2
    w /= w.sum()
    for L in lens:
        cur = int(rng.choice(V, p=probs))
        ids = [cur]
        for _ in range(int(L) - 1):
            cur = int(succ[cur, rng.choice(branch, p=w)])
            ids.append(cur)
        docs.append([i + n_special for i in ids])
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
