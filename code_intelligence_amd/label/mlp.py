"""Transfer-learning MLP head (K9) + the reference MLPWrapper contract.

Reference: py/label_microservice/mlp.py (sklearn MLPClassifier(600,600)
wrapper with per-label threshold search on the P-R curve — keep the
max-precision threshold among points with precision >= 0.7 AND
recall >= 0.5, else None meaning 'never predict' — mlp.py:19-20,65-98;
grid_search 100-114; dill save/load 116-138; weighted-avg AUC 140-163).

MI355X design: the head itself is a torch module (``MLPHead``) so the
frozen-encoder fine-tune runs on GPU under DP=8 (BASELINE.json config 5);
``MLPWrapper`` keeps the reference's fit/predict/threshold API on top of
it. Metrics use sklearn (CPU-side, tiny)."""
from __future__ import annotations

import pickle
from typing import Dict, List, Optional, Sequence

import numpy as np
import torch
from torch import nn


def _to_t(a, device) -> torch.Tensor:
    """float32 tensor from array-like; copies only when non-writable
    (np.load/npz arrays are read-only and trip torch.as_tensor)."""
    arr = np.asarray(a, dtype=np.float32)
    if not arr.flags.writeable:
        arr = arr.copy()
    return torch.as_tensor(arr, device=device)


class MLPHead(nn.Module):
    """input (1600-d truncated embedding) -> hidden(600) -> hidden(600) ->
    sigmoid multi-label output (repo_mlp.ipynb cell 28 shape)."""

    def __init__(self, in_dim: int = 1600, hidden: Sequence[int] = (600, 600),
                 n_labels: int = 1):
        super().__init__()
        layers: List[nn.Module] = []
        d = in_dim
        for h in hidden:
            layers += [nn.Linear(d, h), nn.ReLU()]
            d = h
        layers.append(nn.Linear(d, n_labels))
        self.net = nn.Sequential(*layers)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return self.net(x)


class MLPWrapper:
    precision_threshold = 0.7   # mlp.py:19
    recall_threshold = 0.5      # mlp.py:20

    def __init__(self, clf: Optional[MLPHead] = None, in_dim: int = 1600,
                 hidden: Sequence[int] = (600, 600), n_labels: int = 1,
                 max_iter: int = 3000, lr: float = 1e-3, device: str = "cpu",
                 early_stopping: bool = True,
                 precision_threshold: float = 0.7,
                 recall_threshold: float = 0.5):
        self.in_dim, self.hidden, self.n_labels = in_dim, tuple(hidden), n_labels
        self.max_iter, self.lr = max_iter, lr
        self.device = torch.device(device)
        self.clf = clf or MLPHead(in_dim, hidden, n_labels)
        self.early_stopping = early_stopping
        # per-instance thresholds (reference mlp.py:19-20,37-38 ctor kwargs)
        self.precision_threshold = precision_threshold
        self.recall_threshold = recall_threshold
        self.probability_thresholds: Dict[int, Optional[float]] = {}
        # per-label metrics at the chosen threshold (reference mlp.py:41-43)
        self.precisions: Dict[int, float] = {}
        self.recalls: Dict[int, float] = {}
        self.total_labels_count: Optional[int] = None

    # --- training ---------------------------------------------------------
    def fit(self, X: np.ndarray, y: np.ndarray, epochs: Optional[int] = None,
            batch_size: int = 200, verbose: bool = False) -> "MLPWrapper":
        X_t = _to_t(X, self.device)
        y_t = _to_t(y, self.device)
        if y_t.dim() == 1:
            y_t = y_t.unsqueeze(1)
        self.clf = self.clf.to(self.device)
        opt = torch.optim.AdamW(self.clf.parameters(), lr=self.lr)
        lossf = nn.BCEWithLogitsLoss()
        n = X_t.shape[0]
        n_valid = max(1, int(0.1 * n)) if self.early_stopping and n >= 10 else 0
        perm = torch.randperm(n, device=self.device)
        X_t, y_t = X_t[perm], y_t[perm]
        Xv, yv = X_t[:n_valid], y_t[:n_valid]
        Xtr, ytr = X_t[n_valid:], y_t[n_valid:]
        best, wait, patience = float("inf"), 0, 10
        epochs = epochs or max(1, min(200, self.max_iter // max(1, len(Xtr) // batch_size + 1)))
        self.clf.train()
        for ep in range(epochs):
            for s in range(0, len(Xtr), batch_size):
                xb, yb = Xtr[s:s + batch_size], ytr[s:s + batch_size]
                opt.zero_grad()
                loss = lossf(self.clf(xb), yb)
                loss.backward()
                opt.step()
            if n_valid:
                with torch.no_grad():
                    self.clf.eval()
                    vl = float(lossf(self.clf(Xv), yv))
                    self.clf.train()
                if vl < best - 1e-4:
                    best, wait = vl, 0
                else:
                    wait += 1
                    if wait >= patience:
                        break
        self.clf.eval()
        return self

    def predict_probabilities(self, X: np.ndarray) -> np.ndarray:
        X_t = _to_t(X, self.device)
        with torch.no_grad():
            p = torch.sigmoid(self.clf.to(self.device)(X_t))
        return p.cpu().numpy()

    predict_proba = predict_probabilities

    # --- threshold search (mlp.py:65-98 semantics) ------------------------
    def find_probability_thresholds(self, X: np.ndarray, y: np.ndarray,
                                    test_size: float = 0.3
                                    ) -> Dict[int, Optional[float]]:
        """Reference semantics (mlp.py:65-98): hold out ``test_size`` of the
        data (random_state 1234), REFIT on the rest, and search each label's
        P-R curve on the held-out part; stores ``probability_thresholds``,
        ``precisions``, ``recalls`` (0.0 when no point qualifies) and
        ``total_labels_count``. Pass ``test_size=0`` to search on the given
        data with the already-fitted model instead."""
        from sklearn.metrics import precision_recall_curve
        y = np.asarray(y).copy()  # torch.as_tensor needs writable arrays
        if y.ndim == 1:
            y = y[:, None]
        X = np.asarray(X)
        if test_size and len(X) >= 4:
            from sklearn.model_selection import train_test_split
            X_tr, X_te, y_tr, y_te = train_test_split(
                X, y, test_size=test_size, random_state=1234)
            self.fit(X_tr, y_tr)
        else:
            X_te, y_te = X, y
        probs = self.predict_probabilities(X_te)
        out: Dict[int, Optional[float]] = {}
        self.precisions, self.recalls = {}, {}
        self.total_labels_count = y_te.shape[1]
        for li in range(y_te.shape[1]):
            yt, pp = y_te[:, li], probs[:, li]
            best_p, best_r, best_thr = 0.0, 0.0, None
            if 0 < yt.sum() < len(yt):
                prec, rec, thr = precision_recall_curve(yt, pp)
                # sklearn pairing: precision[i]/recall[i] are the metrics of
                # predicting score >= thresholds[i] (last P/R point has no
                # threshold) — reference mlp.py:84 zips [:-1] with thresholds
                for p_, r_, t_ in zip(prec[:-1], rec[:-1], thr):
                    if p_ >= self.precision_threshold                             and r_ >= self.recall_threshold and p_ > best_p:
                        best_p, best_r, best_thr = p_, r_, float(t_)
            out[li] = best_thr  # None => never predict this label
            self.precisions[li] = float(best_p)
            self.recalls[li] = float(best_r)
        self.probability_thresholds = out
        return out

    # --- grid search (mlp.py:100-114) -------------------------------------
    def grid_search(self, X: np.ndarray, y: np.ndarray,
                    param_grid: Optional[dict] = None) -> dict:
        from sklearn.metrics import roc_auc_score
        param_grid = param_grid or {
            "hidden": [(600, 600), (400, 400), (800,)],
            "lr": [1e-3, 3e-4],
        }
        X = np.asarray(X)
        y2 = np.asarray(y)
        if y2.ndim == 1:
            y2 = y2[:, None]
        n_valid = max(1, len(X) // 5)
        best_auc, best_params, best_clf = -1.0, None, None
        from itertools import product
        keys = sorted(param_grid)
        for combo in product(*(param_grid[k] for k in keys)):
            params = dict(zip(keys, combo))
            cand = MLPWrapper(in_dim=self.in_dim, n_labels=self.n_labels,
                              hidden=params.get("hidden", self.hidden),
                              lr=params.get("lr", self.lr),
                              device=str(self.device))
            cand.fit(X[n_valid:], y2[n_valid:])
            probs = cand.predict_probabilities(X[:n_valid])
            try:
                auc = roc_auc_score(y2[:n_valid], probs, average="weighted")
            except ValueError:
                auc = 0.5
            if auc > best_auc:
                best_auc, best_params, best_clf = auc, params, cand.clf
        if best_clf is not None:
            self.clf = best_clf
        return {"best_params": best_params, "best_auc": best_auc}

    # --- persistence (mlp.py:116-138; .dpkl artifacts) --------------------
    def save_model(self, path, thresholds_path: Optional[str] = None) -> None:
        state = {
            "in_dim": self.in_dim, "hidden": self.hidden,
            "n_labels": self.n_labels,
            "state_dict": {k: v.cpu() for k, v in self.clf.state_dict().items()},
            "probability_thresholds": self.probability_thresholds,
        }
        with open(path, "wb") as f:
            pickle.dump(state, f)
        if thresholds_path:
            import yaml
            with open(thresholds_path, "w") as f:
                yaml.safe_dump({"probability_thresholds":
                                self.probability_thresholds}, f)

    @classmethod
    def load_model(cls, path, device: str = "cpu") -> "MLPWrapper":
        with open(path, "rb") as f:
            state = pickle.load(f)
        w = cls(in_dim=state["in_dim"], hidden=state["hidden"],
                n_labels=state["n_labels"], device=device)
        w.clf.load_state_dict(state["state_dict"])
        w.clf.eval()
        w.probability_thresholds = state.get("probability_thresholds", {})
        return w

    # --- metrics (mlp.py:140-163) -----------------------------------------
    def calculate_auc(self, X: np.ndarray, y: np.ndarray) -> float:
        from sklearn.metrics import roc_auc_score
        probs = self.predict_probabilities(X)
        y = np.asarray(y)
        if y.ndim == 1:
            y = y[:, None]
        return float(roc_auc_score(y, probs, average="weighted"))
