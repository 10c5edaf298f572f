"""Ensemble: run member models serially, keep per-label max probability
(reference: py/label_microservice/combined_model.py:15-54)."""
from __future__ import annotations

from typing import Dict, List, Optional

from .models import IssueLabelModel


class CombinedLabelModels(IssueLabelModel):
    def __init__(self, models: List[IssueLabelModel]):
        self.models = models

    def predict_issue_labels(self, org: str, repo: str, title: str,
                             text: List[str], context: Optional[dict] = None
                             ) -> Dict[str, float]:
        preds = []
        for m in self.models:
            preds.append(m.predict_issue_labels(org, repo, title, text, context))
        return self._combine_predictions(preds)

    @staticmethod
    def _combine_predictions(predictions, right: Optional[Dict[str, float]] = None
                             ) -> Dict[str, float]:
        """Per-label max merge. Accepts either a list of prediction dicts or
        the reference's pairwise form ``_combine_predictions(left, right)``
        (combined_model.py:41-54, exercised by its combined_model_test)."""
        if right is not None or isinstance(predictions, dict):
            predictions = [predictions or {}, right or {}]
        out: Dict[str, float] = {}
        for p in predictions:
            for label, prob in (p or {}).items():
                if label not in out or prob > out[label]:
                    out[label] = prob
        return out
