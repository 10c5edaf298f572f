"""Model interface (reference: py/label_microservice/models.py:5-29)."""
from __future__ import annotations

import abc
from typing import Dict, Optional


class IssueLabelModel(abc.ABC):
    """Predict labels for a GitHub issue."""

    @abc.abstractmethod
    def predict_issue_labels(self, org: str, repo: str, title: str,
                             text: list, context: Optional[dict] = None
                             ) -> Dict[str, float]:
        """Return {label: probability}; only labels worth applying."""
        raise NotImplementedError
