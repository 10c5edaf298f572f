"""Remote text-classification model (reference: automl_model.py).

The reference wraps GCP AutoML Natural Language: build_issue_doc ->
PredictionServiceClient.predict -> keep classifications with confidence
>= 0.5 and map '-' -> '/' once in display names (automl_model.py:17,34-96).
Here the prediction backend is an injectable client (any callable
``predict(doc) -> [(display_name, confidence), ...]``) so the same model
class serves a real remote endpoint (HTTP) or a local model, offline."""
from __future__ import annotations

import logging
from typing import Callable, Dict, List, Optional, Sequence, Tuple

from ..gh.github_util import build_issue_doc
from .models import IssueLabelModel

log = logging.getLogger(__name__)

PredictFn = Callable[[str], Sequence[Tuple[str, float]]]


class AutoMLModel(IssueLabelModel):
    CONFIDENCE_THRESHOLD = 0.5  # reference automl_model.py:17

    def __init__(self, model_name: str, predict_fn: Optional[PredictFn] = None,
                 endpoint: Optional[str] = None, session=None):
        self.model_name = model_name
        if predict_fn is None:
            if endpoint is None:
                raise ValueError("need predict_fn or endpoint")
            if session is None:
                import requests
                session = requests.Session()

            def _http_predict(doc: str):
                r = session.post(endpoint, json={"document": doc,
                                                 "model": model_name})
                r.raise_for_status()
                return [(c["display_name"], c["confidence"])
                        for c in r.json().get("classifications", [])]
            predict_fn = _http_predict
        self.predict_fn = predict_fn

    def predict_issue_labels(self, org: str, repo: str, title: str,
                             text: List[str], context: Optional[dict] = None
                             ) -> Dict[str, float]:
        doc = build_issue_doc(org, repo, title, text)
        try:
            results = self.predict_fn(doc)
        except Exception:
            log.exception("remote prediction failed for %s/%s", org, repo)
            return {}
        out: Dict[str, float] = {}
        for display_name, confidence in results:
            if confidence < self.CONFIDENCE_THRESHOLD:
                continue
            # display names can't contain '/', so '-' encodes it; map ONCE
            # (reference automl_model.py maps the first '-' only)
            label = display_name.replace("-", "/", 1)
            out[label] = float(confidence)
        return out
