"""Model registry/dispatcher (reference: issue_label_predictor.py).

Always loads the universal model; loads per-org and per-repo combined
models from the MODEL_CONFIG yaml (env, populated by a configmap in the
reference — issue_label_predictor.py:63-71); routes an issue to the best
model: '{org}/{repo}_combined' > '{org}_combined' > 'universal'
(146-155); ``predict(payload)`` dispatches on payload keys (183-227)."""
from __future__ import annotations

import logging
import os
from typing import Dict, List, Optional

import yaml

from ..gh.graphql import GraphQLClient
from ..gh import github_util
from ..gh.util import build_issue_url
from .automl_model import AutoMLModel
from .combined_model import CombinedLabelModels
from .models import IssueLabelModel
from .repo_specific_model import RepoSpecificLabelModel
from .universal_kind_label_model import UniversalKindLabelModel

log = logging.getLogger(__name__)

UNIVERSAL_MODEL_NAME = "universal"


class IssueLabelPredictor:
    def __init__(self, model_config: Optional[dict] = None,
                 graphql_client: Optional[GraphQLClient] = None,
                 universal: Optional[IssueLabelModel] = None,
                 embedding_api_endpoint: Optional[str] = None):
        self.client = graphql_client
        self.embedding_api_endpoint = embedding_api_endpoint or os.environ.get(
            "EMBEDDING_API_ENDPOINT", "http://issue-embedding-server")
        self.models: Dict[str, IssueLabelModel] = {}
        self._load_models(model_config, universal)

    def _load_models(self, model_config: Optional[dict],
                     universal: Optional[IssueLabelModel]) -> None:
        """reference issue_label_predictor.py:58-88."""
        if universal is None:
            path = os.environ.get("UNIVERSAL_MODEL_PATH")
            if path and os.path.exists(path):
                universal = UniversalKindLabelModel.load(path)
            else:
                universal = UniversalKindLabelModel()
        self.models[UNIVERSAL_MODEL_NAME] = universal
        if model_config is None:
            cfg_path = os.environ.get("MODEL_CONFIG")
            if cfg_path and os.path.exists(cfg_path):
                with open(cfg_path) as f:
                    model_config = yaml.safe_load(f)
        for spec in (model_config or {}).get("models", []):
            try:
                self._load_one(spec)
            except Exception:
                log.exception("failed to load model spec %s", spec)

    def _load_one(self, spec: dict) -> None:
        kind = spec.get("kind")
        org = spec.get("org")
        repo = spec.get("repo")
        if not org:
            raise ValueError(f"model spec needs 'org': {spec}")
        if kind == "automl":
            m: IssueLabelModel = AutoMLModel(
                model_name=spec["model"], endpoint=spec.get("endpoint"))
        elif kind == "repo_specific":
            if not repo:
                raise ValueError(f"repo_specific spec needs 'repo': {spec}")
            m = RepoSpecificLabelModel.from_repo(
                org, repo, embedding_api_endpoint=self.embedding_api_endpoint)
        else:
            raise ValueError(f"unknown model kind {kind}")
        key = f"{org}/{repo}_combined" if repo else f"{org}_combined"
        if key in self.models and isinstance(self.models[key], CombinedLabelModels):
            self.models[key].models.append(m)
        else:
            self.models[key] = CombinedLabelModels(
                [self.models[UNIVERSAL_MODEL_NAME], m])

    def _model_for(self, org: str, repo: str) -> IssueLabelModel:
        """routing: repo combined > org combined > universal (146-155)."""
        for key in (f"{org}/{repo}_combined", f"{org}_combined",
                    UNIVERSAL_MODEL_NAME):
            if key in self.models:
                log.info("routing %s/%s -> %s", org, repo, key)
                return self.models[key]
        raise KeyError("no universal model loaded")

    def predict_labels_for_data(self, org: str, repo: str, title: str,
                                text: List[str],
                                context: Optional[dict] = None) -> Dict[str, float]:
        model = self._model_for(org, repo)
        return model.predict_issue_labels(org, repo, title, text, context)

    def predict_labels_for_issue(self, org: str, repo: str,
                                 issue_num: int) -> Dict[str, float]:
        if self.client is None:
            raise RuntimeError("predictor has no GraphQL client configured")
        url = build_issue_url(org, repo, issue_num)
        issue = github_util.get_issue(url, self.client)
        return self.predict_labels_for_data(
            org, repo, issue["title"], issue["comments"],
            context={"issue": issue})

    def predict(self, data: dict) -> Dict[str, float]:
        """payload dispatch (183-227): either {repo_owner, repo_name,
        issue_num} or inline {repo_owner, repo_name, title, text}."""
        org = data.get("repo_owner") or data.get("org")
        repo = data.get("repo_name") or data.get("repo")
        if org is None or repo is None:
            raise ValueError(f"payload missing repo_owner/repo_name: {data}")
        if "issue_num" in data and "title" not in data:
            return self.predict_labels_for_issue(org, repo, int(data["issue_num"]))
        title = data.get("title", "")
        text = data.get("text") or ([data["body"]] if data.get("body") else [])
        return self.predict_labels_for_data(org, repo, title, text)
