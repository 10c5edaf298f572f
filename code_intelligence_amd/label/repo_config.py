"""Artifact naming scheme (reference: py/label_microservice/repo_config.py:9-29):
buckets 'repo-models' / 'repo-embeddings', paths '{owner}/{repo}.model.dpkl'
and '{owner}/{repo}.labels.yaml'."""
from __future__ import annotations


class RepoConfig:
    def __init__(self, repo_owner: str, repo_name: str,
                 model_bucket: str = "repo-models",
                 embeddings_bucket: str = "repo-embeddings"):
        self.repo_owner = repo_owner
        self.repo_name = repo_name
        self.model_bucket_name = model_bucket
        self.embeddings_bucket_name = embeddings_bucket

    @property
    def model_file(self) -> str:
        return f"{self.repo_owner}/{self.repo_name}.model.dpkl"

    @property
    def labels_file(self) -> str:
        return f"{self.repo_owner}/{self.repo_name}.labels.yaml"

    @property
    def model_gcs_uri(self) -> str:
        return f"gs://{self.model_bucket_name}/{self.model_file}"

    @property
    def labels_gcs_uri(self) -> str:
        return f"gs://{self.model_bucket_name}/{self.labels_file}"

    @property
    def embeddings_gcs_uri(self) -> str:
        return (f"gs://{self.embeddings_bucket_name}/"
                f"{self.repo_owner}/{self.repo_name}.embeddings.npz")
