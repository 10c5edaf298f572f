"""Label-prediction microservice: model registry,
transfer-learning heads, queue worker, GitHub label application."""
from .models import IssueLabelModel
from .mlp import MLPHead, MLPWrapper
from .combined_model import CombinedLabelModels
from .repo_config import RepoConfig
from .issue_label_predictor import IssueLabelPredictor

__all__ = ["IssueLabelModel", "MLPHead", "MLPWrapper", "CombinedLabelModels",
           "RepoConfig", "IssueLabelPredictor"]
