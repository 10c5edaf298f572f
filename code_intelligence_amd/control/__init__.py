"""Control plane — continuous-retraining reconciler,
needs-sync/needs-train server, model registry, chatbot webhook.

The reference implements these in Go against GCP AutoML + Tekton + k8s
CRDs; here they are process-native Python services around a local model
registry so the same control loop runs on an air-gapped MI355X box."""
from .registry import LocalModelRegistry, ModelRecord
from .modelsync import ModelSync, ModelSyncSpec, PipelineRun

__all__ = ["LocalModelRegistry", "ModelRecord", "ModelSync", "ModelSyncSpec",
           "PipelineRun"]
