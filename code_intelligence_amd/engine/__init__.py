from .inference import InferenceWrapper
from .graph_exec import GraphedEncoder

__all__ = ["InferenceWrapper", "GraphedEncoder"]
