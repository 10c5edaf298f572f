from .lm_loader import LMStreamLoader
from .synthetic import synthetic_issue_tokens, synthetic_issue_texts

__all__ = ["LMStreamLoader", "synthetic_issue_tokens", "synthetic_issue_texts"]
