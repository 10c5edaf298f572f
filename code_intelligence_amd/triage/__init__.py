from .triage import (TriageInfo, IssueTriage, ALLOWED_PRIORITY,
                     REQUIRES_PROJECT, TRIAGE_PROJECT)

__all__ = ["TriageInfo", "IssueTriage", "ALLOWED_PRIORITY",
           "REQUIRES_PROJECT", "TRIAGE_PROJECT"]
