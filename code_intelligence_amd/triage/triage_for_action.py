"""GitHub Action entrypoint (reference: py/issue_triage/triage_for_action.py):
reads INPUT_* env vars the Action runner provides and triages one issue.

action inputs: INPUT_ISSUE_URL (required), INPUT_NEEDS_TRIAGE_PROJECT_CARD_ID
(optional, defaults to the Kubeflow 'Needs Triage' column)."""
from __future__ import annotations

import logging
import os
import sys

from ..gh.util import setup_json_logging
from .triage import IssueTriage


def main() -> dict:
    setup_json_logging()
    url = os.environ.get("INPUT_ISSUE_URL")
    if not url:
        sys.exit("INPUT_ISSUE_URL is required")
    card = os.environ.get("INPUT_NEEDS_TRIAGE_PROJECT_CARD_ID")
    kwargs = {"project_card_id": card} if card else {}
    result = IssueTriage(**kwargs).triage_issue(url)
    logging.info("triage result: %s", result)
    return result


if __name__ == "__main__":
    main()
