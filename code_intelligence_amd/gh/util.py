"""Issue spec/url parsing + structured JSON logging.

Reference semantics: py/code_intelligence/util.py:22-83 (parse_issue_spec,
parse_issue_url, build_issue_url, CustomisedJSONFormatter adding
filename/line/level/time/thread to every record)."""
from __future__ import annotations

import datetime
import json
import logging
import re
from typing import Optional, Tuple

ISSUE_SPEC_RE = re.compile(r"([^/]+)/([^#]+)#(\d+)")
ISSUE_URL_RE = re.compile(
    r"https?://github\.com/([^/]+)/([^/]+)/issues/(\d+)/?")


def parse_issue_spec(spec: str) -> Tuple[Optional[str], Optional[str],
                                         Optional[int]]:
    """'owner/repo#1234' -> (owner, repo, 1234); (None, None, None) if
    malformed — unpacking-safe, reference util_test.py:8-25 semantics."""
    m = ISSUE_SPEC_RE.fullmatch(spec or "")
    if not m:
        return None, None, None
    return m.group(1), m.group(2), int(m.group(3))


def parse_issue_url(url: str) -> Optional[Tuple[str, str, int]]:
    m = ISSUE_URL_RE.fullmatch(url or "")
    if not m:
        return None
    return m.group(1), m.group(2), int(m.group(3))


def build_issue_url(owner: str, repo: str, number) -> str:
    return f"https://github.com/{owner}/{repo}/issues/{number}"


def build_issue_spec(owner: str, repo: str, number) -> str:
    return f"{owner}/{repo}#{number}"


class CustomisedJSONFormatter(logging.Formatter):
    """One JSON object per record with the reference's extra fields."""

    def format(self, record: logging.LogRecord) -> str:
        obj = {
            "message": record.getMessage(),
            "filename": record.filename,
            "line_number": record.lineno,
            "level": record.levelname,
            "time": datetime.datetime.utcfromtimestamp(record.created)
                    .isoformat() + "Z",
            "thread": record.threadName,
        }
        extra = getattr(record, "extra_context", None)
        if isinstance(extra, dict):
            obj.update(extra)
        if record.exc_info:
            obj["exc_info"] = self.formatException(record.exc_info)
        return json.dumps(obj)


def setup_json_logging(level=logging.INFO) -> None:
    handler = logging.StreamHandler()
    handler.setFormatter(CustomisedJSONFormatter())
    root = logging.getLogger()
    root.handlers = [handler]
    root.setLevel(level)
