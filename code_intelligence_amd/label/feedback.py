"""Bot-comment feedback collection.

The worker's probability-table comment asks users to react with
:thumbsup:/:thumbsdown: (reference worker.py comment text). The reference
never closes that loop; this module does: scan a repo's bot comments,
read their reactions, and emit per-issue feedback records that the
retraining pipeline can join against predictions (the label-quality
signal the ModelSync loop retrains on)."""
from __future__ import annotations

import json
import logging
from pathlib import Path
from typing import Dict, List, Optional

from .worker import Worker

log = logging.getLogger(__name__)


class FeedbackCollector:
    def __init__(self, session=None, api_url: str = "https://api.github.com",
                 token: Optional[str] = None):
        if session is None:
            import requests
            session = requests.Session()
        self.session = session
        self.api_url = api_url.rstrip("/")
        self.headers = {"Accept": "application/vnd.github.squirrel-girl-preview+json"}
        if token:
            self.headers["Authorization"] = f"token {token}"

    def _get(self, path: str, **params):
        r = self.session.get(f"{self.api_url}{path}", params=params,
                             headers=self.headers)
        r.raise_for_status()
        return r.json()

    def issue_feedback(self, owner: str, repo: str, issue_num: int
                       ) -> Optional[Dict]:
        """Feedback for one issue's bot comment: {up, down, labels}."""
        comments = self._get(f"/repos/{owner}/{repo}/issues/{issue_num}/comments")
        for c in comments:
            body = c.get("body") or ""
            if Worker.BOT_MARKER not in body:
                continue
            reactions = c.get("reactions") or {}
            up = int(reactions.get("+1", 0))
            down = int(reactions.get("-1", 0))
            labels = [line.split("|")[1].strip()
                      for line in body.splitlines()
                      if line.startswith("|") and "Probability" not in line
                      and "---" not in line]
            return {"owner": owner, "repo": repo, "issue_num": issue_num,
                    "labels": labels, "up": up, "down": down,
                    "score": up - down}
        return None

    def collect(self, owner: str, repo: str, issue_nums: List[int],
                output: Optional[str] = None) -> List[Dict]:
        records = []
        for n in issue_nums:
            try:
                fb = self.issue_feedback(owner, repo, n)
            except Exception:
                log.exception("feedback fetch failed for #%s", n)
                continue
            if fb is not None:
                records.append(fb)
        if output:
            Path(output).parent.mkdir(parents=True, exist_ok=True)
            with open(output, "a") as f:
                for r in records:
                    f.write(json.dumps(r) + "\n")
        log.info("collected feedback for %d/%d issues", len(records),
                 len(issue_nums))
        return records
