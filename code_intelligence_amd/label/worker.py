"""Label-bot worker (reference: py/label_microservice/worker.py).

Consumes issue events one at a time, predicts labels (lazy predictor
construction on first message — the reference does this as a TF
thread-affinity workaround, worker.py:138-145; kept because lazy init also
avoids loading models in producers), filters/aliases via the org+repo
``.github/issue_label_bot.yaml`` config (apply_repo_config,
worker.py:251-297), dedupes against existing AND previously-removed labels
(349-357), applies labels + a probability-table markdown comment
(299-436), and ALWAYS acks (poison-pill avoidance, worker.py:217-231)."""
from __future__ import annotations

import json
import logging
from typing import Dict, List, Optional

from ..gh import github_util
from ..gh.util import build_issue_url
from .issue_label_predictor import IssueLabelPredictor
from .queueing import BaseQueue, Message, queue_from_env

log = logging.getLogger(__name__)


class GitHubIssueClient:
    """Minimal REST surface the worker needs: add labels + comment.
    Injectable; offline tests use a recording fake."""

    def __init__(self, token_generator=None, session=None,
                 api_url: str = "https://api.github.com"):
        self.token_generator = token_generator
        if session is None:
            import requests
            session = requests.Session()
        self.session = session
        self.api_url = api_url.rstrip("/")

    def _headers(self) -> dict:
        h = {"Accept": "application/vnd.github.v3+json"}
        if self.token_generator is not None:
            h.update(self.token_generator.auth_headers())
        return h

    def add_labels(self, owner: str, repo: str, issue_num: int,
                   labels: List[str]) -> None:
        r = self.session.post(
            f"{self.api_url}/repos/{owner}/{repo}/issues/{issue_num}/labels",
            json={"labels": labels}, headers=self._headers())
        r.raise_for_status()

    def add_comment(self, owner: str, repo: str, issue_num: int, body: str) -> None:
        r = self.session.post(
            f"{self.api_url}/repos/{owner}/{repo}/issues/{issue_num}/comments",
            json={"body": body}, headers=self._headers())
        r.raise_for_status()

    def list_comments(self, owner: str, repo: str, issue_num: int) -> List[dict]:
        r = self.session.get(
            f"{self.api_url}/repos/{owner}/{repo}/issues/{issue_num}/comments",
            headers=self._headers())
        r.raise_for_status()
        return r.json()


def wait_for_endpoint(url: str, session=None, timeout_s: float = 120.0,
                      base_delay_s: float = 1.0, max_delay_s: float = 30.0
                      ) -> bool:
    """Exponential-backoff wait for a dependency's /healthz (the analogue
    of the reference's wait_for_gcp_account retry loop, worker.py:446-463:
    dependencies come up in any order; the worker retries with backoff
    instead of crash-looping)."""
    import time
    if session is None:
        import requests
        session = requests.Session()
    deadline = time.monotonic() + timeout_s
    delay = base_delay_s
    while True:
        try:
            r = session.get(url.rstrip("/") + "/healthz", timeout=5)
            if r.status_code == 200:
                return True
        except Exception:
            pass
        if time.monotonic() >= deadline:
            return False
        log.info("dependency %s not ready; retrying in %.1fs", url, delay)
        time.sleep(delay)
        delay = min(delay * 2, max_delay_s)


class Worker:
    BOT_MARKER = "<!-- issue-label-bot -->"

    def __init__(self, queue: Optional[BaseQueue] = None,
                 predictor: Optional[IssueLabelPredictor] = None,
                 github: Optional[GitHubIssueClient] = None,
                 repo_config_fn=None):
        self.queue = queue or queue_from_env()
        self._predictor = predictor
        self.github = github
        # fn(owner, repo) -> yaml dict or None; defaults to live fetch
        self.repo_config_fn = repo_config_fn or github_util.get_yaml

    @classmethod
    def subscribe_from_env(cls, **kw) -> "Worker":
        w = cls(queue=queue_from_env(), **kw)
        w.subscribe()
        return w

    @property
    def predictor(self) -> IssueLabelPredictor:
        if self._predictor is None:
            log.info("lazily constructing predictor")
            self._predictor = IssueLabelPredictor()
        return self._predictor

    # --- config application (worker.py:251-297) ---------------------------
    @staticmethod
    def apply_repo_config(repo_config: Optional[dict],
                          predictions: Dict[str, float]) -> Dict[str, float]:
        if not repo_config:
            return predictions
        out = dict(predictions)
        aliases = repo_config.get("label-alias") or {}
        out = {aliases.get(k, k): v for k, v in out.items()}
        allowed = repo_config.get("predicted-labels")
        if allowed:
            out = {k: v for k, v in out.items() if k in allowed}
        return out

    # --- main callback ----------------------------------------------------
    def callback(self, message: Message) -> None:
        try:
            attrs = message.attributes
            owner = attrs["repo_owner"]
            repo = attrs["repo_name"]
            issue_num = int(attrs["issue_num"])
            ctx = {"repo_owner": owner, "repo_name": repo, "issue_num": issue_num}
            log.info("handling issue event", extra={"extra_context": ctx})
            predictions = self.predictor.predict(
                {"repo_owner": owner, "repo_name": repo, "issue_num": issue_num})
            log.info("predictions %s", predictions, extra={"extra_context": ctx})
            self.add_labels_to_issue(owner, repo, issue_num, predictions)
        except Exception:
            log.exception("failed to process message %s", message.message_id)
        finally:
            message.ack()  # ALWAYS ack (worker.py:231)

    def subscribe(self, stop_event=None) -> None:
        self.queue.subscribe(self.callback, max_messages=1, stop_event=stop_event)

    # --- label application (worker.py:299-436) ----------------------------
    def add_labels_to_issue(self, owner: str, repo: str, issue_num: int,
                            predictions: Dict[str, float],
                            issue_data: Optional[dict] = None) -> List[str]:
        # merge org-level (.github repo) + repo-level config (320-337)
        org_cfg = self.repo_config_fn(owner, ".github") or {}
        repo_cfg = self.repo_config_fn(owner, repo) or {}
        merged = {**org_cfg, **repo_cfg}
        predictions = self.apply_repo_config(merged, predictions)

        client = getattr(self.predictor, "client", None)
        if issue_data is None and client is not None:
            issue_data = github_util.get_issue(
                build_issue_url(owner, repo, issue_num), client)
        issue_data = issue_data or {"labels": [], "removed_labels": []}
        existing = set(issue_data.get("labels") or [])
        removed = set(issue_data.get("removed_labels") or [])
        # dedupe against existing and human-removed labels (349-357)
        to_add = [l for l in predictions
                  if l not in existing and l not in removed]
        if not to_add:
            log.info("no new labels for %s/%s#%s", owner, repo, issue_num)
            return []
        if self.github is not None:
            # skip commenting if the bot already commented (394-413)
            already = any(self.BOT_MARKER in (c.get("body") or "")
                          for c in self.github.list_comments(owner, repo, issue_num))
            self.github.add_labels(owner, repo, issue_num, to_add)
            if not already:
                self.github.add_comment(owner, repo, issue_num,
                                        self._comment(predictions, to_add))
        return to_add

    def _comment(self, predictions: Dict[str, float], added: List[str]) -> str:
        rows = "\n".join(f"| {l} | {predictions[l]:.2f} |" for l in added)
        return (f"{self.BOT_MARKER}\n"
                "Issue-Label Bot is automatically applying the labels below:\n\n"
                "| Label | Probability |\n|---|---|\n" + rows +
                "\n\nPlease mark this comment with :thumbsup: or :thumbsdown: "
                "to give our bot feedback!")


def main():  # pragma: no cover
    from ..gh.util import setup_json_logging
    setup_json_logging()
    Worker.subscribe_from_env()


if __name__ == "__main__":  # pragma: no cover
    main()
