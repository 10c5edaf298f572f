"""Notification hygiene + bulk issue download
(reference: py/notifications/notifications.py).

* ``NotificationManager.mark_read`` — mark GitHub notifications read when
  the policy allows (the reference's non-mention policy: notifications
  whose reason is not a direct mention/assignment/review-request can be
  bulk-cleared — notifications.py:26-75)
* ``download_issues`` — bulk-fetch an org/repo's issues + first comments
  via GraphQL into shard files (notifications.py:106-212)
"""
from __future__ import annotations

import logging
from typing import List, Optional

from ..gh.graphql import GraphQLClient, ShardWriter

log = logging.getLogger(__name__)

# reasons that represent a direct request for THIS user's attention;
# everything else may be bulk-marked read (reference policy)
KEEP_REASONS = {"mention", "assign", "review_requested", "team_mention"}

ISSUES_WITH_COMMENTS_QUERY = """
query issues($org: String!, $repo: String!, $pageSize: Int!, $cursor: String) {
  repository(owner: $org, name: $repo) {
    issues(first: $pageSize, after: $cursor) {
      totalCount
      pageInfo { endCursor hasNextPage }
      edges { node {
        number title url state body createdAt closedAt
        author { login }
        labels(first: 30) { edges { node { name } } }
        comments(first: 10) { edges { node { body author { login } } } }
      } }
    }
  }
}
"""


class NotificationManager:
    def __init__(self, session=None, api_url: str = "https://api.github.com",
                 token: Optional[str] = None):
        if session is None:
            import requests
            session = requests.Session()
        self.session = session
        self.api_url = api_url.rstrip("/")
        self.headers = {"Accept": "application/vnd.github.v3+json"}
        if token:
            self.headers["Authorization"] = f"token {token}"

    def list_notifications(self, all_: bool = False) -> List[dict]:
        out, page = [], 1
        while True:
            r = self.session.get(f"{self.api_url}/notifications",
                                 params={"all": str(all_).lower(),
                                         "page": page, "per_page": 100},
                                 headers=self.headers)
            r.raise_for_status()
            batch = r.json()
            if not batch:
                return out
            out.extend(batch)
            page += 1

    @staticmethod
    def should_mark_read(notification: dict) -> bool:
        """Policy: keep anything that directly requests this user."""
        return notification.get("reason") not in KEEP_REASONS

    def mark_read(self, dry_run: bool = False) -> List[str]:
        """Mark eligible notifications read; returns their thread ids."""
        marked = []
        for n in self.list_notifications():
            if not self.should_mark_read(n):
                continue
            tid = n.get("id")
            if not dry_run:
                r = self.session.patch(
                    f"{self.api_url}/notifications/threads/{tid}",
                    headers=self.headers)
                if r.status_code not in (205, 200, 204):
                    log.warning("failed to mark %s read: %s", tid, r.status_code)
                    continue
            marked.append(tid)
        log.info("marked %d notifications read", len(marked))
        return marked


def process_issue_results(data: dict) -> List[dict]:
    """Unpack one issues-query page into its issue nodes
    (reference notifications.py:44-60; its notifications_test asserts the
    node list shape)."""
    edges = (((data.get("data") or {}).get("repository") or {})
             .get("issues") or {}).get("edges") or []
    return [e["node"] for e in edges]


def download_issues(repo: str, output, client: Optional[GraphQLClient] = None,
                    page_size: int = 100) -> List[dict]:
    """Bulk-download issues + first comments into shard files."""
    client = client or GraphQLClient()
    org, name = repo.split("/")
    writer = ShardWriter(output, total_shards=999)
    cursor = None
    issues: List[dict] = []
    while True:
        data = client.run_query(ISSUES_WITH_COMMENTS_QUERY, {
            "org": org, "repo": name, "pageSize": page_size, "cursor": cursor})
        conn = data["data"]["repository"]["issues"]
        batch = [e["node"] for e in conn.get("edges", [])]
        writer.write_shard(batch)
        issues.extend(batch)
        pi = conn.get("pageInfo") or {}
        if not pi.get("hasNextPage"):
            break
        cursor = pi.get("endCursor")
    return issues


def parse_issue_shards(shard_dir) -> List[dict]:
    """Load issues back from shard files (golden-file test contract)."""
    import json
    from pathlib import Path
    out = []
    for f in sorted(Path(shard_dir).glob("items-*.json")):
        out.extend(json.loads(f.read_text()))
    return out
