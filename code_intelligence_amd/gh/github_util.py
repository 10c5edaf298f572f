"""Issue fetching + per-repo config (reference: py/code_intelligence/github_util.py).

* ``get_yaml(owner, repo)`` — load the repo's .github/issue_label_bot.yaml
  (github_util.py:14-40)
* ``build_issue_doc(org, repo, title, comments)`` — AutoML doc: title +
  lowercased '{org}_{repo}' + comments joined by newline (42-58)
* ``get_issue(url, client)`` — full issue via GraphQL with 3-cursor
  pagination over comments / labels / timeline UNLABELED events; returns
  {title, comments, comment_authors, labels, removed_labels} where
  removed = unlabeled minus currently-present (62-212, removal math at 208)
"""
from __future__ import annotations

import base64
import logging
from typing import Dict, List, Optional

import yaml

from .graphql import GraphQLClient
from .util import parse_issue_url  # noqa: F401 (re-exported for callers)

log = logging.getLogger(__name__)

ISSUE_QUERY = """
query getIssue($url: URI!, $commentsCursor: String, $labelsCursor: String,
               $timelineCursor: String) {
  resource(url: $url) {
    ... on Issue {
      title
      body
      author { login }
      comments(first: 100, after: $commentsCursor) {
        pageInfo { endCursor hasNextPage }
        nodes { body author { login } }
      }
      labels(first: 100, after: $labelsCursor) {
        pageInfo { endCursor hasNextPage }
        nodes { name }
      }
      timelineItems(itemTypes: [UNLABELED_EVENT], first: 100,
                    after: $timelineCursor) {
        pageInfo { endCursor hasNextPage }
        nodes { ... on UnlabeledEvent { label { name } } }
      }
    }
  }
}
"""

_FIELDS = {"comments": "commentsCursor", "labels": "labelsCursor",
           "timelineItems": "timelineCursor"}


def get_issue(url: str, client: GraphQLClient) -> Dict:
    """Fetch title/comments/labels/removed_labels, paginating each of the
    three connections independently until exhausted."""
    comments: List[str] = []
    authors: List[Optional[str]] = []
    labels: List[str] = []
    unlabeled: List[str] = []
    title: Optional[str] = None
    cursors = {c: None for c in _FIELDS.values()}
    # fields still being accumulated this round
    active = set(_FIELDS)
    first = True
    while active:
        data = client.run_query(ISSUE_QUERY, {"url": url, **cursors})
        issue = (data.get("data") or {}).get("resource")
        if issue is None:
            raise ValueError(f"no issue at {url}")
        if first:
            title = issue["title"]
            comments.append(issue.get("body") or "")
            authors.append((issue.get("author") or {}).get("login"))
            first = False
        next_active = set()
        for field in active:
            block = issue.get(field) or {}
            nodes = block.get("nodes") or []
            if field == "comments":
                for n in nodes:
                    comments.append(n.get("body") or "")
                    authors.append((n.get("author") or {}).get("login"))
            elif field == "labels":
                labels.extend(n["name"] for n in nodes)
            else:
                unlabeled.extend(n["label"]["name"] for n in nodes
                                 if n.get("label"))
            pi = block.get("pageInfo") or {}
            if pi.get("hasNextPage"):
                cursors[_FIELDS[field]] = pi.get("endCursor")
                next_active.add(field)
        active = next_active
    removed = [l for l in unlabeled if l not in labels]
    seen: set = set()
    removed = [l for l in removed if not (l in seen or seen.add(l))]
    return {"title": title, "comments": comments, "comment_authors": authors,
            "labels": labels, "removed_labels": removed}


def build_issue_doc(org: str, repo: str, title: str, text: List[str]) -> str:
    """title newline org_repo(lowercase) newline comments (github_util.py:42-58)."""
    pieces = [title, f"{org.lower()}_{repo.lower()}"]
    pieces.extend(text)
    return "\n".join(pieces)


def get_yaml(owner: str, repo: str, session=None,
             api_url: str = "https://api.github.com") -> Optional[dict]:
    """Fetch .github/issue_label_bot.yaml from the repo (github_util.py:14-40)."""
    if session is None:
        import requests
        session = requests.Session()
    url = f"{api_url}/repos/{owner}/{repo}/contents/.github/issue_label_bot.yaml"
    r = session.get(url)
    if r.status_code != 200:
        return None
    try:
        content = base64.b64decode(r.json()["content"])
        return yaml.safe_load(content)
    except Exception:
        log.exception("invalid issue_label_bot.yaml in %s/%s", owner, repo)
        return None
