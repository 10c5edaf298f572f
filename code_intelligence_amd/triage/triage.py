"""Issue-triage rule engine (reference: py/issue_triage/triage.py).

Decides whether an issue "needs triage" from its GraphQL label/project/
timeline history and maintains the "Needs Triage" Kanban project card:

* needs triage when the issue is open AND (no kind/* label event, no
  allowed priority label event, no area/* (or platform*) label event, or a
  P0/P1 priority without an AddedToProject event) — triage.py:118-132
* ``triaged_at`` = the latest of the required label events when all are
  present, else the close time — triage.py:173-195
* ``IssueTriage.triage(repo)`` iterates open issues (paginated GraphQL),
  and per issue adds/removes the triage project card via mutations
  (triage.py:527-777). The GraphQL client is injectable; mutations go
  through ``client.run_query`` so tests use recording fakes.
"""
from __future__ import annotations

import dataclasses
import datetime
import logging
import os
from typing import List, Optional

from ..gh.graphql import GraphQLClient, ShardWriter, unpack_and_split_nodes

log = logging.getLogger(__name__)

# reference constants (triage.py:16-25)
PROJECT_CARD_ID = os.getenv("INPUT_NEEDS_TRIAGE_PROJECT_CARD_ID",
                            "MDEzOlByb2plY3RDb2x1bW41OTM0MzEz")
ALLOWED_PRIORITY = ["priority/p0", "priority/p1", "priority/p2", "priority/p3"]
REQUIRES_PROJECT = ["priority/p0", "priority/p1"]
TRIAGE_PROJECT = "Needs Triage"


def _parse_time(s: Optional[str]) -> Optional[datetime.datetime]:
    if not s:
        return None
    return datetime.datetime.fromisoformat(s.replace("Z", "+00:00"))


@dataclasses.dataclass
class TriageInfo:
    issue: Optional[dict] = None
    triage_project_card: Optional[dict] = None
    kind_time: Optional[datetime.datetime] = None
    priority_time: Optional[datetime.datetime] = None
    project_time: Optional[datetime.datetime] = None
    area_time: Optional[datetime.datetime] = None
    closed_at: Optional[datetime.datetime] = None
    requires_project: bool = False

    @classmethod
    def from_issue(cls, issue: dict) -> "TriageInfo":
        info = cls(issue=issue)
        labels = unpack_and_split_nodes(issue, ["labels", "edges"])
        cards = unpack_and_split_nodes(issue, ["projectCards", "edges"])
        events = unpack_and_split_nodes(issue, ["timelineItems", "edges"])

        for l in labels:
            if l.get("name") in ALLOWED_PRIORITY:
                info.requires_project = l["name"] in REQUIRES_PROJECT

        for c in cards:
            if (c.get("project") or {}).get("name") == TRIAGE_PROJECT:
                info.triage_project_card = c
                break

        for e in events:
            t = _parse_time(e.get("createdAt"))
            if t is None:
                continue
            typ = e.get("__typename")
            if typ == "LabeledEvent":
                name = (e.get("label") or {}).get("name", "")
                if name.startswith("kind") and info.kind_time is None:
                    info.kind_time = t
                if (name.startswith("area") or name.startswith("platform")) \
                        and info.area_time is None:
                    info.area_time = t
                if name in ALLOWED_PRIORITY and info.priority_time is None:
                    info.priority_time = t
            elif typ == "AddedToProjectEvent" and info.project_time is None:
                info.project_time = t

        info.closed_at = _parse_time(issue.get("closedAt"))
        return info

    @property
    def in_triage_project(self) -> bool:
        return self.triage_project_card is not None

    @property
    def needs_triage(self) -> bool:
        if (self.issue or {}).get("state", "").lower() == "closed":
            return False
        if not (self.kind_time and self.priority_time and self.area_time):
            return True
        if self.requires_project and not self.project_time:
            return True
        return False

    @property
    def triaged_at(self) -> Optional[datetime.datetime]:
        if self.needs_triage:
            return None
        events = [self.kind_time, self.priority_time, self.area_time]
        if self.requires_project:
            events.append(self.project_time)
        if all(events):
            return max(events)
        return self.closed_at  # triaged by being closed

    def message(self) -> str:
        if not self.needs_triage:
            return "Issue doesn't need attention."
        lines = ["Issue needs triage:"]
        if not self.kind_time:
            lines.append("\t Issue needs a kind label")
        if not self.priority_time:
            lines.append(f"\t Issue needs one of the priorities {ALLOWED_PRIORITY}")
        if not self.area_time:
            lines.append("\t Issue needs an area label")
        if self.requires_project and not self.project_time:
            lines.append(f"\t Issues with priority in {REQUIRES_PROJECT} "
                         "need to be assigned to a project")
        return "\n".join(lines)


ISSUES_QUERY = """
query issues($org: String!, $repo: String!, $pageSize: Int!, $cursor: String,
             $filter: String) {
  repository(owner: $org, name: $repo) {
    issues(first: $pageSize, after: $cursor, states: [OPEN],
           filterBy: {since: $filter}) {
      totalCount
      pageInfo { endCursor hasNextPage }
      edges { node {
        id number title url state closedAt
        labels(first: 30) { edges { node { name } } }
        projectCards(first: 30) {
          edges { node { id project { name number } } } }
        timelineItems(first: 30, itemTypes: [LABELED_EVENT,
                                             ADDED_TO_PROJECT_EVENT]) {
          edges { node {
            __typename
            ... on LabeledEvent { createdAt label { name } }
            ... on AddedToProjectEvent { createdAt }
          } } }
      } }
    }
  }
}
"""

ISSUE_QUERY = """
query issue($url: URI!) {
  resource(url: $url) {
    ... on Issue {
      id number title url state closedAt
      labels(first: 100) { edges { node { name } } }
      projectCards(first: 100) { edges { node { id project { name number } } } }
      timelineItems(first: 100, itemTypes: [LABELED_EVENT,
                                            ADDED_TO_PROJECT_EVENT]) {
        pageInfo { endCursor hasNextPage }
        edges { node {
          __typename
          ... on LabeledEvent { createdAt label { name } }
          ... on AddedToProjectEvent { createdAt }
        } }
      }
    }
  }
}
"""

ADD_CARD_MUTATION = """
mutation addCard($cardId: ID!, $contentId: ID!) {
  addProjectCard(input: {projectColumnId: $cardId, contentId: $contentId}) {
    clientMutationId
  }
}
"""

DELETE_CARD_MUTATION = """
mutation deleteCard($cardId: ID!) {
  deleteProjectCard(input: {cardId: $cardId}) { clientMutationId }
}
"""


class IssueTriage:
    def __init__(self, client: Optional[GraphQLClient] = None,
                 project_card_id: str = PROJECT_CARD_ID):
        self._client = client
        self.project_card_id = project_card_id

    @property
    def client(self) -> GraphQLClient:
        if self._client is None:
            self._client = GraphQLClient(
                token=os.environ.get("GITHUB_TOKEN"))
        return self._client

    def _iter_issues(self, org: str, repo: str, issue_filter: Optional[str] = None,
                     page_size: int = 100):
        cursor = None
        while True:
            data = self.client.run_query(ISSUES_QUERY, {
                "org": org, "repo": repo, "pageSize": page_size,
                "cursor": cursor, "filter": issue_filter})
            conn = data["data"]["repository"]["issues"]
            issues = [e["node"] for e in conn.get("edges", [])]
            yield issues
            pi = conn.get("pageInfo") or {}
            if not pi.get("hasNextPage"):
                break
            cursor = pi.get("endCursor")

    def download_issues(self, repo: str, output, issue_filter=None) -> List[dict]:
        """Write issue shards like the reference's shard downloader."""
        org, name = repo.split("/")
        writer = ShardWriter(output, total_shards=999)
        all_issues = []
        for batch in self._iter_issues(org, name, issue_filter):
            writer.write_shard(batch)
            all_issues.extend(batch)
        return all_issues

    def triage(self, repo: str, add_comment: bool = False) -> List[dict]:
        """Triage every open issue in org/repo; returns per-issue results."""
        org, name = repo.split("/")
        results = []
        for batch in self._iter_issues(org, name):
            for issue in batch:
                results.append(self._process_issue(issue, add_comment))
        return results

    def triage_issue(self, url: str, add_comment: bool = False) -> dict:
        issue = self._get_issue(url)
        return self._process_issue(issue, add_comment)

    def _get_issue(self, url: str) -> dict:
        data = self.client.run_query(ISSUE_QUERY, {"url": url})
        issue = (data.get("data") or {}).get("resource")
        if issue is None:
            raise ValueError(f"no issue at {url}")
        return issue

    def _process_issue(self, issue: dict, add_comment: bool = False) -> dict:
        info = TriageInfo.from_issue(issue)
        action = "none"
        if info.needs_triage and not info.in_triage_project:
            self._add_triage_project(info)
            action = "added_to_project"
        elif not info.needs_triage and info.in_triage_project:
            self._remove_triage_project(info)
            action = "removed_from_project"
        result = {"url": issue.get("url"), "needs_triage": info.needs_triage,
                  "action": action, "message": info.message()}
        log.info("triage %s", result)
        return result

    def _add_triage_project(self, info: TriageInfo) -> None:
        content_id = info.issue.get("id")
        if content_id is None:
            raise ValueError(
                "issue node id missing (the triage queries request it); "
                "cannot build the addProjectCard mutation")
        self.client.run_query(ADD_CARD_MUTATION, {
            "cardId": self.project_card_id, "contentId": content_id})

    def _remove_triage_project(self, info: TriageInfo) -> None:
        self.client.run_query(DELETE_CARD_MUTATION, {
            "cardId": info.triage_project_card["id"]})


def triage_issue(url: str, client: Optional[GraphQLClient] = None) -> dict:
    return IssueTriage(client=client).triage_issue(url)
