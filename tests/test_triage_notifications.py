"""Triage rule engine + notifications tests (reference techniques:
golden issue fixtures + recording fakes — SURVEY.md §4)."""

import pytest

from code_intelligence_amd.notifications.notifications import (
    NotificationManager, download_issues, parse_issue_shards)
from code_intelligence_amd.triage.triage import (IssueTriage, TriageInfo,
                                                 TRIAGE_PROJECT)


def _edges(items):
    return {"edges": [{"node": n} for n in items]}


def _issue(state="OPEN", labels=(), events=(), cards=(), closed_at=None):
    return {
        "id": "MDU6SXNzdWUx", "number": 1,
        "url": "https://github.com/a/b/issues/1", "state": state,
        "closedAt": closed_at,
        "labels": _edges([{"name": n} for n in labels]),
        "projectCards": _edges(list(cards)),
        "timelineItems": _edges(list(events)),
    }


def _label_event(name, t="2024-01-01T00:00:00Z"):
    return {"__typename": "LabeledEvent", "createdAt": t, "label": {"name": name}}


FULLY_TRIAGED_EVENTS = [
    _label_event("kind/bug", "2024-01-01T00:00:00Z"),
    _label_event("priority/p2", "2024-01-02T00:00:00Z"),
    _label_event("area/ops", "2024-01-03T00:00:00Z"),
]


def test_needs_triage_missing_labels():
    info = TriageInfo.from_issue(_issue())
    assert info.needs_triage
    assert "kind label" in info.message()


def test_fully_labeled_does_not_need_triage():
    info = TriageInfo.from_issue(_issue(events=FULLY_TRIAGED_EVENTS))
    assert not info.needs_triage
    # triaged_at = latest required event
    assert info.triaged_at.isoformat().startswith("2024-01-03")


def test_p0_requires_project():
    events = [
        _label_event("kind/bug"),
        _label_event("priority/p0"),
        _label_event("area/ops"),
    ]
    info = TriageInfo.from_issue(_issue(labels=["priority/p0"], events=events))
    assert info.requires_project
    assert info.needs_triage  # no AddedToProjectEvent yet
    events.append({"__typename": "AddedToProjectEvent",
                   "createdAt": "2024-02-01T00:00:00Z"})
    info2 = TriageInfo.from_issue(_issue(labels=["priority/p0"], events=events))
    assert not info2.needs_triage
    assert info2.triaged_at.isoformat().startswith("2024-02-01")


def test_closed_issue_never_needs_triage():
    info = TriageInfo.from_issue(_issue(state="CLOSED",
                                        closed_at="2024-03-01T00:00:00Z"))
    assert not info.needs_triage
    assert info.triaged_at.isoformat().startswith("2024-03-01")


class RecordingClient:
    """GraphQL fake recording mutations, serving canned queries."""

    def __init__(self, pages):
        self.pages = list(pages)
        self.mutations = []

    def run_query(self, query, variables=None, headers=None):
        if query.strip().startswith("mutation"):
            self.mutations.append((query.split("(")[0].strip(), variables))
            return {"data": {}}
        return self.pages.pop(0)


def test_triage_adds_and_removes_cards():
    needs = _issue()  # needs triage, not in project
    done = _issue(events=FULLY_TRIAGED_EVENTS,
                  cards=[{"id": "card1", "project": {"name": TRIAGE_PROJECT}}])
    page = {"data": {"repository": {"issues": {
        "totalCount": 2,
        "pageInfo": {"hasNextPage": False},
        "edges": [{"node": needs}, {"node": done}],
    }}}}
    client = RecordingClient([page])
    t = IssueTriage(client=client)
    results = t.triage("kubeflow/kubeflow")
    assert [r["action"] for r in results] == ["added_to_project",
                                              "removed_from_project"]
    assert len(client.mutations) == 2
    assert "addCard" in client.mutations[0][0]
    assert client.mutations[0][1]["contentId"] == "MDU6SXNzdWUx"
    assert client.mutations[1][1] == {"cardId": "card1"}


def test_triage_idempotent_noop():
    ok = _issue(events=FULLY_TRIAGED_EVENTS)  # triaged, not in project
    page = {"data": {"repository": {"issues": {
        "totalCount": 1, "pageInfo": {"hasNextPage": False},
        "edges": [{"node": ok}]}}}}
    client = RecordingClient([page])
    results = IssueTriage(client=client).triage("a/b")
    assert results[0]["action"] == "none"
    assert client.mutations == []


# --- notifications -------------------------------------------------------
def test_mark_read_policy():
    assert NotificationManager.should_mark_read({"reason": "subscribed"})
    assert NotificationManager.should_mark_read({"reason": "ci_activity"})
    assert not NotificationManager.should_mark_read({"reason": "mention"})
    assert not NotificationManager.should_mark_read({"reason": "assign"})


def test_download_issues_shards(tmp_path):
    issues = [{"number": i, "title": f"t{i}"} for i in range(5)]
    pages = [
        {"data": {"repository": {"issues": {
            "pageInfo": {"hasNextPage": True, "endCursor": "c1"},
            "edges": [{"node": n} for n in issues[:3]]}}}},
        {"data": {"repository": {"issues": {
            "pageInfo": {"hasNextPage": False},
            "edges": [{"node": n} for n in issues[3:]]}}}},
    ]
    client = RecordingClient(pages)
    got = download_issues("kubeflow/kubeflow", tmp_path, client=client)
    assert len(got) == 5
    assert len(parse_issue_shards(tmp_path)) == 5


def test_mark_read_dry_run_and_patch():
    class Sess:
        def __init__(self):
            self.patched = []
            self.page = 0

        def get(self, url, params=None, headers=None):
            self.page += 1
            batch = [{"id": "1", "reason": "subscribed"},
                     {"id": "2", "reason": "mention"}] if self.page == 1 else []

            class R:
                status_code = 200

                def raise_for_status(self):
                    pass

                def json(self2):
                    return batch
            return R()

        def patch(self, url, headers=None):
            self.patched.append(url)

            class R:
                status_code = 205
            return R()

    s = Sess()
    nm = NotificationManager(session=s, token="t")
    marked = nm.mark_read(dry_run=True)
    assert marked == ["1"]            # mention kept, nothing patched
    assert s.patched == []
    s.page = 0
    marked = nm.mark_read()
    assert marked == ["1"]
    assert s.patched and s.patched[0].endswith("/notifications/threads/1")


def test_archive_max_age_filter(tmp_path):
    import pandas as pd
    from code_intelligence_amd.gh import bigquery
    now = pd.Timestamp.now(tz="UTC")
    events = [
        {"org": "o", "repo": "r", "issue_num": 1, "title": "new", "body": "",
         "labels": [], "updated_at": (now - pd.Timedelta(days=2)).isoformat()},
        {"org": "o", "repo": "r", "issue_num": 2, "title": "old", "body": "",
         "labels": [], "updated_at": (now - pd.Timedelta(days=90)).isoformat()},
    ]
    bigquery.write_archive_events(events, tmp_path / "e.jsonl")
    df = bigquery.get_issues("o", max_age_days=30, archive_root=tmp_path)
    assert df["issue_num"].tolist() == [1]


def test_process_issue_results():
    """reference notifications_test.py:6-12 contract."""
    from code_intelligence_amd.notifications.notifications import (
        process_issue_results)
    page = {"data": {"repository": {"issues": {
        "edges": [{"node": {"title": f"t{i}", "number": i}}
                  for i in range(100)]}}}}
    issues = process_issue_results(page)
    assert len(issues) == 100
    assert "title" in issues[0]
