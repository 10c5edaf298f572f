"""Issue-archive queries (reference: py/code_intelligence/github_bigquery.py).

The reference pulls an org's issues from the public ``githubarchive``
BigQuery dataset, dedupes to the latest event per issue, and parses labels
(github_bigquery.py:8-68). Offline, the same contract is served from a
local archive directory of JSONL event files (one event per line with
{org, repo, issue_num, title, body, labels, updated_at}); ``get_issues``
returns the deduped latest-state DataFrame with parsed label lists."""
from __future__ import annotations

import json
import os
from pathlib import Path
from typing import Optional

import pandas as pd


def _archive_root() -> Path:
    return Path(os.environ.get("CI_ISSUE_ARCHIVE", "/tmp/ci_issue_archive"))


def get_issues(login: str, project: Optional[str] = None,
               max_age_days: Optional[int] = None,
               archive_root: Optional[str] = None) -> pd.DataFrame:
    """Latest event per issue for an org, labels parsed to lists."""
    root = Path(archive_root) if archive_root else _archive_root()
    rows = []
    for f in sorted(root.glob("*.jsonl")):
        with open(f) as fh:
            for line in fh:
                if not line.strip():
                    continue
                ev = json.loads(line)
                if ev.get("org") != login:
                    continue
                rows.append(ev)
    if not rows:
        return pd.DataFrame(columns=["org", "repo", "issue_num", "title",
                                     "body", "labels", "updated_at"])
    df = pd.DataFrame(rows)
    df["updated_at"] = pd.to_datetime(df["updated_at"], utc=True)
    if max_age_days is not None:
        cutoff = pd.Timestamp.now(tz="UTC") - pd.Timedelta(days=max_age_days)
        df = df[df["updated_at"] >= cutoff]
    # dedupe to the latest event per (repo, issue)
    df = df.sort_values("updated_at").groupby(
        ["repo", "issue_num"], as_index=False).last()
    df["labels"] = df["labels"].apply(
        lambda l: l if isinstance(l, list) else
        [s.strip() for s in str(l).split(",") if s.strip()])
    return df


def write_archive_events(events, path) -> None:
    """Test/dev helper: write events as one JSONL shard."""
    Path(path).parent.mkdir(parents=True, exist_ok=True)
    with open(path, "w") as f:
        for ev in events:
            f.write(json.dumps(ev) + "\n")
