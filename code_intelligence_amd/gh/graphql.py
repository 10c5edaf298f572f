"""GitHub GraphQL v4 client (reference: py/code_intelligence/graphql.py).

``GraphQLClient.run_query(query, variables, headers)`` POSTs to the API and
raises on transport or GraphQL-level errors; ``unpack_and_split_nodes``
walks a result path and flattens node lists; ``ShardWriter`` writes
items-%03d-of-%03d.json shards (graphql.py:41-121). The HTTP session is
injectable for offline tests."""
from __future__ import annotations

import json
import logging
from pathlib import Path
from typing import Any, Dict, List, Optional

log = logging.getLogger(__name__)

GITHUB_GRAPHQL_URL = "https://api.github.com/graphql"


class GraphQLError(RuntimeError):
    def __init__(self, errors):
        super().__init__(f"GraphQL query returned errors: {errors}")
        self.errors = errors


class GraphQLClient:
    def __init__(self, url: str = GITHUB_GRAPHQL_URL, session=None,
                 token: Optional[str] = None, headers: Optional[dict] = None):
        if session is None:
            import requests
            session = requests.Session()
        self.session = session
        self.url = url
        self.headers = dict(headers or {})
        if token:
            self.headers["Authorization"] = f"Bearer {token}"

    def run_query(self, query: str, variables: Optional[dict] = None,
                  headers: Optional[dict] = None) -> Dict[str, Any]:
        hdrs = {**self.headers, **(headers or {})}
        resp = self.session.post(
            self.url, json={"query": query, "variables": variables or {}},
            headers=hdrs)
        if resp.status_code != 200:
            raise RuntimeError(
                f"GraphQL query failed with code {resp.status_code}: {resp.text[:500]}")
        data = resp.json()
        if data.get("errors"):
            raise GraphQLError(data["errors"])
        return data


def unpack_and_split_nodes(data: dict, path: List[str]) -> List[dict]:
    """Walk ``path`` into ``data`` and return the list under 'edges'/'nodes'.
    (reference graphql.py:73-102)"""
    node = data
    for p in path:
        if node is None:
            return []
        node = node.get(p)
    if node is None:
        return []
    if isinstance(node, list):
        items = node
    else:
        items = node.get("edges", node.get("nodes", []))
    out = []
    for it in items:
        out.append(it.get("node", it) if isinstance(it, dict) else it)
    return out


class ShardWriter:
    """Write item shards as items-%03d-of-%03d.json (graphql.py:104-121)."""

    def __init__(self, output_dir, total_shards: int):
        self.output_dir = Path(output_dir)
        self.output_dir.mkdir(parents=True, exist_ok=True)
        self.total = total_shards
        self.shard = 0

    def write_shard(self, items: List[dict]) -> Path:
        name = self.output_dir / f"items-{self.shard:03d}-of-{self.total:03d}.json"
        with open(name, "w") as f:
            json.dump(items, f)
        log.info("wrote %d items to %s", len(items), name)
        self.shard += 1
        return name
