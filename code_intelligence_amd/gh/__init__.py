"""GitHub / cloud integration layer — offline-friendly:
every network client takes an injectable transport so the full test suite
runs with fakes (the reference's own test technique)."""
from .util import (parse_issue_spec, parse_issue_url, build_issue_url,
                   CustomisedJSONFormatter)
from .graphql import GraphQLClient, unpack_and_split_nodes, ShardWriter
from .github_app import (GitHubApp, FixedAccessTokenGenerator,
                         GitHubAppTokenGenerator)
from . import github_util, gcs_util, bigquery

__all__ = [
    "parse_issue_spec", "parse_issue_url", "build_issue_url",
    "CustomisedJSONFormatter", "GraphQLClient", "unpack_and_split_nodes",
    "ShardWriter", "GitHubApp", "FixedAccessTokenGenerator",
    "GitHubAppTokenGenerator", "github_util", "gcs_util", "bigquery",
]
