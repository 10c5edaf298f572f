"""GitHub/cloud integration-layer tests (offline: fake transports)."""
import base64
import json
import logging
import subprocess

import pytest

from code_intelligence_amd.gh import bigquery, gcs_util, github_util
from code_intelligence_amd.gh.graphql import (GraphQLClient, ShardWriter,
                                              unpack_and_split_nodes)
from code_intelligence_amd.gh.rs256 import app_jwt, parse_rsa_private_key_pem, \
    sign_pkcs1_sha256
from code_intelligence_amd.gh.util import (CustomisedJSONFormatter,
                                           build_issue_url, parse_issue_spec,
                                           parse_issue_url)


class FakeResponse:
    def __init__(self, status_code=200, payload=None, content=b""):
        self.status_code = status_code
        self._payload = payload
        self.content = content
        self.text = json.dumps(payload) if payload is not None else ""

    def json(self):
        return self._payload

    def raise_for_status(self):
        if self.status_code >= 400:
            raise RuntimeError(f"http {self.status_code}")


class FakeSession:
    def __init__(self, responses):
        self.responses = list(responses)
        self.requests = []

    def post(self, url, json=None, headers=None, **kw):
        self.requests.append(("POST", url, json))
        return self.responses.pop(0)

    def get(self, url, headers=None, **kw):
        self.requests.append(("GET", url, None))
        return self.responses.pop(0)


def test_parse_issue_spec_and_url():
    assert parse_issue_spec("kubeflow/tf-operator#123") == ("kubeflow", "tf-operator", 123)
    # malformed -> unpackable (None, None, None): reference util_test.py:8-25
    assert parse_issue_spec("garbage") == (None, None, None)
    assert parse_issue_spec("kubeflow/tfjob/tfjob") == (None, None, None)
    assert parse_issue_url("https://github.com/a/b/issues/7") == ("a", "b", 7)
    assert build_issue_url("a", "b", 7) == "https://github.com/a/b/issues/7"


def test_json_formatter_fields():
    rec = logging.LogRecord("n", logging.INFO, "file.py", 42, "hello %s", ("x",), None)
    rec.extra_context = {"repo": "kubeflow"}
    obj = json.loads(CustomisedJSONFormatter().format(rec))
    assert obj["message"] == "hello x"
    assert obj["line_number"] == 42
    assert obj["level"] == "INFO"
    assert obj["repo"] == "kubeflow"


def test_rs256_against_openssl(tmp_path):
    key = tmp_path / "key.pem"
    subprocess.run(["openssl", "genrsa", "-out", str(key), "2048"],
                   check=True, capture_output=True)
    pub = tmp_path / "pub.pem"
    subprocess.run(["openssl", "rsa", "-in", str(key), "-pubout", "-out", str(pub)],
                   check=True, capture_output=True)
    pem = key.read_text()
    msg = b"the quick brown fox"
    sig = sign_pkcs1_sha256(msg, parse_rsa_private_key_pem(pem))
    msg_f, sig_f = tmp_path / "m", tmp_path / "s"
    msg_f.write_bytes(msg)
    sig_f.write_bytes(sig)
    r = subprocess.run(["openssl", "dgst", "-sha256", "-verify", str(pub),
                        "-signature", str(sig_f), str(msg_f)],
                       capture_output=True)
    assert r.returncode == 0, r.stdout + r.stderr

    tok = app_jwt("12345", pem)
    h, b, s = tok.split(".")
    hdr = json.loads(base64.urlsafe_b64decode(h + "=="))
    assert hdr == {"alg": "RS256", "typ": "JWT"}
    payload = json.loads(base64.urlsafe_b64decode(b + "=="))
    assert payload["iss"] == "12345"
    assert payload["exp"] - payload["iat"] == 70  # iat backdated 10s, ttl 60


def test_graphql_client_and_unpack():
    payload = {"data": {"x": {"edges": [{"node": {"a": 1}},
                                        {"node": {"a": 2}}]}}}
    sess = FakeSession([FakeResponse(200, payload)])
    c = GraphQLClient(session=sess, token="tok")
    data = c.run_query("query {}", {"v": 1})
    assert sess.requests[0][2]["variables"] == {"v": 1}
    nodes = unpack_and_split_nodes(data, ["data", "x"])
    assert nodes == [{"a": 1}, {"a": 2}]


def test_graphql_raises_on_errors():
    sess = FakeSession([FakeResponse(200, {"errors": [{"message": "nope"}]})])
    with pytest.raises(Exception):
        GraphQLClient(session=sess).run_query("q")


def test_shard_writer(tmp_path):
    w = ShardWriter(tmp_path, total_shards=3)
    p = w.write_shard([{"a": 1}])
    assert p.name == "items-000-of-003.json"
    assert json.loads(p.read_text()) == [{"a": 1}]


def _issue_page(has_next_comments=False):
    return {"data": {"resource": {
        "title": "Crash", "body": "It broke", "author": {"login": "alice"},
        "comments": {"pageInfo": {"endCursor": "c1",
                                  "hasNextPage": has_next_comments},
                     "nodes": [{"body": "me too", "author": {"login": "bob"}}]},
        "labels": {"pageInfo": {"hasNextPage": False},
                   "nodes": [{"name": "kind/bug"}]},
        "timelineItems": {"pageInfo": {"hasNextPage": False},
                          "nodes": [{"label": {"name": "priority/p0"}},
                                    {"label": {"name": "kind/bug"}}]},
    }}}


def test_get_issue_fields_and_removed_labels():
    sess = FakeSession([FakeResponse(200, _issue_page())])
    issue = github_util.get_issue("https://github.com/a/b/issues/1",
                                  GraphQLClient(session=sess))
    assert issue["title"] == "Crash"
    assert issue["comments"] == ["It broke", "me too"]
    assert issue["comment_authors"] == ["alice", "bob"]
    assert issue["labels"] == ["kind/bug"]
    # removed = unlabeled minus currently-present (github_util.py:208)
    assert issue["removed_labels"] == ["priority/p0"]


def test_get_issue_paginates_comments():
    page2 = _issue_page()
    page2["data"]["resource"]["comments"]["nodes"] = [
        {"body": "third", "author": None}]
    sess = FakeSession([FakeResponse(200, _issue_page(has_next_comments=True)),
                        FakeResponse(200, page2)])
    issue = github_util.get_issue("u", GraphQLClient(session=sess))
    assert issue["comments"] == ["It broke", "me too", "third"]
    # labels were exhausted on page 1: not duplicated
    assert issue["labels"] == ["kind/bug"]


def test_build_issue_doc_exact():
    """reference github_util_test.py:7-16 format."""
    doc = github_util.build_issue_doc("KubeFlow", "Examples", "title here",
                                      ["line1", "line2"])
    assert doc == "title here\nkubeflow_examples\nline1\nline2"


def test_get_yaml_decodes_content():
    payload = {"content": base64.b64encode(b"predicted-labels:\n  - bug\n").decode()}
    sess = FakeSession([FakeResponse(200, payload)])
    cfg = github_util.get_yaml("o", "r", session=sess)
    assert cfg == {"predicted-labels": ["bug"]}
    sess404 = FakeSession([FakeResponse(404)])
    assert github_util.get_yaml("o", "r", session=sess404) is None


def test_object_store_roundtrip(tmp_path):
    store = gcs_util.ObjectStore(root=tmp_path)
    assert gcs_util.split_gcs_uri("gs://bkt/a/b.txt") == ("bkt", "a/b.txt")
    f = tmp_path / "local.txt"
    f.write_text("hello")
    store.upload(str(f), "gs://bkt/a/b.txt")
    assert store.exists("gs://bkt/a/b.txt")
    out = tmp_path / "out.txt"
    store.download("gs://bkt/a/b.txt", str(out))
    assert out.read_text() == "hello"


def test_bigquery_archive_dedupe(tmp_path):
    events = [
        {"org": "kubeflow", "repo": "kf", "issue_num": 1, "title": "old",
         "body": "", "labels": ["bug"], "updated_at": "2024-01-01T00:00:00Z"},
        {"org": "kubeflow", "repo": "kf", "issue_num": 1, "title": "new",
         "body": "", "labels": "bug, feature", "updated_at": "2024-02-01T00:00:00Z"},
        {"org": "other", "repo": "x", "issue_num": 2, "title": "skip",
         "body": "", "labels": [], "updated_at": "2024-01-01T00:00:00Z"},
    ]
    bigquery.write_archive_events(events, tmp_path / "shard.jsonl")
    df = bigquery.get_issues("kubeflow", archive_root=tmp_path)
    assert len(df) == 1
    assert df.iloc[0]["title"] == "new"
    assert df.iloc[0]["labels"] == ["bug", "feature"]


def test_es256_against_openssl(tmp_path):
    from code_intelligence_amd.gh.es256 import (chatbot_test_jwt,
                                                parse_ec_private_key_pem,
                                                sign_es256)
    key = tmp_path / "ec.pem"
    subprocess.run(["openssl", "ecparam", "-genkey", "-name", "prime256v1",
                    "-noout", "-out", str(key)], check=True, capture_output=True)
    pub = tmp_path / "ecpub.pem"
    subprocess.run(["openssl", "ec", "-in", str(key), "-pubout", "-out", str(pub)],
                   check=True, capture_output=True)
    d = parse_ec_private_key_pem(key.read_text())
    msg = b"chatbot webhook test"
    r, s = sign_es256(msg, d)
    # DER-encode (r, s) for openssl verification
    def _int(v):
        b = v.to_bytes(32, "big").lstrip(b"\x00")
        if b[0] & 0x80:
            b = b"\x00" + b
        return b"\x02" + bytes([len(b)]) + b
    body = _int(r) + _int(s)
    der = b"\x30" + bytes([len(body)]) + body
    (tmp_path / "m").write_bytes(msg)
    (tmp_path / "sig").write_bytes(der)
    res = subprocess.run(["openssl", "dgst", "-sha256", "-verify", str(pub),
                          "-signature", str(tmp_path / "sig"), str(tmp_path / "m")],
                         capture_output=True)
    assert res.returncode == 0, res.stdout + res.stderr

    tok = chatbot_test_jwt(key.read_text(), audience="chatbot")
    h, b, sg = tok.split(".")
    hdr = json.loads(base64.urlsafe_b64decode(h + "=="))
    assert hdr["alg"] == "ES256"
    assert len(base64.urlsafe_b64decode(sg + "==")) == 64  # raw r||s


def test_github_app_repo_listing_and_reaction(tmp_path):
    from code_intelligence_amd.gh.github_app import GitHubApp
    import subprocess as sp
    key = tmp_path / "rsa.pem"
    sp.run(["openssl", "genrsa", "-out", str(key), "2048"], check=True,
           capture_output=True)
    sess = FakeSession([
        FakeResponse(200, {"token": "tkn", "expires_at": "2099-01-01T00:00:00Z"}),
        FakeResponse(200, {"repositories": [{"full_name": "o/r"}]}),
        FakeResponse(200, {"id": 1, "content": "+1"}),
    ])
    app = GitHubApp(pem_path=str(key), app_id="7", session=sess)
    repos = app.list_installation_repos(42)
    assert repos == [{"full_name": "o/r"}]
    out = app.add_reaction("o", "r", 5, "+1", token="tkn")
    assert out["content"] == "+1"


def test_token_generator_refreshes_on_expiry():
    """GitHubAppTokenGenerator: caches until expiry-skew, then re-mints
    (reference github_app.py:305-364 refresh semantics)."""
    import datetime
    from code_intelligence_amd.gh.github_app import GitHubAppTokenGenerator

    class App:
        def __init__(self):
            self.mints = 0

        def get_installation_id(self, owner, repo=None):
            return 42

        def get_installation_access_token(self, iid):
            self.mints += 1
            exp = (datetime.datetime.now(datetime.timezone.utc)
                   + datetime.timedelta(seconds=3600 if self.mints > 1 else 30))
            return {"token": f"t{self.mints}",
                    "expires_at": exp.isoformat().replace("+00:00", "Z")}

    app = App()
    gen = GitHubAppTokenGenerator(app, "org", skew_s=60)
    assert gen.token == "t1"
    # first token expires within the 60 s skew -> next access re-mints
    assert gen.token == "t2"
    assert app.mints == 2
    # fresh hour-long token -> cached
    assert gen.token == "t2"
    assert app.mints == 2
    assert gen.auth_headers() == {"Authorization": "token t2"}
