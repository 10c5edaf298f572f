"""Label-microservice tests (offline fakes — reference test techniques,
SURVEY.md §4: mocked prediction clients, fake embedding service, crafted
threshold datasets)."""
import numpy as np
import pytest
import torch
import yaml

from code_intelligence_amd.label.automl_model import AutoMLModel
from code_intelligence_amd.label.combined_model import CombinedLabelModels
from code_intelligence_amd.label.issue_label_predictor import IssueLabelPredictor
from code_intelligence_amd.label.mlp import MLPWrapper
from code_intelligence_amd.label.models import IssueLabelModel
from code_intelligence_amd.label.queueing import LocalQueue
from code_intelligence_amd.label.repo_config import RepoConfig
from code_intelligence_amd.label.repo_specific_model import RepoSpecificLabelModel
from code_intelligence_amd.label.universal_kind_label_model import (
    UniversalKindLabelModel)
from code_intelligence_amd.label.worker import Worker


class StaticModel(IssueLabelModel):
    def __init__(self, preds):
        self.preds = preds

    def predict_issue_labels(self, org, repo, title, text, context=None):
        return dict(self.preds)


# --- MLP wrapper --------------------------------------------------------
def test_mlp_fit_and_predict_separable():
    rng = np.random.default_rng(0)
    X = rng.normal(size=(400, 8)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float32)
    w = MLPWrapper(in_dim=8, hidden=(16,), n_labels=1, lr=1e-2)
    w.fit(X, y)
    auc = w.calculate_auc(X, y)
    assert auc > 0.95, auc


def test_threshold_search_excludes_unsatisfiable_labels():
    """reference test_mlp.py:8-59: labels whose P-R curve can't reach
    precision>=0.7 & recall>=0.5 get threshold None (never predict)."""
    w = MLPWrapper(in_dim=4, hidden=(8,), n_labels=3)

    class FixedProb(MLPWrapper):
        pass

    X = np.zeros((6, 4), dtype=np.float32)
    y = np.array([[1, 0, 1], [1, 0, 0], [1, 0, 1],
                  [0, 1, 0], [0, 1, 1], [0, 1, 0]], dtype=np.float32)
    probs = np.array([[.9, .1, .5], [.8, .2, .5], [.85, .15, .5],
                      [.1, .2, .5], [.2, .1, .5], [.15, .3, .5]],
                     dtype=np.float32)
    w.predict_probabilities = lambda X_: probs  # type: ignore
    thr = w.find_probability_thresholds(X, y, test_size=0)
    assert thr[0] is not None        # label 0 perfectly separable
    assert thr[1] is None            # label 1: probs anti-correlated
    # label 2: constant 0.5 prob, precision 0.5 < 0.7 -> None
    assert thr[2] is None
    # reference test_mlp.py:56-59 also inspects the stored metrics
    assert w.precisions[0] >= 0.7 and w.recalls[0] >= 0.5
    assert w.precisions[1] == 0.0 and w.recalls[2] == 0.0
    assert w.total_labels_count == 3


def test_mlp_save_load_roundtrip(tmp_path):
    w = MLPWrapper(in_dim=4, hidden=(8,), n_labels=2)
    w.probability_thresholds = {0: 0.6, 1: None}
    p = tmp_path / "m.dpkl"
    w.save_model(p)
    w2 = MLPWrapper.load_model(p)
    assert w2.probability_thresholds == {0: 0.6, 1: None}
    X = np.random.default_rng(0).normal(size=(3, 4)).astype(np.float32)
    assert np.allclose(w.predict_probabilities(X), w2.predict_probabilities(X))


# --- combined model (reference combined_model_test.py) ------------------
def test_combined_max_merge():
    m = CombinedLabelModels([StaticModel({"bug": 0.9, "area/ops": 0.4}),
                             StaticModel({"bug": 0.7, "feature": 0.8})])
    preds = m.predict_issue_labels("o", "r", "t", [])
    assert preds == {"bug": 0.9, "area/ops": 0.4, "feature": 0.8}


# --- automl model (reference automl_model_test.py) ----------------------
def test_automl_threshold_and_dash_mapping():
    m = AutoMLModel("m", predict_fn=lambda doc: [
        ("kind-bug", 0.9), ("area-platform-gcp", 0.6), ("lowconf", 0.2)])
    preds = m.predict_issue_labels("org", "repo", "title", ["text"])
    # 0.5 threshold filters lowconf; '-' -> '/' mapped ONCE
    assert preds == {"kind/bug": 0.9, "area/platform-gcp": 0.6}


# --- repo-specific model (reference repo_specific_model_test.py) --------
class FakeEmbeddingSession:
    def __init__(self, status=200, dim=2400):
        self.status = status
        self.dim = dim

    def post(self, url, json=None, **kw):
        class R:
            pass
        r = R()
        r.status_code = self.status
        vec = np.arange(self.dim, dtype="<f4")
        r.content = vec.tobytes()
        return r


def test_repo_specific_predicts_with_thresholds():
    w = MLPWrapper(in_dim=1600, hidden=(4,), n_labels=2)
    w.predict_probabilities = lambda X: np.array([[0.2, 0.9]])  # type: ignore
    m = RepoSpecificLabelModel(w, ["bug", "feature"], {0: 0.5, 1: 0.5},
                               session=FakeEmbeddingSession())
    preds = m.predict_issue_labels("o", "r", "t", ["b"])
    assert preds == {"feature": pytest.approx(0.9)}


def test_repo_specific_embedding_service_down_returns_empty():
    w = MLPWrapper(in_dim=1600, hidden=(4,), n_labels=1)
    m = RepoSpecificLabelModel(w, ["bug"], {0: 0.5},
                               session=FakeEmbeddingSession(status=404))
    assert m.predict_issue_labels("o", "r", "t", ["b"]) == {}


def test_repo_specific_truncates_to_1600():
    captured = {}
    w = MLPWrapper(in_dim=1600, hidden=(4,), n_labels=1)

    def capture(X):
        captured["shape"] = X.shape
        return np.array([[0.9]])
    w.predict_probabilities = capture  # type: ignore
    m = RepoSpecificLabelModel(w, ["bug"], {0: 0.5},
                               session=FakeEmbeddingSession(dim=2400))
    m.predict_issue_labels("o", "r", "t", [])
    assert captured["shape"] == (1, 1600)  # embeddings.py:116 truncation


def test_repo_config_paths():
    c = RepoConfig("kubeflow", "examples")
    assert c.model_gcs_uri == "gs://repo-models/kubeflow/examples.model.dpkl"
    assert c.labels_gcs_uri == "gs://repo-models/kubeflow/examples.labels.yaml"


def test_repo_specific_from_repo_loads_artifacts(tmp_path):
    from code_intelligence_amd.gh.gcs_util import ObjectStore
    store = ObjectStore(root=tmp_path)
    w = MLPWrapper(in_dim=1600, hidden=(4,), n_labels=2)
    w.probability_thresholds = {0: 0.5, 1: None}
    local = tmp_path / "m.dpkl"
    w.save_model(local)
    cfg = RepoConfig("o", "r")
    store.upload(str(local), cfg.model_gcs_uri)
    store.write_bytes(cfg.labels_gcs_uri, yaml.safe_dump(
        {"labels": ["bug", "feature"],
         "probability_thresholds": {0: 0.5, 1: None}}).encode())
    m = RepoSpecificLabelModel.from_repo("o", "r", store=store,
                                         session=FakeEmbeddingSession())
    assert m.label_names == ["bug", "feature"]
    assert m.thresholds == {0: 0.5, 1: None}


# --- universal model ----------------------------------------------------
def test_universal_model_thresholds(tmp_path):
    m = UniversalKindLabelModel()
    # force logits so sigmoid probs are deterministic
    with torch.no_grad():
        m.net.out.weight.zero_()
        m.net.out.bias.copy_(torch.tensor([2.0, -2.0, 0.35]))
    preds = m.predict_issue_labels("o", "r", "crash", ["boom"])
    assert "bug" in preds            # sigmoid(2.0)=0.88 >= 0.52
    assert "feature" not in preds    # 0.12 < 0.52
    assert "question" not in preds   # sigmoid(0.35)=0.587 < 0.60
    m.save(tmp_path / "u")
    m2 = UniversalKindLabelModel.load(tmp_path / "u")
    assert m2.predict_issue_labels("o", "r", "crash", ["boom"]).keys() == preds.keys()


# --- predictor routing (issue_label_predictor.py:146-155) ---------------
def test_predictor_routing_specificity():
    p = IssueLabelPredictor(model_config={}, universal=StaticModel({"u": 0.9}))
    p.models["kubeflow_combined"] = StaticModel({"org": 0.9})
    p.models["kubeflow/examples_combined"] = StaticModel({"repo": 0.9})
    assert p.predict_labels_for_data("kubeflow", "examples", "t", []) == {"repo": 0.9}
    assert p.predict_labels_for_data("kubeflow", "other", "t", []) == {"org": 0.9}
    assert p.predict_labels_for_data("x", "y", "t", []) == {"u": 0.9}


def test_predictor_payload_dispatch():
    p = IssueLabelPredictor(model_config={}, universal=StaticModel({"bug": 0.8}))
    preds = p.predict({"repo_owner": "o", "repo_name": "r",
                       "title": "t", "text": ["b"]})
    assert preds == {"bug": 0.8}
    with pytest.raises(ValueError):
        p.predict({"title": "no repo"})


# --- worker end-to-end (fakes) ------------------------------------------
class RecordingGitHub:
    def __init__(self, existing_comments=None):
        self.labels = []
        self.comments = []
        self.existing = existing_comments or []

    def add_labels(self, owner, repo, num, labels):
        self.labels.append((owner, repo, num, labels))

    def add_comment(self, owner, repo, num, body):
        self.comments.append(body)

    def list_comments(self, owner, repo, num):
        return self.existing


def _mk_worker(predictions, repo_cfg=None, issue_data=None, gh=None):
    pred = IssueLabelPredictor(model_config={},
                               universal=StaticModel(predictions))
    gh = gh or RecordingGitHub()
    w = Worker(queue=LocalQueue(), predictor=pred, github=gh,
               repo_config_fn=lambda o, r: repo_cfg)
    return w, gh


def test_worker_applies_labels_and_comments():
    w, gh = _mk_worker({"bug": 0.9})
    added = w.add_labels_to_issue("o", "r", 1, {"bug": 0.9},
                                  issue_data={"labels": [], "removed_labels": []})
    assert added == ["bug"]
    assert gh.labels == [("o", "r", 1, ["bug"])]
    assert "| bug | 0.90 |" in gh.comments[0]


def test_worker_dedupes_existing_and_removed():
    w, gh = _mk_worker({})
    added = w.add_labels_to_issue(
        "o", "r", 1, {"bug": 0.9, "feature": 0.8},
        issue_data={"labels": ["bug"], "removed_labels": ["feature"]})
    assert added == []           # human removed 'feature'; 'bug' already there
    assert gh.labels == []


def test_worker_alias_and_allowlist():
    cfg = {"label-alias": {"bug": "kind/bug"},
           "predicted-labels": ["kind/bug"]}
    out = Worker.apply_repo_config(cfg, {"bug": 0.9, "feature": 0.8})
    assert out == {"kind/bug": 0.9}


def test_worker_callback_acks_even_on_failure():
    from code_intelligence_amd.label.queueing import Message
    w, gh = _mk_worker({"bug": 0.9})
    msg = Message(attributes={})  # missing repo_owner -> KeyError inside
    w.callback(msg)
    assert msg._acked  # always ack (worker.py:231)


def test_worker_skips_duplicate_comment():
    gh = RecordingGitHub(existing_comments=[
        {"body": Worker.BOT_MARKER + " earlier"}])
    w, _ = _mk_worker({"bug": 0.9}, gh=gh)
    w.add_labels_to_issue("o", "r", 1, {"bug": 0.9},
                          issue_data={"labels": [], "removed_labels": []})
    assert gh.labels  # label applied
    assert gh.comments == []  # no second comment


def test_local_queue_spool_roundtrip(tmp_path):
    spool = tmp_path / "spool.jsonl"
    prod = LocalQueue(spool_path=str(spool))
    prod.publish(repo_owner="o", repo_name="r", issue_num=3)
    cons = LocalQueue(spool_path=str(spool))
    msg = cons.pull(timeout=0.2)
    assert msg is not None
    assert msg.attributes["issue_num"] == "3"


def test_local_broker_idempotent(tmp_path):
    from code_intelligence_amd.label.queueing import LocalBroker
    b = LocalBroker(tmp_path)
    assert not b.check_topic_exists("events")
    b.create_topic_if_not_exists("events")
    b.create_topic_if_not_exists("events")  # idempotent
    assert b.check_topic_exists("events")
    assert not b.check_subscription_name_exists("events", "worker")
    q1 = b.create_subscription_if_not_exists("events", "worker")
    q2 = b.create_subscription_if_not_exists("events", "worker2")
    assert b.check_subscription_name_exists("events", "worker")
    b.publish("events", repo_owner="o", repo_name="r", issue_num=1)
    m1, m2 = q1.pull(0.2), q2.pull(0.2)  # fan-out to both subscriptions
    assert m1.attributes["issue_num"] == "1"
    assert m2.attributes["issue_num"] == "1"


def test_wait_for_endpoint_backoff():
    from code_intelligence_amd.label.worker import wait_for_endpoint

    class FlakySession:
        def __init__(self):
            self.calls = 0

        def get(self, url, timeout=None):
            self.calls += 1
            class R:
                status_code = 503 if self.calls < 3 else 200
            return R()

    s = FlakySession()
    assert wait_for_endpoint("http://svc", session=s, timeout_s=30,
                             base_delay_s=0.01)
    assert s.calls == 3

    class DownSession:
        def get(self, url, timeout=None):
            raise ConnectionError()
    assert not wait_for_endpoint("http://svc", session=DownSession(),
                                 timeout_s=0.05, base_delay_s=0.01)


def test_feedback_collector(tmp_path):
    from code_intelligence_amd.label.feedback import FeedbackCollector
    bot_body = (Worker.BOT_MARKER + "\n"
                "Issue-Label Bot is automatically applying the labels below:\n\n"
                "| Label | Probability |\n|---|---|\n"
                "| kind/bug | 0.91 |\n| area/ops | 0.66 |\n\nthanks!")

    class Sess:
        def get(self, url, params=None, headers=None):
            class R:
                status_code = 200

                def raise_for_status(self):
                    pass

                def json(self):
                    if url.endswith("/issues/7/comments"):
                        return [{"body": "unrelated"},
                                {"body": bot_body,
                                 "reactions": {"+1": 3, "-1": 1}}]
                    return []
            return R()

    fc = FeedbackCollector(session=Sess())
    out = tmp_path / "fb.jsonl"
    records = fc.collect("o", "r", [7, 8], output=str(out))
    assert len(records) == 1
    assert records[0]["labels"] == ["kind/bug", "area/ops"]
    assert records[0]["score"] == 2
    assert out.read_text().count("\n") == 1


def test_github_issue_client_rest_contract():
    """GitHubIssueClient hits the documented REST endpoints with the
    token header (worker's GitHub boundary)."""
    from code_intelligence_amd.label.worker import GitHubIssueClient

    calls = []

    class Sess:
        def post(self, url, json=None, headers=None):
            calls.append(("POST", url, json, headers))
            class R:
                def raise_for_status(self):
                    pass
                def json(self):
                    return {}
            return R()

        def get(self, url, headers=None):
            calls.append(("GET", url, None, headers))
            class R:
                def raise_for_status(self):
                    pass
                def json(self):
                    return [{"body": "x"}]
            return R()

    class Tok:
        def auth_headers(self):
            return {"Authorization": "token sekrit"}

    c = GitHubIssueClient(token_generator=Tok(), session=Sess())
    c.add_labels("o", "r", 5, ["kind/bug"])
    c.add_comment("o", "r", 5, "hello")
    assert c.list_comments("o", "r", 5) == [{"body": "x"}]
    assert calls[0][1].endswith("/repos/o/r/issues/5/labels")
    assert calls[0][2] == {"labels": ["kind/bug"]}
    assert calls[0][3]["Authorization"] == "token sekrit"
    assert calls[1][1].endswith("/repos/o/r/issues/5/comments")
    assert calls[2][0] == "GET"


def test_combine_predictions_pairwise_reference_form():
    """reference combined_model_test.py:10-31 calls the static method with
    (left, right) dicts."""
    left = {"a": .1, "b": .9, "c": .5}
    right = {"a": .9, "b": .1, "d": .6}
    got = CombinedLabelModels._combine_predictions(left, right)
    assert got == {"a": .9, "b": .9, "c": .5, "d": .6}


def test_spool_queue_durable_acks(tmp_path):
    """A restarted consumer re-delivers only UNacked messages (Pub/Sub
    at-least-once semantics — not replay-all)."""
    from code_intelligence_amd.label.queueing import LocalQueue
    spool = tmp_path / "q.jsonl"
    q1 = LocalQueue(spool_path=spool)
    for i in range(30):
        q1.publish(repo_owner="o", repo_name="r", issue_num=i)
    for _ in range(12):
        m = q1.pull(timeout=0.05)
        m.ack()
    # message 13 pulled but NOT acked -> must come back after restart
    q1.pull(timeout=0.05)
    del q1
    q2 = LocalQueue(spool_path=spool)
    nums = []
    while True:
        m = q2.pull(timeout=0.05)
        if m is None:
            break
        nums.append(int(m.attributes["issue_num"]))
        m.ack()
    assert nums == list(range(12, 30))  # unacked 12..29 redelivered once
    del q2
    q3 = LocalQueue(spool_path=spool)  # everything acked now
    assert q3.pull(timeout=0.05) is None


def test_shipped_universal_artifact_loads_and_separates():
    """The committed model_files/universal artifact (trained by
    scripts/make_universal_artifact.py, report in
    docs/universal_model_report.md) loads and separates the three kinds
    on obviously-typed issues."""
    from pathlib import Path
    from code_intelligence_amd.label.universal_kind_label_model import \
        UniversalKindLabelModel
    root = Path(__file__).resolve().parents[1] / "model_files" / "universal"
    model = UniversalKindLabelModel.load(root)
    bug = model.predict_issue_labels(
        "o", "r", "crash with traceback", ["segfault error broken fails"])
    feat = model.predict_issue_labels(
        "o", "r", "add support for flag", ["implement option enhancement"])
    q = model.predict_issue_labels(
        "o", "r", "how to understand usage", ["question help why example"])
    assert "kind/bug" in bug and "kind/feature" not in bug
    assert "kind/feature" in feat
    assert "kind/question" in q


# ---- Pub/Sub REST adapter contract tests (recorded wire fixtures,
# the python analogue of the reference's RoundTripper tests:
# go/cmd/automl/pkg/client/client_test.go:18-31) -----------------------------

class _RecordedTransport:
    """Transport double: asserts request shapes, replays canned wire
    responses, records every call."""

    def __init__(self, fixtures):
        self.fixtures = list(fixtures)
        self.calls = []

    def __call__(self, method, url, body, headers):
        self.calls.append((method, url, body))
        for match, (status, payload) in self.fixtures:
            if match in url:
                return status, payload
        raise AssertionError(f"unexpected wire call {method} {url}")


def test_pubsub_rest_pull_ack_contract():
    import base64
    from code_intelligence_amd.label.queueing import PubSubRestQueue
    wire_msg = {"receivedMessages": [{
        "ackId": "ACK123",
        "message": {"data": base64.b64encode(b"payload").decode(),
                    "attributes": {"repo_owner": "kubeflow",
                                   "repo_name": "code-intelligence",
                                   "issue_num": "42"},
                    "messageId": "m-1"}}]}
    tr = _RecordedTransport([
        (":pull", (200, wire_msg)),
        (":acknowledge", (200, {})),
    ])
    q = PubSubRestQueue("proj", "events", "bot-sub", transport=tr,
                        token_provider=lambda: "tok")
    msg = q.pull()
    assert msg.data == b"payload"
    assert msg.attributes["issue_num"] == "42"
    msg.ack()
    methods = [(m, u.split("/v1/")[1]) for m, u, _ in tr.calls]
    assert methods[0] == ("POST", "projects/proj/subscriptions/bot-sub:pull")
    assert methods[1] == ("POST",
                          "projects/proj/subscriptions/bot-sub:acknowledge")
    assert tr.calls[1][2] == {"ackIds": ["ACK123"]}
    assert tr.calls[0][2] == {"maxMessages": 1}


def test_pubsub_rest_publish_and_empty_pull():
    import base64
    from code_intelligence_amd.label.queueing import PubSubRestQueue
    tr = _RecordedTransport([
        (":publish", (200, {"messageIds": ["555"]})),
        (":pull", (200, {})),
    ])
    q = PubSubRestQueue("proj", "events", "bot-sub", transport=tr)
    mid = q.publish(b"hello", installation_id=7)
    assert mid == "555"
    body = tr.calls[0][2]["messages"][0]
    assert base64.b64decode(body["data"]) == b"hello"
    assert body["attributes"] == {"installation_id": "7"}
    assert q.pull() is None  # empty pull -> no message, no crash


def test_pubsub_rest_idempotent_creates_and_errors():
    import pytest as _pytest
    from code_intelligence_amd.label.queueing import PubSubRestQueue
    tr = _RecordedTransport([
        ("/topics/events", (409, {"error": {"status": "ALREADY_EXISTS"}})),
        ("/subscriptions/bot-sub", (200, {})),
    ])
    q = PubSubRestQueue("proj", "events", "bot-sub", transport=tr)
    q.create_topic_if_not_exists()       # 409 tolerated (idempotent)
    q.create_subscription_if_not_exists()
    assert tr.calls[1][2]["topic"] == "projects/proj/topics/events"
    tr2 = _RecordedTransport([(":pull", (403, {"error": "denied"}))])
    q2 = PubSubRestQueue("proj", "events", "bot-sub", transport=tr2)
    with _pytest.raises(RuntimeError, match="403"):
        q2.pull()


def test_queue_from_env_selects_rest_adapter(monkeypatch):
    from code_intelligence_amd.label import queueing
    monkeypatch.setenv("PROJECT", "proj")
    monkeypatch.setenv("ISSUE_EVENT_TOPIC", "events")
    monkeypatch.setenv("ISSUE_EVENT_SUBSCRIPTION", "bot-sub")
    tr = _RecordedTransport([
        ("/topics/events", (200, {})),
        ("/subscriptions/bot-sub", (409, {})),
    ])
    q = queueing.queue_from_env(transport=tr)
    assert isinstance(q, queueing.PubSubRestQueue)
    assert len(tr.calls) == 2  # idempotent create calls on construction
