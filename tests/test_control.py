"""Control-plane tests: registry, ModelSync reconcile, needs-sync server,
chatbot label matching (reference: Go test techniques — fakes, goldens)."""

import pytest
import yaml

from code_intelligence_amd.control.chatbot import KubeflowLabels, create_app as chatbot_app
from code_intelligence_amd.control.modelsync import ModelSync, ModelSyncSpec
from code_intelligence_amd.control.needs_sync_server import (
    create_app as sync_app, read_deployed_setter)
from code_intelligence_amd.control.registry import LocalModelRegistry


def test_registry_lifecycle(tmp_path):
    reg = LocalModelRegistry(tmp_path)
    rec = reg.create_training("labels")
    assert reg.is_training("labels")
    assert reg.latest_trained("labels") is None
    reg.finish_training(rec.name, evaluation={"precision": 0.8, "recall": 0.6,
                                              "confidence": 0.5})
    assert not reg.is_training("labels")
    assert reg.latest_trained("labels").name == rec.name
    reg.deploy(rec.name)
    assert reg.latest_deployed("labels").name == rec.name
    # a newer deploy undeploys the old one
    rec2 = reg.create_training("labels")
    reg.finish_training(rec2.name)
    reg.deploy(rec2.name)
    assert reg.latest_deployed("labels").name == rec2.name
    assert not reg.get(rec.name).deployed
    assert reg.evaluation_at_confidence(rec.name)["precision"] == 0.8


class FakeHTTP:
    def __init__(self, payloads):
        self.payloads = list(payloads)

    def get(self, url, **kw):
        class R:
            def __init__(self, p):
                self._p = p
                self.status_code = 200

            def json(self):
                return self._p

            def raise_for_status(self):
                pass
        return R(self.payloads.pop(0))


def test_modelsync_reconcile_creates_and_skips_runs():
    ran = []

    def runner(cmd):
        ran.append(cmd)
        return 0

    spec = ModelSyncSpec(name="sync", needs_sync_url="http://x/needsSync",
                         run_command=["train"],
                         parameter_mapping={"name": "model"})
    ms = ModelSync(spec, session=FakeHTTP([
        {"needsSync": True, "parameters": {"name": "m-123"}},
        {"needsSync": True, "parameters": {"name": "m-123"}},
        {"needsSync": False},
    ]), runner=runner)
    r1 = ms.reconcile()
    assert "created" in r1
    assert ran == [["train", "--model=m-123"]]  # parameter mapping applied
    r2 = ms.reconcile()   # run Succeeded already (sync runner) -> creates again
    assert "created" in r2
    r3 = ms.reconcile()
    assert r3["needs_sync"] is False
    assert r3["succeeded"] == 2


def test_modelsync_requeues_on_error():
    class Boom:
        def get(self, url, **kw):
            raise ConnectionError("down")
    spec = ModelSyncSpec(name="s", needs_sync_url="http://x", run_command=["t"])
    ms = ModelSync(spec, session=Boom())
    r = ms.reconcile()
    assert r["requeue_after_s"] == spec.requeue_after_s


def test_modelsync_history_gc():
    spec = ModelSyncSpec(name="s", needs_sync_url="u", run_command=["t"],
                         successful_runs_history_limit=2)
    ms = ModelSync(spec, session=FakeHTTP(
        [{"needsSync": True}] * 5 + [{"needsSync": False}]),
        runner=lambda cmd: 0)
    for _ in range(5):
        ms.reconcile()
    r = ms.reconcile()
    assert r["succeeded"] == 2  # GC'd down to the history limit


def test_needs_sync_server(tmp_path):
    reg = LocalModelRegistry(tmp_path / "reg")
    rec = reg.create_training("labels")
    reg.finish_training(rec.name)
    cfg = tmp_path / "Kptfile.yaml"
    cfg.write_text(yaml.safe_dump({"openAPI": {"definitions": {
        "io.k8s.cli.setters.automl-model": {
            "x-k8s-cli": {"setter": {"name": "automl-model",
                                     "value": "old-model"}}}}}}))
    assert read_deployed_setter(cfg) == "old-model"
    app = sync_app(reg, "labels", cfg)
    c = app.test_client()
    r = c.get("/needsSync").get_json()
    assert r["needsSync"] is True
    assert r["parameters"]["name"] == rec.name
    # fresh model -> no retrain needed
    assert c.get("/needsTrain").get_json()["needsTrain"] is False


def test_chatbot_label_matching(tmp_path):
    labels = KubeflowLabels([
        {"name": "area/ops", "owners": ["alice"]},
        {"name": "platform/gcp", "owners": ["bob", "carol"]},
        {"name": "area/docs", "owners": []},
    ])
    # server.go table-driven matchLabels semantics
    assert [l["name"] for l in labels.match_labels("area")] == \
        ["area/ops", "area/docs"]
    assert [l["name"] for l in labels.match_labels("platform", "gcp")] == \
        ["platform/gcp"]
    app = chatbot_app(labels)
    c = app.test_client()
    r = c.post("/dialogflow/webhook", json={
        "queryResult": {"parameters": {"area": "platform", "value": "gcp"}}})
    text = r.get_json()["fulfillmentText"]
    assert "platform/gcp" in text and "bob" in text
    r2 = c.post("/dialogflow/webhook", json={
        "queryResult": {"parameters": {"area": "nosuch"}}})
    assert "could not find" in r2.get_json()["fulfillmentText"]


def test_k8s_manifests_parse_and_mirror_reference_topology():
    """deploy/k8s base: valid YAML, /healthz readiness probe kept, worker
    env contract (PROJECT/ISSUE_EVENT_*/MODEL_CONFIG) kept, 1-GPU
    embedding server replaces the reference's 9 CPU replicas."""
    import glob
    from pathlib import Path
    import yaml
    root = Path(__file__).resolve().parents[1] / "deploy" / "k8s"
    docs = []
    for f in glob.glob(str(root / "**" / "*.yaml"), recursive=True):
        docs += [d for d in yaml.safe_load_all(open(f)) if d]
    kinds = {(d["kind"], d["metadata"]["name"]): d for d in docs
             if "metadata" in d}
    emb = kinds[("Deployment", "issue-embedding-server")]
    c = emb["spec"]["template"]["spec"]["containers"][0]
    assert c["resources"]["limits"]["amd.com/gpu"] == 1
    assert c["readinessProbe"]["httpGet"]["path"] == "/healthz"
    worker = kinds[("Deployment", "label-worker")]
    assert worker["spec"]["replicas"] == 5
    env = {e["name"] for e in
           worker["spec"]["template"]["spec"]["containers"][0]["env"]}
    assert {"PROJECT", "ISSUE_EVENT_TOPIC", "ISSUE_EVENT_SUBSCRIPTION",
            "MODEL_CONFIG", "ISSUE_EMBEDDING_SERVICE"} <= env
    assert ("Service", "issue-embedding-server") in kinds
    assert ("Deployment", "modelsync") in kinds


def test_pipeline_runner_two_step_contract(tmp_path):
    """KFP-equivalent pipeline artifact: ordered steps, param templating,
    fail-fast, JSON run record (reference Training_Pipeline.ipynb's 2-step
    scrape -> train)."""
    import json
    from code_intelligence_amd.control.pipeline import (PipelineRunner,
                                                        load_pipeline)
    spec = load_pipeline("deploy/pipelines/scrape_train_pipeline.yaml")
    assert [s.name for s in spec.steps] == ["scrape-issue-embeddings",
                                            "train-repo-mlp"]
    calls = []

    def fake(cmd):
        calls.append(cmd)
        return 0

    rec = PipelineRunner(spec, run_dir=tmp_path, runner=fake).run(
        org="acme", repo="widgets")
    assert rec["status"] == "Succeeded" and len(calls) == 2
    assert "--org" in calls[0] and "acme" in calls[0]
    assert "widgets" in calls[1]
    saved = json.loads(next(tmp_path.glob("*.json")).read_text())
    assert saved["steps"][1]["status"] == "Succeeded"

    # fail-fast: step 1 failure stops the run before step 2
    def fail_first(cmd):
        return 3 if "embed_repo" in " ".join(cmd) else 0

    rec2 = PipelineRunner(spec, runner=fail_first).run()
    assert rec2["status"] == "Failed"
    assert len(rec2["steps"]) == 1 and rec2["steps"][0]["returncode"] == 3


def test_pipeline_cli_main(tmp_path):
    """CLI entry: runs a pipeline YAML end-to-end as subprocesses and
    exits nonzero on step failure."""
    import json
    import subprocess
    import sys
    from pathlib import Path
    pipe = tmp_path / "p.yaml"
    pipe.write_text("""
name: demo
params: {msg: world}
steps:
  - name: one
    command: [python, -c, "print('hello {msg}')"]
  - name: two
    command: [python, -c, "import sys; sys.exit(0)"]
""")
    root = Path(__file__).resolve().parents[1]
    r = subprocess.run(
        [sys.executable, "-m", "code_intelligence_amd.control.pipeline",
         str(pipe), "--run_dir", str(tmp_path / "runs"),
         "--param", "msg=there"],
        capture_output=True, text=True, timeout=120, cwd=root)
    assert r.returncode == 0, r.stderr
    out = json.loads(r.stdout.strip().splitlines()[-1])
    assert out["status"] == "Succeeded"
    rec = json.loads(next((tmp_path / "runs").glob("*.json")).read_text())
    assert rec["params"]["msg"] == "there"

    bad = tmp_path / "bad.yaml"
    bad.write_text("""
name: demo2
steps:
  - name: boom
    command: [python, -c, "import sys; sys.exit(3)"]
  - name: never
    command: [python, -c, "print('x')"]
""")
    r2 = subprocess.run(
        [sys.executable, "-m", "code_intelligence_amd.control.pipeline",
         str(bad), "--run_dir", str(tmp_path / "runs2")],
        capture_output=True, text=True, timeout=120, cwd=root)
    assert r2.returncode == 1
