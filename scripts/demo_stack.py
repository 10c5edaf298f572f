"""End-to-end stack demo on synthetic data (CPU or GPU):
archive -> LM artifacts -> per-repo classifier pipeline -> embedding
server (in-process) -> predictor -> worker labels a queued issue event.

  python scripts/demo_stack.py [--workdir /tmp/ci_demo]
"""
import sys
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import argparse
import json

import torch


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--workdir", default="/tmp/ci_demo")
    args = ap.parse_args()
    wd = Path(args.workdir)
    wd.mkdir(parents=True, exist_ok=True)

    from code_intelligence_amd.engine.inference import InferenceWrapper
    from code_intelligence_amd.gh import bigquery
    from code_intelligence_amd.gh.gcs_util import ObjectStore
    from code_intelligence_amd.label.issue_label_predictor import IssueLabelPredictor
    from code_intelligence_amd.label.queueing import LocalBroker
    from code_intelligence_amd.label.repo_specific_model import RepoSpecificLabelModel
    from code_intelligence_amd.label.trainers import run_training_pipeline
    from code_intelligence_amd.label.worker import Worker
    from code_intelligence_amd.models.awd_lstm import AWDLSTM
    from code_intelligence_amd.serve.app import create_app
    from code_intelligence_amd.text.tokenizer import Vocab, defaults_specials

    print("== 1. synthetic issue archive")
    events = []
    for i in range(120):
        label = "bug" if i % 2 == 0 else "feature"
        word = "crash" if label == "bug" else "request"
        events.append({"org": "demo", "repo": "repo", "issue_num": i,
                       "title": f"{word} w{i % 25}", "body": f"{word} body text",
                       "labels": [label], "updated_at": "2024-01-01T00:00:00Z"})
    bigquery.write_archive_events(events, wd / "archive" / "events.jsonl")

    print("== 2. LM encoder (random-init demo weights)")
    torch.manual_seed(0)
    vocab = Vocab(defaults_specials + [f"w{i}" for i in range(500)] +
                  ["crash", "request", "body", "text"])
    model = AWDLSTM(vocab_sz=len(vocab), emb_sz=32, n_hid=64, n_layers=2)
    wrapper = InferenceWrapper(encoder=model.encoder, vocab=vocab)

    print("== 3. per-repo classifier pipeline")
    store = ObjectStore(root=wd / "store")
    result = run_training_pipeline("demo", "repo", wrapper, store=store,
                                   archive_root=wd / "archive")
    print(json.dumps({k: v for k, v in result.items() if k != "thresholds"},
                     default=str))

    print("== 4. embedding server (in-process flask test client)")
    app = create_app(wrapper=wrapper)
    client = app.test_client()

    class Session:
        def post(self, url, json=None, **kw):
            path = "/" + url.split("/", 3)[-1] if "://" in url else url
            r = client.post(path, json=json)

            class R:
                status_code = r.status_code
                content = r.data
            return R()

    print("== 5. predictor + worker consume a queued event")
    repo_model = RepoSpecificLabelModel.from_repo("demo", "repo", store=store,
                                                  session=Session())
    predictor = IssueLabelPredictor(model_config={}, universal=repo_model)

    class PrintGitHub:
        def add_labels(self, o, r, n, labels):
            print(f"   -> would label {o}/{r}#{n}: {labels}")

        def add_comment(self, o, r, n, body):
            print("   -> would comment:\n" +
                  "\n".join("      " + l for l in body.splitlines()[:6]))

        def list_comments(self, o, r, n):
            return []

    broker = LocalBroker(wd / "broker")
    q = broker.create_subscription_if_not_exists("issue-events", "worker")
    broker.publish("issue-events", repo_owner="demo", repo_name="repo",
                   issue_num=3)
    worker = Worker(queue=q, predictor=predictor, github=PrintGitHub(),
                    repo_config_fn=lambda o, r: None)
    msg = q.pull(0.5)
    preds = predictor.predict_labels_for_data(
        "demo", "repo", events[3]["title"], [events[3]["body"]])
    print(f"   predictions for #{msg.attributes['issue_num']}: {preds}")
    worker.add_labels_to_issue("demo", "repo", 3, preds,
                               issue_data={"labels": [], "removed_labels": []})
    msg.ack()
    print("== demo complete")


if __name__ == "__main__":
    main()
