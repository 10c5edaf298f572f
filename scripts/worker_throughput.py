"""Label-worker throughput on MI355X: events/s through the FULL path
(queue pull -> predictor -> repo-MLP over the GPU embedding server ->
label application), deployed-shape encoder, repo MLP trained on the fly.
"""
import sys, time, json
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
import torch

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

wd = Path("/tmp/wt")
wd.mkdir(parents=True, exist_ok=True)
N_EVENTS = 300
events = []
for i in range(max(N_EVENTS, 120)):
    label = "bug" if i % 2 == 0 else "feature"
    word = "crash" if label == "bug" else "request"
    events.append({"org": "demo", "repo": "repo", "issue_num": i,
                   "title": f"{word} w{i % 25}", "body": f"{word} body text",
                   "labels": [label], "updated_at": "2024-01-01T00:00:00Z"})
bigquery.write_archive_events(events, wd / "archive" / "events.jsonl")

torch.manual_seed(0)
vocab = Vocab(defaults_specials + [f"w{i}" for i in range(59900)] +
              ["crash", "request", "body", "text"])
model = AWDLSTM(vocab_sz=len(vocab), emb_sz=800, n_hid=2400, n_layers=4)
wrapper = InferenceWrapper(encoder=model.encoder, vocab=vocab)
store = ObjectStore(root=wd / "store")
run_training_pipeline("demo", "repo", wrapper, store=store,
                      archive_root=wd / "archive")
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


repo_model = RepoSpecificLabelModel.from_repo("demo", "repo", store=store,
                                              session=Session())
predictor = IssueLabelPredictor(model_config={}, universal=repo_model)
applied = [0]


class GH:
    def add_labels(self, o, r, n, labels):
        applied[0] += 1

    def add_comment(self, o, r, n, body):
        pass

    def list_comments(self, o, r, n):
        return []


broker = LocalBroker(wd / "broker")
q = broker.create_subscription_if_not_exists("issue-events", "wt")
for i in range(N_EVENTS):
    broker.publish("issue-events", repo_owner="demo", repo_name="repo",
                   issue_num=i)
worker = Worker(queue=q, predictor=predictor, github=GH(),
                repo_config_fn=lambda o, r: None)
# warmup a few
for _ in range(10):
    m = q.pull(0.5)
    ev = events[int(m.attributes["issue_num"])]
    preds = predictor.predict_labels_for_data("demo", "repo", ev["title"],
                                              [ev["body"]])
    worker.add_labels_to_issue("demo", "repo", 0, preds,
                               issue_data={"labels": [],
                                           "removed_labels": []})
    m.ack()
t0 = time.perf_counter()
done = 0
while done < N_EVENTS - 10:
    m = q.pull(0.5)
    if m is None:
        break
    ev = events[int(m.attributes["issue_num"])]
    preds = predictor.predict_labels_for_data("demo", "repo", ev["title"],
                                              [ev["body"]])
    worker.add_labels_to_issue("demo", "repo", int(m.attributes["issue_num"]),
                               preds, issue_data={"labels": [],
                                                  "removed_labels": []})
    m.ack()
    done += 1
dt = time.perf_counter() - t0
print(json.dumps({"events": done, "events_per_s": round(done / dt, 1),
                  "labels_applied": applied[0],
                  "p_mean_ms": round(dt / done * 1e3, 2)}))
