"""Train and ship the universal 3-kind (bug/feature/question) model
artifact ("an untrained architecture is a shell").

There is no network (and therefore no real issue corpus) in this image,
so the corpus is synthetic-but-structured: each issue's text is drawn
from kind-correlated vocabulary pools plus shared noise, the honest
offline stand-in for the reference's 'kind' labels. The artifact (~1 MB)
is committed under model_files/universal/ together with a held-out
AUC / threshold report (docs/universal_model_report.md) evaluated at the
reference's decision thresholds (.52 bug/feature, .60 question —
/root/reference/py/label_microservice/universal_kind_label_model.py:50-51).

Run: python scripts/make_universal_artifact.py [--out model_files/universal]
"""
import sys
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import argparse
import json
import random

import numpy as np
import torch

KIND_WORDS = {
    "bug": ["crash", "error", "traceback", "exception", "segfault", "fails",
            "broken", "regression", "stacktrace", "panic", "hang", "leak"],
    "feature": ["add", "support", "implement", "proposal", "enhancement",
                "would", "feature", "extend", "option", "flag", "allow"],
    "question": ["how", "why", "what", "help", "question", "clarify",
                 "understand", "documentation", "usage", "example", "where"],
}
COMMON = ["the", "issue", "when", "running", "version", "kubernetes",
          "pipeline", "model", "training", "cluster", "deploy", "config",
          "log", "file", "install", "python", "container", "node", "pod",
          "image", "release", "update", "works", "using", "setup", "tests"]


def synth_issue(rng: random.Random, kinds):
    words = []
    for k in kinds:
        words += rng.choices(KIND_WORDS[k], k=rng.randint(3, 7))
    words += rng.choices(COMMON, k=rng.randint(15, 40))
    # cross-talk noise: a few words from a random other pool
    noise_kind = rng.choice(list(KIND_WORDS))
    words += rng.choices(KIND_WORDS[noise_kind], k=rng.randint(0, 2))
    rng.shuffle(words)
    split = rng.randint(4, 9)
    return " ".join(words[:split]), " ".join(words[split:])


def make_corpus(n, seed):
    rng = random.Random(seed)
    events = []
    for i in range(n):
        r = rng.random()
        if r < 0.12:
            kinds = rng.sample(list(KIND_WORDS), 2)  # multi-label
        elif r < 0.2:
            kinds = []                               # unlabeled
        else:
            kinds = [rng.choice(list(KIND_WORDS))]
        title, body = synth_issue(rng, kinds)
        events.append({"org": "kubeflow", "repo": "synthetic",
                       "issue_num": i + 1, "title": title, "body": body,
                       "labels": [f"kind/{k}" for k in kinds],
                       "updated_at": "2026-01-01T00:00:00Z"})
    return events


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--out", default="model_files/universal")
    p.add_argument("--report", default="docs/universal_model_report.md")
    p.add_argument("--n", type=int, default=4000)
    p.add_argument("--epochs", type=int, default=8)
    args = p.parse_args()
    torch.manual_seed(0)

    import tempfile
    from code_intelligence_amd.gh.bigquery import write_archive_events
    from code_intelligence_amd.label.trainers import (train_universal_model,
                                                      kind_targets)
    from code_intelligence_amd.label.universal_kind_label_model import \
        UniversalKindLabelModel

    events = make_corpus(args.n, seed=11)
    test_events = make_corpus(800, seed=99)
    with tempfile.TemporaryDirectory() as td:
        write_archive_events(events, Path(td) / "events.jsonl")
        model = train_universal_model("kubeflow", archive_root=td,
                                      epochs=args.epochs, max_vocab=4000,
                                      prefix="kind/")
    model.save(args.out)

    # held-out evaluation at the reference thresholds (raw sigmoid probs)
    y = kind_targets([e["labels"] for e in test_events])
    raw = np.zeros_like(y)
    with torch.no_grad():
        for i, e in enumerate(test_events):
            logits = model.net(model._encode(e["title"], [e["body"]]))
            raw[i] = torch.sigmoid(logits)[0].numpy()
    report = ["# Universal kind model — held-out report (synthetic corpus)",
              "",
              f"Train n={args.n}, test n=800, vocab 4000, epochs "
              f"{args.epochs}. Reference thresholds .52/.52/.60 "
              "(universal_kind_label_model.py:50-51).", "",
              "| class | AUC | precision@thr | recall@thr | support |",
              "|---|---|---|---|---|"]
    for j, cls in enumerate(model.CLASS_NAMES):
        thr = model.thresholds[cls]
        p_ = raw[:, j]
        t = y[:, j]
        order = np.argsort(-p_)
        ts = t[order]
        pos = ts.sum()
        neg = len(ts) - pos
        ranks = np.argsort(np.argsort(p_))
        auc = (ranks[t == 1].sum() - pos * (pos - 1) / 2) / max(pos * neg, 1)
        sel = p_ >= thr
        prec = t[sel].mean() if sel.any() else float("nan")
        rec = t[sel].sum() / max(pos, 1)
        report.append(f"| kind/{cls} | {auc:.4f} | {prec:.3f} | {rec:.3f} "
                      f"| {int(pos)} |")
    Path(args.report).write_text("\n".join(report) + "\n")
    print("\n".join(report))


if __name__ == "__main__":
    main()
