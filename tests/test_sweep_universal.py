"""Sweep runner + universal-model training tests."""

import numpy as np

from code_intelligence_amd.gh import bigquery
from code_intelligence_amd.label.trainers import kind_targets, train_universal_model
from code_intelligence_amd.train.sweep import SweepRunner, sample_space


def test_sample_space_random_and_grid():
    spec = {"method": "random", "parameters": {
        "lr": {"min": 1e-4, "max": 1e-2, "log": True},
        "n_hid": {"min": 100, "max": 200, "type": "int"},
        "one_cycle": {"values": [True, False]}}}
    trials = sample_space(spec, 20, seed=1)
    assert len(trials) == 20
    assert all(1e-4 <= t["lr"] <= 1e-2 for t in trials)
    assert all(isinstance(t["n_hid"], int) for t in trials)
    grid = sample_space({"method": "grid", "parameters": {
        "a": {"values": [1, 2]}, "b": {"values": ["x", "y", "z"]}}}, 0)
    assert len(grid) == 6


def test_sweep_runner_leaderboard(tmp_path):
    def fake_trial(cfg, base, out, tid, gpu):
        return {"trial": tid, "config": cfg, "gpu": gpu, "returncode": 0,
                "metrics": {"valid_loss": cfg["lr"] * 100}}
    spec = {"method": "grid", "parameters": {"lr": {"values": [0.03, 0.01, 0.02]}}}
    r = SweepRunner(spec, tmp_path, n_gpus=2, trial_fn=fake_trial)
    results = r.run(0)
    assert len(results) == 3
    best = r.best("valid_loss")
    assert best["config"]["lr"] == 0.01
    assert (tmp_path / "leaderboard.jsonl").exists()


def test_kind_targets():
    y = kind_targets([["kind/bug"], ["feature", "area/x"], ["question", "bug"], []])
    assert y.tolist() == [[1, 0, 0], [0, 1, 0], [1, 0, 1], [0, 0, 0]]


def test_train_universal_model_learns(tmp_path):
    rng = np.random.default_rng(0)
    events = []
    for i in range(120):
        kind = ["bug", "feature", "question"][i % 3]
        word = {"bug": "crash", "feature": "request", "question": "how"}[kind]
        events.append({"org": "o", "repo": "r", "issue_num": i,
                       "title": f"{word} issue {word}", "body": f"{word} body",
                       "labels": [f"kind/{kind}"],
                       "updated_at": "2024-01-01T00:00:00Z"})
    bigquery.write_archive_events(events, tmp_path / "a.jsonl")
    m = train_universal_model("o", archive_root=tmp_path, epochs=12)
    preds = m.predict_issue_labels("o", "r", "crash issue crash", ["crash body"])
    assert "bug" in preds, preds
    assert "feature" not in preds
    m.save(tmp_path / "model")
