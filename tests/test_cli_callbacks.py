"""Coverage for CLIs, callbacks, schedules (table-driven, offline)."""
import json

import pytest
import torch

from code_intelligence_amd.control.automl_cli import main as automl_main
from code_intelligence_amd.control.registry import LocalModelRegistry
from code_intelligence_amd.label.cli import main as label_cli_main
from code_intelligence_amd.train.callbacks import (CSVLogger, EarlyStopping,
                                                   JSONRunLogger,
                                                   ReduceLROnPlateau)
from code_intelligence_amd.train.schedules import FlatSchedule, OneCycle
from code_intelligence_amd.train.train_cli import main as train_main


class _T:  # minimal trainer stand-in for callbacks
    lr_scale = 1.0
    model = None


def test_one_cycle_shape():
    s = OneCycle(max_lr=1.0, pct_start=0.3, div=25)
    lr0, mom0 = s.at(0.0)
    lr_peak, mom_peak = s.at(0.3)
    lr_end, mom_end = s.at(1.0)
    assert lr0 == pytest.approx(1.0 / 25)
    assert lr_peak == pytest.approx(1.0)
    assert lr_end < 1e-4                      # anneal to max_lr/final_div
    assert mom0 == pytest.approx(0.95)
    assert mom_peak == pytest.approx(0.85)
    assert mom_end == pytest.approx(0.95)
    # monotone up then down
    ups = [s.at(f)[0] for f in (0.0, 0.1, 0.2, 0.3)]
    downs = [s.at(f)[0] for f in (0.3, 0.5, 0.8, 1.0)]
    assert ups == sorted(ups)
    assert downs == sorted(downs, reverse=True)
    assert FlatSchedule(0.5).at(0.7) == (0.5, None)


def test_early_stopping_patience():
    es = EarlyStopping(patience=2)
    t = _T()
    assert not es.on_epoch_end(t, 0, {"valid_loss": 1.0})
    assert not es.on_epoch_end(t, 1, {"valid_loss": 1.1})
    assert not es.on_epoch_end(t, 2, {"valid_loss": 1.2})
    assert es.on_epoch_end(t, 3, {"valid_loss": 1.3})  # patience exceeded
    # improvement resets
    es2 = EarlyStopping(patience=1)
    es2.on_epoch_end(t, 0, {"valid_loss": 1.0})
    es2.on_epoch_end(t, 1, {"valid_loss": 1.5})
    assert not es2.on_epoch_end(t, 2, {"valid_loss": 0.5})


def test_reduce_lr_on_plateau():
    cb = ReduceLROnPlateau(patience=1, factor=0.5)
    t = _T()
    cb.on_epoch_end(t, 0, {"valid_loss": 1.0})
    cb.on_epoch_end(t, 1, {"valid_loss": 1.2})
    assert t.lr_scale == 1.0
    cb.on_epoch_end(t, 2, {"valid_loss": 1.3})
    assert t.lr_scale == 0.5


def test_csv_and_json_loggers(tmp_path):
    csv_cb = CSVLogger(tmp_path / "h.csv")
    run_cb = JSONRunLogger(tmp_path / "r.jsonl", config={"lr": 1}, every=2)
    t = _T()
    run_cb.on_train_begin(t)
    run_cb.on_step_end(t, 2, 0.5)
    run_cb.on_step_end(t, 3, 0.4)  # not logged (every=2)
    csv_cb.on_epoch_end(t, 0, {"train_loss": 1.0})
    run_cb.on_epoch_end(t, 0, {"train_loss": 1.0})
    assert "train_loss" in (tmp_path / "h.csv").read_text()
    lines = [json.loads(l) for l in open(tmp_path / "r.jsonl")]
    assert lines[0]["event"] == "run_begin"
    assert sum(1 for l in lines if l["event"] == "step") == 1


def test_train_cli_end_to_end(tmp_path, capsys):
    m = train_main(["--data_path", "synthetic:60", "--emb_sz", "16",
                    "--n_hid", "24", "--n_layers", "2", "--vocab_sz", "200",
                    "--bs", "4", "--bptt", "16", "--epochs", "1",
                    "--model_path", str(tmp_path), "--dtype", "fp32"])
    assert "valid_loss" in m
    assert (tmp_path / "best_enc.pth").exists()   # SaveModel encoder artifact
    assert (tmp_path / "history.csv").exists()
    sd = torch.load(tmp_path / "best_enc.pth", weights_only=True)
    assert "encoder.weight" in sd                 # fastai layout


def test_automl_cli_roundtrip(tmp_path, capsys):
    reg_dir = str(tmp_path / "reg")
    reg = LocalModelRegistry(reg_dir)
    rec = reg.create_training("d")
    reg.finish_training(rec.name)
    automl_main(["--registry", reg_dir, "get", "--dataset", "d"])
    out = json.loads(capsys.readouterr().out)
    assert out["latest_trained"]["name"] == rec.name
    automl_main(["--registry", reg_dir, "deploy", "--name", rec.name])
    assert json.loads(capsys.readouterr().out)["deployed"] is True
    automl_main(["--registry", reg_dir, "is-training", "--dataset", "d"])
    assert json.loads(capsys.readouterr().out) == {"isTraining": False}


def test_label_cli_publish_and_logs(tmp_path, capsys):
    spool = tmp_path / "spool.jsonl"
    label_cli_main(["label-issue", "--issue", "kubeflow/kf#7",
                    "--spool", str(spool)])
    assert "published" in capsys.readouterr().out
    from code_intelligence_amd.label.queueing import LocalQueue
    msg = LocalQueue(spool_path=str(spool)).pull(0.2)
    assert msg.attributes["repo_owner"] == "kubeflow"
    assert msg.attributes["issue_num"] == "7"
    logf = tmp_path / "w.jsonl"
    logf.write_text(json.dumps({"time": "t", "level": "INFO",
                                "message": "hello", "repo": "kf"}) + "\n")
    label_cli_main(["logs", "--path", str(logf)])
    assert "hello" in capsys.readouterr().out


def test_terminate_on_nan():
    from code_intelligence_amd.train.callbacks import TerminateOnNaN
    cb = TerminateOnNaN()
    cb.on_step_end(None, 5, 1.25)  # finite: no-op
    with pytest.raises(FloatingPointError):
        cb.on_step_end(None, 6, float("nan"))
    with pytest.raises(FloatingPointError):
        cb.on_step_end(None, 7, float("inf"))


def test_train_cli_qrnn_end_to_end(tmp_path):
    m = train_main(["--data_path", "synthetic:60", "--emb_sz", "16",
                    "--n_hid", "24", "--n_layers", "2", "--vocab_sz", "200",
                    "--bs", "4", "--bptt", "16", "--epochs", "1",
                    "--qrnn", "true",
                    "--model_path", str(tmp_path), "--dtype", "fp32"])
    assert "valid_loss" in m and m["valid_loss"] == m["valid_loss"]
    sd = torch.load(tmp_path / "best_enc.pth", weights_only=True)
    assert "rnns.0.weight_raw" in sd              # QRNN layout persisted


def test_train_cli_resume_continues(tmp_path):
    common = ["--data_path", "synthetic-markov:80", "--emb_sz", "16",
              "--n_hid", "24", "--n_layers", "2", "--vocab_sz", "150",
              "--bs", "4", "--bptt", "16", "--dtype", "fp32",
              "--one_cycle", "false", "--model_path", str(tmp_path)]
    ck = str(tmp_path / "resume.pt")
    m1 = train_main(common + ["--epochs", "1", "--save_checkpoint", ck])
    m2 = train_main(common + ["--epochs", "3", "--resume", ck])
    assert m2["valid_loss"] < m1["valid_loss"]  # training continued downhill


def test_periodic_checkpoint_resume(tmp_path):
    """PeriodicCheckpoint writes step-N.ckpt mid-epoch, rotates old ones,
    and the newest checkpoint resumes the exact training trajectory."""
    import torch
    from code_intelligence_amd.models.awd_lstm import AWDLSTM
    from code_intelligence_amd.train.trainer import LMTrainer, TrainConfig
    from code_intelligence_amd.train.callbacks import PeriodicCheckpoint

    torch.manual_seed(0)
    m = AWDLSTM(vocab_sz=64, emb_sz=16, n_hid=24, n_layers=2)
    tr = LMTrainer(m, TrainConfig(alpha=0, beta=0, one_cycle=False))
    cb = PeriodicCheckpoint(tmp_path, every_steps=2, keep=2)
    g = torch.Generator().manual_seed(5)

    def step(t):
        x = torch.randint(2, 64, (4, 8), generator=g)
        loss = t.train_step(x, torch.roll(x, -1, 1), lr=1e-3)
        cb.on_step_end(t, t.global_step, loss)
        return loss

    for _ in range(6):
        step(tr)
    ckpts = sorted(tmp_path.glob("step-*.ckpt"))
    assert [c.name for c in ckpts] == ["step-4.ckpt", "step-6.ckpt"]  # keep=2

    # resume from step-6 and continue 2 steps; compare against continuing
    # the original trainer with the same data stream
    torch.manual_seed(0)
    m2 = AWDLSTM(vocab_sz=64, emb_sz=16, n_hid=24, n_layers=2)
    tr2 = LMTrainer(m2, TrainConfig(alpha=0, beta=0, one_cycle=False))
    tr2.load_checkpoint(tmp_path / "step-6.ckpt")
    assert tr2.global_step == 6
    g2 = torch.Generator().manual_seed(99)
    g = torch.Generator().manual_seed(99)  # same continuation stream

    def step_with(t, gen):
        x = torch.randint(2, 64, (4, 8), generator=gen)
        return t.train_step(x, torch.roll(x, -1, 1), lr=1e-3)

    a = [step_with(tr, g) for _ in range(2)]
    b = [step_with(tr2, g2) for _ in range(2)]
    # hidden-state carry differs (resume resets hidden), so compare with
    # modest tolerance on the loss trajectory
    assert abs(a[0] - b[0]) < 0.2 and abs(a[1] - b[1]) < 0.2, (a, b)
