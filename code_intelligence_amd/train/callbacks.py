"""Training callbacks with the reference's callback set semantics
(train.py:97-102: EarlyStoppingCallback(patience=2), SaveModelCallback
best-on-valid, ReduceLROnPlateauCallback(patience=1), CSVLogger, W&B
logging every 100 iterations -> JSONRunLogger here, offline)."""
from __future__ import annotations

import csv
import json
import math
import time
from pathlib import Path
from typing import List, Optional


class Callback:
    def on_train_begin(self, trainer): ...
    def on_epoch_end(self, trainer, epoch: int, metrics: dict) -> bool:
        """Return True to request early stop."""
        return False
    def on_step_end(self, trainer, step: int, loss: float): ...
    def on_train_end(self, trainer): ...


class CallbackList(Callback):
    def __init__(self, cbs: Optional[List[Callback]] = None):
        self.cbs = cbs or []

    def on_train_begin(self, trainer):
        for c in self.cbs:
            c.on_train_begin(trainer)

    def on_epoch_end(self, trainer, epoch, metrics):
        stop = False
        for c in self.cbs:
            stop = bool(c.on_epoch_end(trainer, epoch, metrics)) or stop
        return stop

    def on_step_end(self, trainer, step, loss):
        for c in self.cbs:
            c.on_step_end(trainer, step, loss)

    def on_train_end(self, trainer):
        for c in self.cbs:
            c.on_train_end(trainer)


class EarlyStopping(Callback):
    def __init__(self, monitor: str = "valid_loss", patience: int = 2,
                 min_delta: float = 0.0):
        self.monitor, self.patience, self.min_delta = monitor, patience, min_delta
        self.best = math.inf
        self.wait = 0

    def on_epoch_end(self, trainer, epoch, metrics):
        v = metrics.get(self.monitor)
        if v is None:
            return False
        if v < self.best - self.min_delta:
            self.best, self.wait = v, 0
            return False
        self.wait += 1
        return self.wait > self.patience


class SaveModel(Callback):
    """Save best-on-valid encoder+full model state (fastai SaveModelCallback)."""

    def __init__(self, path, monitor: str = "valid_loss", name: str = "best"):
        self.path, self.monitor, self.name = Path(path), monitor, name
        self.best = math.inf

    def on_epoch_end(self, trainer, epoch, metrics):
        import torch
        v = metrics.get(self.monitor)
        if v is not None and v < self.best:
            self.best = v
            self.path.mkdir(parents=True, exist_ok=True)
            torch.save(trainer.model.state_dict(), self.path / f"{self.name}.pth")
            trainer.model.save_encoder(self.path / f"{self.name}_enc.pth")
        return False


class ReduceLROnPlateau(Callback):
    def __init__(self, monitor: str = "valid_loss", patience: int = 1,
                 factor: float = 0.2):
        self.monitor, self.patience, self.factor = monitor, patience, factor
        self.best = math.inf
        self.wait = 0

    def on_epoch_end(self, trainer, epoch, metrics):
        v = metrics.get(self.monitor)
        if v is None:
            return False
        if v < self.best:
            self.best, self.wait = v, 0
        else:
            self.wait += 1
            if self.wait > self.patience:
                trainer.lr_scale *= self.factor
                self.wait = 0
        return False


class CSVLogger(Callback):
    def __init__(self, path):
        self.path = Path(path)
        self.rows = []

    def on_epoch_end(self, trainer, epoch, metrics):
        self.rows.append({"epoch": epoch, **metrics})
        self.path.parent.mkdir(parents=True, exist_ok=True)
        with open(self.path, "w", newline="") as f:
            w = csv.DictWriter(f, fieldnames=sorted({k for r in self.rows for k in r}))
            w.writeheader()
            w.writerows(self.rows)
        return False


class JSONRunLogger(Callback):
    """Offline stand-in for the reference's W&B run logging (train.py:75-81,
    36-38: config + loss every `every` steps) — appends JSON lines."""

    def __init__(self, path, config: Optional[dict] = None, every: int = 100):
        self.path = Path(path)
        self.config = config or {}
        self.every = every
        self.t0 = time.time()

    def on_train_begin(self, trainer):
        self.path.parent.mkdir(parents=True, exist_ok=True)
        with open(self.path, "a") as f:
            f.write(json.dumps({"event": "run_begin", "config": self.config}) + "\n")

    def _log(self, obj):
        with open(self.path, "a") as f:
            f.write(json.dumps(obj) + "\n")

    def on_step_end(self, trainer, step, loss):
        if step % self.every == 0:
            self._log({"event": "step", "step": step, "loss": loss,
                       "elapsed_s": round(time.time() - self.t0, 3)})

    def on_epoch_end(self, trainer, epoch, metrics):
        self._log({"event": "epoch", "epoch": epoch, **metrics})
        return False


class TerminateOnNaN(Callback):
    """Abort training on the first non-finite loss instead of burning the
    rest of the schedule (the reference's fastai loop would run to
    completion on NaN). Raises FloatingPointError with step context."""

    def on_step_end(self, trainer, step, loss):
        if loss != loss or loss in (float("inf"), float("-inf")):
            raise FloatingPointError(
                f"non-finite training loss {loss} at step {step} "
                f"(lr={getattr(trainer, 'last_lr', None)})")


class PeriodicCheckpoint(Callback):
    """Full-state checkpoint every N optimizer steps (mid-epoch).

    The reference only checkpointed per-epoch via SaveModelCallback; at
    the reference's pretraining scale (a week of one-cycle on 16.7M
    issues) an epoch is hours, so step-granular resume matters. Writes
    ``<path>.tmp`` then atomically renames, keeps the last ``keep``
    files (``step-<N>.ckpt``)."""

    def __init__(self, out_dir, every_steps: int = 1000, keep: int = 2):
        from pathlib import Path
        self.out_dir = Path(out_dir)
        self.every = max(1, int(every_steps))
        self.keep = keep

    def on_step_end(self, trainer, step: int, loss: float):
        if step == 0 or step % self.every:
            return
        import os
        self.out_dir.mkdir(parents=True, exist_ok=True)
        path = self.out_dir / f"step-{step}.ckpt"
        tmp = path.with_suffix(".ckpt.tmp")
        trainer.save_checkpoint(tmp)
        os.replace(tmp, path)
        ckpts = sorted(self.out_dir.glob("step-*.ckpt"),
                       key=lambda p: int(p.stem.split("-")[1]))
        for old in ckpts[: -self.keep]:
            old.unlink(missing_ok=True)
