"""LM pretraining loop — the MI355X-native equivalent of the reference's
fastai fit/fit_one_cycle (train.py:104-113) including AR/TAR activation
regularization (fastai RNNTrainer defaults alpha=2, beta=1) and the
callback set of train.py:97-102.

The loss path never materializes full logits: encoder -> output dropout ->
fused tied-decoder CE (ops/crossentropy.py, K6)."""
from __future__ import annotations

import dataclasses
import math
import time
from typing import Iterable, Optional

import torch
from torch import nn

from ..models.awd_lstm import AWDLSTM
from ..ops.crossentropy import tied_decoder_accuracy, tied_decoder_ce
from ..ops.adam import FusedAdamW
from ..parallel.ddp import DistributedGrads
from .callbacks import Callback, CallbackList
from .schedules import OneCycle, FlatSchedule


@dataclasses.dataclass
class TrainConfig:
    lr: float = 1.3e-3            # reference default train.py:44
    wd: float = 0.012             # train.py:45
    one_cycle: bool = True
    cycle_len: int = 1
    betas: tuple = (0.9, 0.99)    # fastai Adam default (0.9, 0.99)
    alpha: float = 2.0            # AR
    beta: float = 1.0             # TAR
    clip: float = 0.0
    bucket_mb: float = 64.0


class LMTrainer:
    def __init__(self, model: AWDLSTM, cfg: TrainConfig = TrainConfig(),
                 callbacks: Optional[list[Callback]] = None,
                 distributed: bool = False):
        self.model = model
        self.cfg = cfg
        self.cbs = CallbackList(callbacks)
        self.opt = FusedAdamW(self._param_groups(), lr=cfg.lr,
                              betas=cfg.betas, weight_decay=cfg.wd)
        self.lr_scale = 1.0
        if distributed:
            from ..parallel.ddp import broadcast_parameters
            broadcast_parameters(model)
            self.dist = DistributedGrads(model, bucket_mb=cfg.bucket_mb)
        else:
            self.dist = None
        self.global_step = 0
        self.epoch = 0  # epochs completed (checkpoint resume)

    def _param_groups(self):
        seen, uniq = set(), []
        for p in self.model.parameters():
            if p.requires_grad and id(p) not in seen:
                seen.add(id(p))
                uniq.append(p)
        return uniq

    def loss_on_batch(self, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        enc = self.model.encoder
        dec = self.model.decoder
        raw_outputs, outputs = enc(x)
        out = dec.output_dp(outputs[-1])
        loss = tied_decoder_ce(out.reshape(-1, out.shape[-1]),
                               dec.decoder.weight, dec.decoder.bias,
                               y.reshape(-1))
        if self.model.training and (self.cfg.alpha or self.cfg.beta):
            # fused AR/TAR kernel: one pass over the (B,T,H) activations
            # instead of ~6 eager elementwise/reduce kernels (ops/artar.py)
            from ..ops.artar import artar_loss
            loss = loss + artar_loss(out, raw_outputs[-1],
                                     self.cfg.alpha, self.cfg.beta)
        return loss

    def train_step(self, x, y, lr: float, mom: Optional[float] = None,
                   micro_batches: Optional[list] = None) -> float:
        """One optimizer step. ``micro_batches`` (list of (x, y)) enables
        gradient accumulation: losses averaged, all-reduce only on the
        final micro-step (DDP no_sync)."""
        for g in self.opt.param_groups:
            g["lr"] = lr * self.lr_scale
            if mom is not None:
                g["betas"] = (mom, g["betas"][1])
        self.opt.zero_grad(set_to_none=True)
        if micro_batches:
            n = len(micro_batches)
            total = 0.0
            for i, (mx, my) in enumerate(micro_batches):
                last = i == n - 1
                if self.dist is not None:
                    self.dist.prepare(sync=last)
                loss = self.loss_on_batch(mx, my) / n
                loss.backward()
                total += float(loss.detach())
            loss = total  # reported value: mean loss over micro-batches
        else:
            if self.dist is not None:
                self.dist.prepare()
            loss = self.loss_on_batch(x, y)
            loss.backward()
        ref_x = x if x is not None else (micro_batches[0][0]
                                         if micro_batches else None)
        if ref_x is not None and ref_x.is_cuda:
            from ..ops.lstm import sync_dw_stream
            sync_dw_stream()  # side-stream dW grads (CI_SIDE_DW) ordered
        if self.dist is not None:
            self.dist.finalize()
        if self.cfg.clip:
            torch.nn.utils.clip_grad_norm_(self._param_groups(), self.cfg.clip)
        self.opt.step()
        self.global_step += 1
        return float(loss.detach()) if torch.is_tensor(loss) else float(loss)

    @torch.no_grad()
    def evaluate(self, loader: Iterable, with_accuracy: bool = True) -> dict:
        self.model.eval()
        self.model.reset()
        tot, n, correct = 0.0, 0, 0.0
        dec = self.model.decoder
        for x, y in loader:
            enc_raw, enc_out = self.model.encoder(x)
            out = dec.output_dp(enc_out[-1])
            loss = tied_decoder_ce(out.reshape(-1, out.shape[-1]),
                                   dec.decoder.weight, dec.decoder.bias,
                                   y.reshape(-1))
            tot += float(loss) * x.numel()
            if with_accuracy:
                correct += float(tied_decoder_accuracy(
                    out, dec.decoder.weight, dec.decoder.bias, y)) * x.numel()
            n += x.numel()
        self.model.train()
        self.model.reset()
        vl = tot / max(n, 1)
        metrics = {"valid_loss": vl, "valid_ppl": math.exp(min(vl, 30.0))}
        if with_accuracy:
            metrics["valid_acc"] = correct / max(n, 1)
        return metrics

    # --- checkpoint/resume (full training state; the reference only
    # checkpoints model weights via fastai SaveModelCallback) -------------
    def save_checkpoint(self, path) -> None:
        state = {
            "model": self.model.state_dict(),
            "optimizer": self.opt.state_dict(),
            "lr_scale": self.lr_scale,
            "global_step": self.global_step,
            "epoch": self.epoch,
            "rng": torch.get_rng_state(),
        }
        if torch.cuda.is_available():
            state["rng_cuda"] = torch.cuda.get_rng_state_all()
        torch.save(state, path)

    def load_checkpoint(self, path, map_location="cpu") -> None:
        ckpt = torch.load(path, map_location=map_location, weights_only=False)
        self.model.load_state_dict(ckpt["model"])
        self.opt.load_state_dict(ckpt["optimizer"])
        self.lr_scale = ckpt.get("lr_scale", 1.0)
        self.global_step = ckpt.get("global_step", 0)
        self.epoch = ckpt.get("epoch", 0)
        if "rng" in ckpt:
            # RNG states must stay CPU ByteTensors even when the rest of
            # the checkpoint is mapped to cuda (map_location="cuda"
            # otherwise breaks set_rng_state — caught by the GPU
            # deployed-shape resume check, round 2)
            torch.set_rng_state(ckpt["rng"].cpu().to(torch.uint8))
        if "rng_cuda" in ckpt and torch.cuda.is_available():
            torch.cuda.set_rng_state_all(
                [s.cpu().to(torch.uint8) for s in ckpt["rng_cuda"]])

    def fit(self, train_loader, valid_loader=None, epochs: int = 1,
            one_cycle: Optional[bool] = None) -> dict:
        """Train up to ``epochs`` TOTAL epochs. ``self.epoch`` (epochs already
        completed — restored by load_checkpoint, advanced by previous fit
        calls) is the starting point, so a resumed or repeated fit continues
        the same trajectory instead of restarting; set ``trainer.epoch = 0``
        for a fresh run."""
        cfg = self.cfg
        use_oc = cfg.one_cycle if one_cycle is None else one_cycle
        # reference: fit_one_cycle(cycle_len, max_lr=lr*2) (train.py:109-111)
        sched = OneCycle(cfg.lr * 2) if use_oc else FlatSchedule(cfg.lr)
        n_total = None
        try:
            n_total = len(train_loader) * epochs
        except TypeError:
            pass
        self.cbs.on_train_begin(self)
        metrics: dict = {}
        self.model.train()
        t0 = time.time()
        start_epoch = self.epoch  # resume skips completed epochs
        step = start_epoch * (n_total // epochs) if n_total else 0
        if start_epoch and hasattr(train_loader, "epoch"):
            train_loader.epoch = start_epoch  # reproduce the shuffle order
        for epoch in range(start_epoch, epochs):
            # fresh hidden state per epoch (fastai RNNTrainer.on_epoch_begin
            # parity; also makes epoch-granular checkpoint resume exact)
            self.model.reset()
            losses = []
            for x, y in train_loader:
                frac = (step / n_total) if n_total else 0.5
                lr, mom = sched.at(frac) if use_oc else (cfg.lr, None)
                loss = self.train_step(x, y, lr, mom)
                losses.append(loss)
                self.cbs.on_step_end(self, self.global_step, loss)
                step += 1
            self.epoch = epoch + 1
            metrics = {"train_loss": sum(losses) / max(len(losses), 1),
                       "time_s": round(time.time() - t0, 2)}
            if valid_loader is not None:
                metrics.update(self.evaluate(valid_loader))
            if self.cbs.on_epoch_end(self, epoch, metrics):
                break
        self.cbs.on_train_end(self)
        return metrics
