"""Fused multi-tensor Adam with decoupled weight decay (K8).

Reference semantics: fastai one-cycle AdamW (fit_one_cycle, train.py:106-113;
wd=0.012 default train.py:45). On ROCm the step is a single multi-tensor HIP
kernel per dtype-group: fp32 master weights + Adam moments update, then a
cast-back to the (bf16) working parameters. CPU path: plain torch loop.
"""
from __future__ import annotations

from typing import Iterable, List

import torch
from torch import Tensor

from . import extension as ext

__all__ = ["FusedAdamW"]


class FusedAdamW(torch.optim.Optimizer):
    def __init__(self, params: Iterable[Tensor], lr: float = 1e-3,
                 betas=(0.9, 0.99), eps: float = 1e-8, weight_decay: float = 0.01):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)

    def load_state_dict(self, state_dict):
        """torch's Optimizer.load_state_dict casts floating state to each
        PARAM's dtype — for bf16 params that silently downcasts the fp32
        master weights/moments on every resume (and the fused kernel then
        rejects the bf16 masters). Restore fp32 after the base load.
        Caught by the GPU deployed-shape resume check (round 2)."""
        super().load_state_dict(state_dict)
        for st in self.state.values():
            for k in ("master", "exp_avg", "exp_avg_sq"):
                v = st.get(k)
                if torch.is_tensor(v) and v.dtype != torch.float32:
                    st[k] = v.to(torch.float32)

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        for group in self.param_groups:
            lr = group["lr"]
            beta1, beta2 = group["betas"]
            eps, wd = group["eps"], group["weight_decay"]
            # Bias correction depends on each param's own step count, which
            # can diverge when grads are intermittently None (gradual
            # unfreezing). Batch the fused launch per step-count — in
            # steady state that is one launch, exactly as before.
            by_step: dict = {}
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = 0
                    state["master"] = p.detach().to(torch.float32).clone() \
                        if p.dtype != torch.float32 else p
                    state["exp_avg"] = torch.zeros_like(state["master"])
                    state["exp_avg_sq"] = torch.zeros_like(state["master"])
                state["step"] += 1
                by_step.setdefault(state["step"], []).append((p, state))
            for step_t, entries in by_step.items():
                params: List[Tensor] = [p for p, _ in entries]
                grads: List[Tensor] = [p.grad for p, _ in entries]
                masters = [s["master"] for _, s in entries]
                exp_avgs = [s["exp_avg"] for _, s in entries]
                exp_avg_sqs = [s["exp_avg_sq"] for _, s in entries]
                bc1 = 1 - beta1 ** step_t
                bc2 = 1 - beta2 ** step_t
                if params[0].is_cuda:
                    lib = ext.require()
                    lib.fused_adamw(params, grads, masters, exp_avgs,
                                    exp_avg_sqs, lr, beta1, beta2, eps, wd,
                                    bc1, bc2)
                else:
                    for p, g, m, ea, eas in zip(params, grads, masters,
                                                exp_avgs, exp_avg_sqs):
                        gf = g.to(torch.float32)
                        m.mul_(1 - lr * wd)
                        ea.mul_(beta1).add_(gf, alpha=1 - beta1)
                        eas.mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
                        denom = (eas / bc2).sqrt_().add_(eps)
                        m.addcdiv_(ea / bc1, denom, value=-lr)
                        if m.data_ptr() != p.data_ptr():
                            p.copy_(m.to(p.dtype))
        return loss
