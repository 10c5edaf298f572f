"""CI_CE_FP8R quality check: train the deployed-shape model for K steps
from the same seed with bf16-resident vs fp8-resident CE and print the
loss trajectories. Decides whether fp8r ships as default.

Run on an MI355X box: python scripts/fp8r_check.py --steps 30
"""
import sys
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import argparse
import json
import os
import time

import torch


def run(mode: str, steps: int, bs: int, seq: int, args) -> list:
    os.environ["CI_CE_FP8R"] = "1" if mode == "fp8r" else "0"
    from code_intelligence_amd.data.synthetic import synthetic_issue_tokens
    from code_intelligence_amd.models.awd_lstm import AWDLSTM
    from code_intelligence_amd.train.trainer import LMTrainer, TrainConfig

    torch.manual_seed(1234)
    model = AWDLSTM(vocab_sz=args.vocab, emb_sz=args.emb, n_hid=args.hid,
                    n_layers=args.layers).to("cuda", torch.bfloat16)
    trainer = LMTrainer(model, TrainConfig())
    model.train()
    model.reset(bs)
    docs = synthetic_issue_tokens(64, args.vocab, seed=7, mean_len=300)
    stream = torch.tensor([t for d in docs for t in d], dtype=torch.int64)
    need = bs * (seq + 1) * steps
    stream = stream.repeat(need // stream.numel() + 1)[:need] \
        .view(-1, seq + 1).to("cuda")
    losses = []
    t0 = time.perf_counter()
    for i in range(steps):
        s = (i * bs) % (stream.shape[0] - bs + 1)
        win = stream[s: s + bs]
        losses.append(trainer.train_step(win[:, :-1], win[:, 1:], 1e-3))
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(json.dumps({"mode": mode, "ms_per_step": round(dt / steps * 1e3, 1),
                      "losses": [round(l, 4) for l in losses]}))
    return losses


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--bs", type=int, default=256)
    p.add_argument("--seq", type=int, default=256)
    p.add_argument("--vocab", type=int, default=60000)
    p.add_argument("--emb", type=int, default=800)
    p.add_argument("--hid", type=int, default=2400)
    p.add_argument("--layers", type=int, default=4)
    args = p.parse_args()
    lb = run("bf16", args.steps, args.bs, args.seq, args)
    lf = run("fp8r", args.steps, args.bs, args.seq, args)
    tail_b = sum(lb[-5:]) / 5
    tail_f = sum(lf[-5:]) / 5
    print(json.dumps({"tail_loss_bf16": round(tail_b, 4),
                      "tail_loss_fp8r": round(tail_f, 4),
                      "rel_diff": round(abs(tail_f - tail_b) / tail_b, 5)}))


if __name__ == "__main__":
    main()
