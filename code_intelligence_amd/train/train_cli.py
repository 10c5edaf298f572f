"""LM pretrain CLI — same flag surface as the reference fire CLI
(/root/reference/Issue_Embeddings/train.py:41-120 ``LangModel``): defaults
emb_sz=800, n_layers=4, n_hid=2500, bs=104, bptt=67, lr=0.0013, wd=0.012,
one_cycle, cycle_len. ``--data_path`` accepts a directory with tokenized
docs (docs.pt: List[List[int]] + vocab.json) or 'synthetic[:N]' for the
offline synthetic corpus.

Usage: python -m code_intelligence_amd.train --data_path synthetic:2000 \
           --emb_sz 400 --n_hid 1150 --n_layers 3 --bs 32 --bptt 64
"""
from __future__ import annotations

import argparse
import json
from pathlib import Path

import torch

from ..data.lm_loader import LMStreamLoader
from ..data.synthetic import synthetic_issue_tokens
from ..models.awd_lstm import AWDLSTM
from .callbacks import (PeriodicCheckpoint, CSVLogger, EarlyStopping, JSONRunLogger,
                        ReduceLROnPlateau, SaveModel)
from .trainer import LMTrainer, TrainConfig


def build_argparser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--data_path", type=str, default="synthetic:2000")
    p.add_argument("--model_path", type=str, default="model_files")
    p.add_argument("--emb_sz", type=int, default=800)
    p.add_argument("--n_layers", type=int, default=4)
    p.add_argument("--n_hid", type=int, default=2500)
    p.add_argument("--vocab_sz", type=int, default=60000)
    p.add_argument("--bs", type=int, default=104)
    p.add_argument("--bptt", type=int, default=67)
    p.add_argument("--qrnn", type=lambda s: s.lower() == "true", default=False,
                   help="QRNN encoder instead of LSTM (reference train.py:43)")
    p.add_argument("--lr", type=float, default=0.0013)
    p.add_argument("--wd", type=float, default=0.012)
    p.add_argument("--one_cycle", type=lambda s: s.lower() != "false", default=True)
    p.add_argument("--cycle_len", type=int, default=1)
    p.add_argument("--epochs", type=int, default=1)
    p.add_argument("--dtype", type=str, default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--device", type=str, default=None)
    p.add_argument("--seed", type=int, default=42)
    p.add_argument("--resume", type=str, default=None,
                   help="checkpoint path (trainer.save_checkpoint) to resume "
                        "from: restores model+optimizer+epoch+RNG and "
                        "continues the same trajectory")
    p.add_argument("--checkpoint_every", type=int, default=0,
                   help="full-state checkpoint every N steps (mid-epoch "
                        "resume for long pretraining runs); 0 = off")
    p.add_argument("--save_checkpoint", type=str, default=None,
                   help="write a full resume checkpoint here after training")
    return p


def load_docs(data_path: str, vocab_sz: int):
    if data_path.startswith("synthetic"):
        n = int(data_path.split(":")[1]) if ":" in data_path else 2000
        docs = synthetic_issue_tokens(n, vocab_sz,
                                      markov="markov" in data_path)
        return docs, vocab_sz
    root = Path(data_path)
    try:
        # compact corpora memory-map straight from disk (no RAM copy of
        # the flat token tensor until windows touch it)
        docs = torch.load(root / "docs.pt", weights_only=True, mmap=True)
    except (RuntimeError, TypeError):
        docs = torch.load(root / "docs.pt", weights_only=True)
    # docs.pt is either the compact {flat, offsets} corpus (prepare_data)
    # or a legacy list-of-lists; both feed LMStreamLoader
    vocab = json.loads((root / "vocab.json").read_text())
    return docs, len(vocab)


def main(argv=None) -> dict:
    args = build_argparser().parse_args(argv)
    torch.manual_seed(args.seed)
    device = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    docs, vocab_sz = load_docs(args.data_path, args.vocab_sz)
    if isinstance(docs, dict):
        from ..data.lm_loader import split_compact
        n_docs = docs["offsets"].numel() - 1
        valid_docs, train_docs = split_compact(docs, max(1, n_docs // 10))
    else:
        n_valid = max(1, len(docs) // 10)
        train_docs, valid_docs = docs[n_valid:], docs[:n_valid]
    model = AWDLSTM(vocab_sz=vocab_sz, emb_sz=args.emb_sz, n_hid=args.n_hid,
                    n_layers=args.n_layers, qrnn=args.qrnn)
    dtype = torch.bfloat16 if (args.dtype == "bf16" and device != "cpu") else torch.float32
    model = model.to(device=device, dtype=dtype)
    dev = torch.device(device)
    train_loader = LMStreamLoader(train_docs, args.bs, args.bptt, device=dev)
    valid_loader = LMStreamLoader(valid_docs, args.bs, args.bptt, device=dev,
                                  shuffle=False)
    out = Path(args.model_path)
    cfg = TrainConfig(lr=args.lr, wd=args.wd, one_cycle=args.one_cycle,
                      cycle_len=args.cycle_len)
    trainer = LMTrainer(model, cfg, callbacks=[
        EarlyStopping(patience=2),
        SaveModel(out),
        ReduceLROnPlateau(patience=1),
        CSVLogger(out / "history.csv"),
        JSONRunLogger(out / "run.jsonl", config=vars(args)),
    ] + ([PeriodicCheckpoint(out / "checkpoints", args.checkpoint_every)]
         if args.checkpoint_every else []))
    if args.resume:
        trainer.load_checkpoint(args.resume, map_location=device)
    metrics = trainer.fit(train_loader, valid_loader,
                          epochs=args.epochs * max(args.cycle_len, 1))
    if args.save_checkpoint:
        trainer.save_checkpoint(args.save_checkpoint)
    print(json.dumps({"final": metrics}))
    return metrics


if __name__ == "__main__":
    main()
