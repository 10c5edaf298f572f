"""Serve-path benchmark: issue-embeddings/sec (BASELINE.json config 4).

Measures the InferenceWrapper batched bulk path (df_to_embedding) and the
single-request path, with and without hipGraph capture, on synthetic
(title, body) issues at the deployed model shape. Prints one JSON line
per configuration.

Run on an MI355X box: python scripts/serve_bench.py [--n 2000]
"""
import sys
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))


import argparse
import json
import time

import torch

from code_intelligence_amd.data.synthetic import synthetic_issue_texts
from code_intelligence_amd.engine.inference import InferenceWrapper
from code_intelligence_amd.models.awd_lstm import AWDLSTM
from code_intelligence_amd.text.tokenizer import Vocab, defaults_specials


def build_wrapper(use_graphs: bool, emb=800, hid=2400, layers=4, vocab=60000,
                  qrnn=False):
    torch.manual_seed(0)
    words = [f"w{i}" for i in range(vocab - len(defaults_specials))]
    v = Vocab(defaults_specials + words)
    model = AWDLSTM(vocab_sz=len(v), emb_sz=emb, n_hid=hid, n_layers=layers,
                    qrnn=qrnn)
    return InferenceWrapper(encoder=model.encoder, vocab=v,
                            use_graphs=use_graphs)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--n", type=int, default=2000)
    p.add_argument("--bs", type=int, default=200)  # reference: "200 stable"
    p.add_argument("--single", type=int, default=50)
    p.add_argument("--qrnn", type=lambda v: v.lower() == "true", default=False)
    args = p.parse_args()
    issues = synthetic_issue_texts(args.n, seed=3)

    for use_graphs in (False, True):
        if use_graphs and (not torch.cuda.is_available() or args.qrnn):
            continue  # graph capture unsupported for QRNN (see inference.py)
        w = build_wrapper(use_graphs, qrnn=args.qrnn)
        texts = [w.process_dict(d)["text"] for d in issues]
        # warmup (captures graphs for the bucket shapes)
        w.texts_to_embedding(texts[:256], bs=args.bs)
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        out = w.texts_to_embedding(texts, bs=args.bs)
        if torch.cuda.is_available():
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        print(json.dumps({
            "metric": "issue-embeddings/sec (bulk)",
            "value": round(len(texts) / dt, 1),
            "unit": "embeddings/s", "n": len(texts), "bs": args.bs,
            "hipgraph": use_graphs, "qrnn": args.qrnn,
            "dim": int(out.shape[1]),
            "ms_total": round(dt * 1e3, 1)}))

        # single-request latency (flask /text path without HTTP)
        lat = []
        for d in issues[: args.single]:
            t0 = time.perf_counter()
            w.get_pooled_features(w.process_dict(d)["text"])
            if torch.cuda.is_available():
                torch.cuda.synchronize()
            lat.append(time.perf_counter() - t0)
        lat.sort()
        print(json.dumps({
            "metric": "single-request latency",
            "p50_ms": round(lat[len(lat) // 2] * 1e3, 2),
            "p95_ms": round(lat[int(len(lat) * 0.95)] * 1e3, 2),
            "hipgraph": use_graphs}))


if __name__ == "__main__":
    main()
