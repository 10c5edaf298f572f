"""Data pipeline: issue archive -> tokenized LM corpus
(reference: Issue_Embeddings notebooks 01_AcquireData + 02_fastai_DataBunch:
GHArchive -> mdparse pre-rules -> tokenize -> vocab -> databunch).

Offline equivalent: reads JSONL archive shards (gh/bigquery.py format, or
'synthetic:N' to generate), builds 'xxxfldtitle .. xxxfldbody ..' docs
(process_dict), tokenizes with the framework rules in a process pool,
builds a 60k vocab (min_freq 2), and writes docs.pt + vocab.json — the
layout the train CLI consumes.

Parallelism: thread-chunked ``Tokenizer.process_all`` — the C++ core
releases the GIL, so threads overlap its work while the Python pre-rules
serialize; measured ~10x the old per-text process-pool (pickling-bound).

  python scripts/prepare_data.py --archive /path/or/synthetic:5000 \
      --out data_dir [--max_vocab 60000] [--workers 8]
"""
import sys
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))


import argparse
import json
import multiprocessing as mp
from concurrent.futures import ThreadPoolExecutor
from pathlib import Path

import torch

from code_intelligence_amd.data.lm_loader import docs_to_compact
from code_intelligence_amd.text.tokenizer import (Tokenizer, Vocab,
                                                  process_dict)

def load_archive_docs(archive: str) -> list[str]:
    if archive.startswith("synthetic"):
        from code_intelligence_amd.data.synthetic import synthetic_issue_texts
        n = int(archive.split(":")[1]) if ":" in archive else 5000
        raw = synthetic_issue_texts(n)
    else:
        raw = []
        for f in sorted(Path(archive).glob("*.jsonl")):
            for line in open(f):
                if line.strip():
                    ev = json.loads(line)
                    raw.append({"title": ev.get("title", ""),
                                "body": ev.get("body", "")})
    return [process_dict(d)["text"] for d in raw]


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--archive", required=True)
    p.add_argument("--out", required=True)
    p.add_argument("--max_vocab", type=int, default=60000)
    p.add_argument("--min_freq", type=int, default=2)
    p.add_argument("--workers", type=int, default=mp.cpu_count())
    args = p.parse_args()

    texts = load_archive_docs(args.archive)
    print(f"{len(texts)} documents")
    tok = Tokenizer()
    if args.workers > 1 and len(texts) > args.workers:
        chunks = [texts[i::args.workers] for i in range(args.workers)]
        with ThreadPoolExecutor(args.workers) as ex:
            parts = list(ex.map(tok.process_all, chunks))
        # un-interleave back to original order
        token_docs = [None] * len(texts)
        for w, part in enumerate(parts):
            for j, toks in enumerate(part):
                token_docs[w + j * args.workers] = toks
    else:
        token_docs = tok.process_all(texts)
    vocab = Vocab.create(token_docs, max_vocab=args.max_vocab,
                         min_freq=args.min_freq)
    docs = [vocab.numericalize(t) for t in token_docs]
    out = Path(args.out)
    out.mkdir(parents=True, exist_ok=True)
    # compact corpus: flat int32 + offsets — 8x less RAM than list-of-lists
    # at the reference's 16.7M-issue scale; LMStreamLoader consumes it
    # directly (data/lm_loader.py)
    torch.save(docs_to_compact(docs), out / "docs.pt")
    vocab.save(out / "vocab.json")
    n_tok = sum(len(d) for d in docs)
    # corpus stats (the counts the reference's 01 notebook reports)
    (out / "corpus_stats.json").write_text(json.dumps({
        "n_docs": len(docs), "n_tokens": n_tok, "vocab_size": len(vocab),
        "mean_doc_tokens": round(n_tok / max(1, len(docs)), 1),
        "max_vocab": args.max_vocab, "min_freq": args.min_freq}))
    print(f"vocab {len(vocab)}; {n_tok} tokens -> {out}")


if __name__ == "__main__":
    main()
