"""Data pipeline: issue archive -> tokenized LM corpus
(reference: Issue_Embeddings notebooks 01_AcquireData + 02_fastai_DataBunch:
GHArchive -> mdparse pre-rules -> tokenize -> vocab -> databunch).

Offline equivalent: reads JSONL archive shards (gh/bigquery.py format, or
'synthetic:N' to generate), builds 'xxxfldtitle .. xxxfldbody ..' docs
(process_dict), tokenizes with the framework rules, builds a 60k vocab
(min_freq 2), and writes the COMPACT corpus docs.pt (flat int32 +
offsets — data/lm_loader.py) + vocab.json — the layout the train CLI
consumes.

Parallelism: thread-chunked ``Tokenizer.process_all`` — the C++ core
releases the GIL, so threads overlap its work while the Python pre-rules
serialize; measured ~8x the old per-text process-pool (pickling-bound).

``--streaming`` runs TWO passes over the archive in bounded memory
(pass 1 counts the vocab, pass 2 numericalizes straight into the flat
int32 buffer): RAM stays O(chunk + vocab + flat tokens) instead of
holding every raw text and token list at once — the mode for the
reference-scale 16.7M-issue corpus.

  python scripts/prepare_data.py --archive /path/or/synthetic:5000 \
      --out data_dir [--max_vocab 60000] [--workers 8] [--streaming]
"""
import sys
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))


import argparse
import json
import multiprocessing as mp
from collections import Counter
from concurrent.futures import ThreadPoolExecutor
from pathlib import Path

import torch

from code_intelligence_amd.data.lm_loader import docs_to_compact
from code_intelligence_amd.text.tokenizer import (Tokenizer, Vocab,
                                                  process_dict)


def iter_archive_texts(archive: str, chunk: int = 50000):
    """Yield lists of <= chunk document texts, deterministically ordered
    (both streaming passes see the same sequence)."""
    buf = []
    if archive.startswith("synthetic"):
        from code_intelligence_amd.data.synthetic import synthetic_issue_texts
        n = int(archive.split(":")[1]) if ":" in archive else 5000
        for d in synthetic_issue_texts(n):
            buf.append(process_dict(d)["text"])
            if len(buf) >= chunk:
                yield buf
                buf = []
    else:
        for f in sorted(Path(archive).glob("*.jsonl")):
            for line in open(f):
                if line.strip():
                    ev = json.loads(line)
                    buf.append(process_dict(
                        {"title": ev.get("title", ""),
                         "body": ev.get("body", "")})["text"])
                    if len(buf) >= chunk:
                        yield buf
                        buf = []
    if buf:
        yield buf


def tokenize_chunk(tok: Tokenizer, texts: list, workers: int):
    if workers > 1 and len(texts) > workers:
        shards = [texts[i::workers] for i in range(workers)]
        with ThreadPoolExecutor(workers) as ex:
            parts = list(ex.map(tok.process_all, shards))
        out = [None] * len(texts)
        for w, part in enumerate(parts):
            for j, toks in enumerate(part):
                out[w + j * workers] = toks
        return out
    return tok.process_all(texts)


def run_streaming(args, out: Path) -> tuple:
    tok = Tokenizer()
    counts: Counter = Counter()
    n_docs = 0
    for texts in iter_archive_texts(args.archive, args.chunk):
        for toks in tokenize_chunk(tok, texts, args.workers):
            counts.update(toks)
            n_docs += 1
    vocab = Vocab.from_counter(counts, max_vocab=args.max_vocab,
                               min_freq=args.min_freq)
    lengths = []
    flats = []
    for texts in iter_archive_texts(args.archive, args.chunk):
        for toks in tokenize_chunk(tok, texts, args.workers):
            ids = vocab.numericalize(toks)
            lengths.append(len(ids))
            flats.append(torch.tensor(ids, dtype=torch.int32))
    flat = torch.cat(flats) if flats else torch.empty(0, dtype=torch.int32)
    offsets = torch.zeros(len(lengths) + 1, dtype=torch.int64)
    torch.cumsum(torch.tensor(lengths, dtype=torch.int64), 0, out=offsets[1:])
    torch.save({"flat": flat, "offsets": offsets}, out / "docs.pt")
    return vocab, len(lengths), int(flat.numel())


def run_in_memory(args, out: Path) -> tuple:
    texts = [t for chunk in iter_archive_texts(args.archive, args.chunk)
             for t in chunk]
    print(f"{len(texts)} documents")
    tok = Tokenizer()
    token_docs = tokenize_chunk(tok, texts, args.workers)
    vocab = Vocab.create(token_docs, max_vocab=args.max_vocab,
                         min_freq=args.min_freq)
    docs = [vocab.numericalize(t) for t in token_docs]
    torch.save(docs_to_compact(docs), out / "docs.pt")
    return vocab, len(docs), sum(len(d) for d in docs)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--archive", required=True)
    p.add_argument("--out", required=True)
    p.add_argument("--max_vocab", type=int, default=60000)
    p.add_argument("--min_freq", type=int, default=2)
    p.add_argument("--workers", type=int, default=mp.cpu_count())
    p.add_argument("--streaming", action="store_true",
                   help="two-pass bounded-memory mode (reference-scale corpora)")
    p.add_argument("--chunk", type=int, default=50000,
                   help="documents per streaming chunk")
    args = p.parse_args()

    out = Path(args.out)
    out.mkdir(parents=True, exist_ok=True)
    if args.streaming:
        vocab, n_docs, n_tok = run_streaming(args, out)
    else:
        vocab, n_docs, n_tok = run_in_memory(args, out)
    vocab.save(out / "vocab.json")
    # corpus stats (the counts the reference's 01 notebook reports)
    (out / "corpus_stats.json").write_text(json.dumps({
        "n_docs": n_docs, "n_tokens": n_tok, "vocab_size": len(vocab),
        "mean_doc_tokens": round(n_tok / max(1, n_docs), 1),
        "max_vocab": args.max_vocab, "min_freq": args.min_freq,
        "streaming": bool(args.streaming)}))
    print(f"vocab {len(vocab)}; {n_tok} tokens -> {out}")


if __name__ == "__main__":
    main()
