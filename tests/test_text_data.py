"""Tokenizer rules, vocab, loader-stream semantics (CPU)."""
import torch

from code_intelligence_amd.data.lm_loader import LMStreamLoader
from code_intelligence_amd.data.synthetic import synthetic_issue_tokens
from code_intelligence_amd.text.tokenizer import (Tokenizer, Vocab, PAD, UNK,
                                                  TK_MAJ, TK_REP, TK_UP,
                                                  defaults_specials,
                                                  process_dict, replace_rep,
                                                  replace_all_caps, deal_caps)


def test_special_token_indices():
    v = Vocab(defaults_specials + ["hello", "world"])
    assert v.stoi[UNK] == 0
    assert v.stoi[PAD] == 1  # pad_token=1 everywhere in the reference


def test_replace_rep():
    assert TK_REP in replace_rep("soooooo cool")
    assert replace_rep("soo cool") == "soo cool"  # <4 reps untouched


def test_caps_rules():
    assert replace_all_caps(["HELLO", "hi"]) == [TK_UP, "hello", "hi"]
    assert deal_caps(["Hello", "world"]) == [TK_MAJ, "hello", "world"]


def test_tokenizer_end_to_end():
    tok = Tokenizer()
    toks = tok.process_text("# Bug report\n`pip install` FAILS with ```trace```")
    assert "xxcd" in toks  # inline code collapsed
    assert "xxcdb" in toks  # code block collapsed
    assert "xxup" in toks  # FAILS -> xxup fails
    assert "fails" in toks


def test_vocab_create_and_numericalize():
    tok = Tokenizer()
    docs = [tok.process_text("the bug the bug the bug"),
            tok.process_text("a feature a feature a feature")]
    v = Vocab.create(docs, max_vocab=100, min_freq=2)
    ids = v.numericalize(["the", "zzzunknown"])
    assert ids[0] >= len(defaults_specials)
    assert ids[1] == v.stoi[UNK]


def test_process_dict_reference_format():
    d = process_dict({"title": "Crash on start", "body": "It **fails**."})
    assert d["text"].startswith("xxxfldtitle")
    assert "xxxfldbody" in d["text"]
    assert process_dict({"title": None, "body": None})["text"] is not None


def test_lm_loader_stream_continuity():
    docs = [[10, 11, 12, 13, 14, 15, 16, 17]] * 8
    dl = LMStreamLoader(docs, bs=2, bptt=4, bos_idx=2, shuffle=False)
    batches = list(dl)
    assert len(batches) == len(dl)
    x0, y0 = batches[0]
    assert x0.shape == (2, 4)
    # y is x shifted by one within the stream
    assert torch.equal(y0[:, :-1], x0[:, 1:])
    # next window continues exactly where the previous ended
    x1, y1 = batches[1]
    assert torch.equal(x1[:, 0], y0[:, -1])


def test_synthetic_tokens_in_range():
    docs = synthetic_issue_tokens(10, vocab_sz=1000, seed=0)
    flat = [t for d in docs for t in d]
    assert min(flat) >= 9 and max(flat) < 1000


def test_prepare_data_script_end_to_end(tmp_path):
    """scripts/prepare_data.py: archive -> docs.pt + vocab.json the train
    CLI consumes (single-process path)."""
    import subprocess
    import sys
    out = tmp_path / "corpus"
    r = subprocess.run(
        [sys.executable, "scripts/prepare_data.py", "--archive", "synthetic:40",
         "--out", str(out), "--max_vocab", "500", "--workers", "1"],
        capture_output=True, text=True, cwd=".")
    assert r.returncode == 0, r.stderr
    docs = torch.load(out / "docs.pt", weights_only=True)
    # compact corpus format: flat int32 + per-doc offsets
    assert docs["offsets"].numel() - 1 == 40
    from code_intelligence_amd.text.tokenizer import Vocab
    v = Vocab.load(out / "vocab.json")
    assert int(docs["flat"].min()) >= 0
    assert int(docs["flat"].max()) < len(v)
    # consumable by the train CLI loader + stream loader
    from code_intelligence_amd.train.train_cli import load_docs
    docs2, vsz = load_docs(str(out), 0)
    assert vsz == len(v)
    ld = LMStreamLoader(docs2, bs=2, bptt=8)
    x, y = next(iter(ld))
    assert x.shape == (2, 8) and x.dtype == torch.int64


def test_tokenizer_fuzz_no_crash():
    """Arbitrary unicode/markdown garbage never crashes the pipeline."""
    import random
    from code_intelligence_amd.text.tokenizer import Tokenizer, process_dict
    rng = random.Random(0)
    tok = Tokenizer()
    pool = "abc ABC 123 #/\\`*[]()!&;\n\té中�" + "s" * 10
    for _ in range(50):
        s = "".join(rng.choice(pool) for _ in range(rng.randint(0, 200)))
        toks = tok.process_text(s)
        assert isinstance(toks, list)
        d = process_dict({"title": s, "body": s})
        assert isinstance(d["text"], str)


def test_loader_reshuffles_between_epochs():
    docs = [[10 + i] * 6 for i in range(50)]
    dl = LMStreamLoader(docs, bs=2, bptt=5, shuffle=True, seed=1)
    e1 = torch.cat([x.flatten() for x, _ in dl])
    e2 = torch.cat([x.flatten() for x, _ in dl])
    assert not torch.equal(e1, e2)  # epoch-order reshuffle
    assert e1.numel() == e2.numel()  # same window count (tail drop is
    # order-dependent, like fastai's LMDataLoader)


def test_native_tokenizer_parity():
    """C++ tokenizer core is token-exact vs the Python rules on ASCII
    (non-ASCII inputs transparently fall back to Python)."""
    import random
    import string
    from code_intelligence_amd.text.tokenizer import Tokenizer
    nat = Tokenizer(native=True)
    py = Tokenizer(native=False)
    if nat._native is None:
        import pytest
        pytest.skip("native extension not built")
    rng = random.Random(1)
    pool = string.ascii_letters + string.digits + " \n\t.,!?#/\\()[]{}`*->:;\"'"
    for i in range(300):
        s = "".join(rng.choice(pool) for _ in range(rng.randint(0, 250)))
        if i % 4 == 0:
            s += " soooooo COOL Cool go go go go go !!!!!! "
        assert nat.process_text(s) == py.process_text(s), s[:120]
    # non-ASCII routes through Python: identical by construction
    s = "héllo wörld CRASH"
    assert nat.process_text(s) == py.process_text(s)
    # batch API equals per-doc API
    docs = ["a b C", "x " * 30, "sooo COOL"]
    assert nat.process_all(docs) == [nat.process_text(d) for d in docs]


def test_vocab_save_load_roundtrip(tmp_path):
    v = Vocab(defaults_specials + ["alpha", "beta", "gamma"])
    v.save(tmp_path / "v.json")
    v2 = Vocab.load(tmp_path / "v.json")
    assert v2.itos == v.itos
    assert v2.numericalize(["alpha", "nope", "gamma"]) == \
        v.numericalize(["alpha", "nope", "gamma"])
    # pad stays at index 1 (fastai contract the encoder relies on)
    assert v2.itos[1] == "xxpad"


def test_compact_corpus_equivalent_to_lists():
    from code_intelligence_amd.data.lm_loader import (docs_to_compact,
                                                      split_compact)
    docs = [[(i * 7 + j) % 90 + 9 for j in range(5 + i % 11)]
            for i in range(40)]
    compact = docs_to_compact(docs)
    a = LMStreamLoader(docs, bs=4, bptt=6, seed=3)
    b = LMStreamLoader(compact, bs=4, bptt=6, seed=3)
    assert len(a) == len(b)
    for (xa, ya), (xb, yb) in zip(a, b):
        assert torch.equal(xa, xb) and torch.equal(ya, yb)
    # split: first 10 docs / rest, content preserved
    first, rest = split_compact(compact, 10)
    assert first["offsets"].numel() - 1 == 10
    assert rest["offsets"].numel() - 1 == 30
    re_flat = torch.cat([first["flat"], rest["flat"]])
    assert torch.equal(re_flat, compact["flat"])
    assert int(rest["offsets"][0]) == 0


def test_prepare_data_streaming_matches_in_memory(tmp_path):
    """--streaming (two-pass bounded-memory) produces a byte-identical
    corpus + vocab to the in-memory path."""
    import subprocess
    import sys
    a, b = tmp_path / "mem", tmp_path / "str"
    for extra, out in ((["--workers", "2"], a),
                       (["--streaming", "--chunk", "16", "--workers", "2"], b)):
        r = subprocess.run(
            [sys.executable, "scripts/prepare_data.py", "--archive",
             "synthetic:60", "--out", str(out), "--max_vocab", "300"] + extra,
            capture_output=True, text=True, cwd=".")
        assert r.returncode == 0, r.stderr
    da = torch.load(a / "docs.pt", weights_only=True)
    db = torch.load(b / "docs.pt", weights_only=True)
    assert torch.equal(da["flat"], db["flat"])
    assert torch.equal(da["offsets"], db["offsets"])
    assert (a / "vocab.json").read_text() == (b / "vocab.json").read_text()


def test_reference_golden_tokens():
    """Token-for-token parity against goldens recorded in a REFERENCE
    environment (scripts/dump_reference_tokens.py). Skipped until a
    fixture is recorded — spacy/fastai are not in this image (NOTES.md
    gap 2); the harness is the deliverable here."""
    import json
    from pathlib import Path
    import pytest as _pytest
    fixture = Path(__file__).parent / "data" / "reference_tokens.json"
    if not fixture.exists():
        _pytest.skip("no recorded reference tokenization fixture "
                     "(record with scripts/dump_reference_tokens.py in a "
                     "fastai+spacy environment)")
    from code_intelligence_amd.text.tokenizer import Tokenizer
    tok = Tokenizer()
    mismatches = []
    for case in json.loads(fixture.read_text()):
        got = tok.process_text(case["text"])
        if got != case["tokens"]:
            mismatches.append((case["text"][:40], got[:8],
                               case["tokens"][:8]))
    assert not mismatches, mismatches
