"""Golden-tokenization dumper — run in a REFERENCE environment (fastai
1.0.5x + spacy + mdparse installed; not this image) to record the exact
token streams the reference produces, so the air-gapped framework can be
checked token-for-token (NOTES.md gap 2).

  python scripts/dump_reference_tokens.py texts.json golden_tokens.json

texts.json: ["raw text", ...]  (defaults to the built-in probe set)
Output: [{"text": ..., "tokens": [...]}, ...] — drop the file at
tests/data/reference_tokens.json and tests/test_text_data.py's
test_reference_golden_tokens (skipped when absent) enforces parity.
"""
import json
import sys

PROBE_TEXTS = [
    "xxxfldtitle Crash in train.py xxxfldbody It FAILS with a Traceback:\n"
    "```python\nValueError: bad\n```\nSee https://github.com/a/b#issue-1",
    "Add support for `--qrnn` flag (PLEASE!!)  multi   space\ttab",
    "HTML entities &amp; escapes <code>x&lt;1</code> and CAPS WORDS",
    "repeated letters loooooool and 1234 numbers v1.2.3",
    "unicode café — emdash … ellipsis",
]


def main():
    texts = PROBE_TEXTS
    if len(sys.argv) > 1 and sys.argv[1] != "-":
        texts = json.loads(open(sys.argv[1]).read())
    out_path = sys.argv[2] if len(sys.argv) > 2 else "golden_tokens.json"
    try:
        from fastai.text import Tokenizer as FTokenizer  # noqa
        from fastai.text.transform import defaults  # noqa
    except ImportError:
        raise SystemExit("run this in the reference environment "
                         "(fastai 1.0.5x + spacy)")
    tok = FTokenizer()
    token_lists = tok.process_all(texts)
    json.dump([{"text": t, "tokens": toks}
               for t, toks in zip(texts, token_lists)],
              open(out_path, "w"), indent=1)
    print(f"wrote {len(texts)} golden tokenizations to {out_path}")


if __name__ == "__main__":
    main()
