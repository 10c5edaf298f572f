"""Tokenization + numericalization compatible with the reference pipeline.

The reference tokenizes with fastai 1.0 rules plus mdparse markdown
pre-rules (py/code_intelligence/inference.py:46-53 ``parse``: it composes
``transform_pre_rules + default_pre_rules``) and builds
``'xxxfldtitle <title> xxxfldbody <body>'`` documents
(inference.py:95-126 ``process_dict``). This module re-creates:

* the fastai special tokens and pre/post rules (fix_html, replace_rep,
  replace_wrep, spec_add_spaces, rm_useless_spaces; replace_all_caps,
  deal_caps),
* a markdown normalizer standing in for mdparse (code blocks/inline
  code/links/images collapsed to stable tokens — same role, own
  implementation; divergence documented here rather than copied),
* a regex word tokenizer standing in for spacy (the heavyweight spacy dep
  is not part of the MI355X runtime; tokenization is CPU-side and
  vocabulary-compatible as long as train/serve use the same tokenizer).

Numericalization: ``Vocab`` with fastai's itos/stoi contract
(max_vocab, min_freq — 02_fastai_DataBunch.ipynb uses 60k vocab).
"""
from __future__ import annotations

import html
import re
from collections import Counter
from typing import Dict, Iterable, List, Optional

UNK, PAD, BOS, EOS, FLD = "xxunk", "xxpad", "xxbos", "xxeos", "xxfld"
TK_MAJ, TK_UP, TK_REP, TK_WREP = "xxmaj", "xxup", "xxrep", "xxwrep"

# fastai text.transform.defaults.text_spec_tok order — index of PAD must stay
# 1 (pad_token=1 throughout the reference model configs).
defaults_specials = [UNK, PAD, BOS, EOS, FLD, TK_MAJ, TK_UP, TK_REP, TK_WREP]

_re_rep = re.compile(r"(\S)(\1{3,})")
_re_wrep = re.compile(r"(?:\s|^)(\w+)((?:\s+\1){3,})(\s|$)")
_re_spec = re.compile(r"([/#\\])")
_re_space = re.compile(r" {2,}")


def fix_html(x: str) -> str:
    """fastai fix_html semantics: un-escape HTML artifacts from scraped text."""
    x = (x.replace("#39;", "'").replace("amp;", "&").replace("#146;", "'")
         .replace("nbsp;", " ").replace("#36;", "$").replace("\\n", "\n")
         .replace("quot;", "'").replace("<br />", "\n").replace('\\"', '"')
         .replace("<unk>", UNK).replace(" @.@ ", ".").replace(" @-@ ", "-")
         .replace(" @,@ ", ",").replace("\\", " \\ "))
    return html.unescape(x)


def replace_rep(x: str) -> str:
    """ccccc -> TK_REP 5 c  (character repeated 4+ times)."""
    def _repl(m):
        c, cc = m.groups()
        return f" {TK_REP} {len(cc) + 1} {c} "
    return _re_rep.sub(_repl, x)


def replace_wrep(x: str) -> str:
    """word word word word -> TK_WREP 4 word (word repeated 4+ times)."""
    def _repl(m):
        w, ws, end = m.groups()
        return f" {TK_WREP} {len(ws.split()) + 1} {w} {end}"
    return _re_wrep.sub(_repl, x)


def spec_add_spaces(x: str) -> str:
    return _re_spec.sub(r" \1 ", x)


def rm_useless_spaces(x: str) -> str:
    return _re_space.sub(" ", x)


_re_codeblock = re.compile(r"```.*?```|~~~.*?~~~", re.S)
_re_inline_code = re.compile(r"`[^`\n]+`")
_re_image = re.compile(r"!\[[^\]]*\]\([^\)]*\)")
_re_link = re.compile(r"\[([^\]]*)\]\([^\)]*\)")
_re_url = re.compile(r"https?://\S+")
_re_header = re.compile(r"^#{1,6}\s*", re.M)
_re_quote = re.compile(r"^>\s*", re.M)


def markdown_rules(x: str) -> str:
    """mdparse-equivalent markdown normalization (own implementation):
    code blocks -> xxcdb token, inline code -> xxcd, images dropped,
    links -> their text, bare URLs -> xxurl, header/quote markers stripped."""
    x = _re_codeblock.sub(" xxcdb ", x)
    x = _re_inline_code.sub(" xxcd ", x)
    x = _re_image.sub(" ", x)
    x = _re_link.sub(r" \1 ", x)
    x = _re_url.sub(" xxurl ", x)
    x = _re_header.sub(" ", x)
    x = _re_quote.sub(" ", x)
    return x


default_pre_rules = [fix_html, replace_rep, replace_wrep, spec_add_spaces,
                     rm_useless_spaces]
markdown_pre_rules = [markdown_rules] + default_pre_rules

_re_word = re.compile(r"\w+|[^\w\s]")


def replace_all_caps(toks: List[str]) -> List[str]:
    out: List[str] = []
    for t in toks:
        if len(t) > 1 and t.isupper() and t.isalpha():
            out.append(TK_UP)
            out.append(t.lower())
        else:
            out.append(t)
    return out


def deal_caps(toks: List[str]) -> List[str]:
    out: List[str] = []
    for t in toks:
        if len(t) > 1 and t[0].isupper() and t[1:].islower() and t.isalpha():
            out.append(TK_MAJ)
            out.append(t.lower())
        else:
            out.append(t)
    return out


default_post_rules = [replace_all_caps, deal_caps]


class Tokenizer:
    """pre_rules (str->str) -> regex word split -> post_rules (tokens->tokens).

    With ``native=True`` (default) and default rules, pure-ASCII documents
    run through the C++ tokenizer core (ops/csrc/tokenizer.cpp — token-exact
    for replace_rep/replace_wrep/split/caps; fix_html + markdown
    normalization stay in Python); non-ASCII or custom-rule inputs use the
    Python path, so results are identical either way."""

    def __init__(self, pre_rules=None, post_rules=None, markdown: bool = True,
                 native: bool = True):
        self.pre_rules = pre_rules if pre_rules is not None else (
            markdown_pre_rules if markdown else default_pre_rules)
        self.post_rules = post_rules if post_rules is not None else default_post_rules
        self._default_rules = pre_rules is None and post_rules is None
        self._native = None
        if native and self._default_rules:
            try:
                from ..ops import extension as _ext
                lib = _ext.load(required=False)
                if lib is not None and hasattr(lib, "tokenize_core"):
                    self._native = lib
            except Exception:
                self._native = None
        # string-level rules that precede the native core (everything except
        # the token-level-equivalent rep/wrep/space rules it implements)
        self._native_pre = ([markdown_rules] if markdown else []) + [fix_html]

    def process_text(self, text: str) -> List[str]:
        if self._native is not None:
            pre = text
            for rule in self._native_pre:
                pre = rule(pre)
            if pre.isascii():
                return self._native.tokenize_core(pre)
        for rule in self.pre_rules:
            text = rule(text)
        toks = _re_word.findall(text)
        for rule in self.post_rules:
            toks = rule(toks)
        return toks

    def process_all(self, texts: Iterable[str]) -> List[List[str]]:
        if self._native is not None:
            pres, idx_native, out = [], [], {}
            texts = list(texts)
            for i, t in enumerate(texts):
                pre = t
                for rule in self._native_pre:
                    pre = rule(pre)
                if pre.isascii():
                    pres.append(pre)
                    idx_native.append(i)
            if pres:
                for i, toks in zip(idx_native,
                                   self._native.tokenize_core_batch(pres)):
                    out[i] = toks
            return [out[i] if i in out else self.process_text(texts[i])
                    for i in range(len(texts))]
        return [self.process_text(t) for t in texts]


class Vocab:
    """itos/stoi with fastai contract; index 0 = xxunk, 1 = xxpad."""

    def __init__(self, itos: List[str]):
        self.itos = list(itos)
        self.stoi: Dict[str, int] = {s: i for i, s in enumerate(self.itos)}

    def __len__(self) -> int:
        return len(self.itos)

    def numericalize(self, toks: List[str]) -> List[int]:
        unk = self.stoi.get(UNK, 0)
        return [self.stoi.get(t, unk) for t in toks]

    def textify(self, ids: Iterable[int], sep: str = " ") -> str:
        return sep.join(self.itos[i] for i in ids)

    @classmethod
    def create(cls, tokens: Iterable[List[str]], max_vocab: int = 60000,
               min_freq: int = 2) -> "Vocab":
        return cls.from_counter(Counter(t for doc in tokens for t in doc),
                                max_vocab=max_vocab, min_freq=min_freq)

    @classmethod
    def from_counter(cls, counts: Counter, max_vocab: int = 60000,
                     min_freq: int = 2) -> "Vocab":
        """Build from pre-accumulated counts (streaming prepare_data)."""
        itos = [s for s, c in counts.most_common(max_vocab)
                if c >= min_freq and s not in defaults_specials]
        itos = defaults_specials + itos
        return cls(itos[: max_vocab + len(defaults_specials)])

    def save(self, path) -> None:
        import json
        with open(path, "w") as f:
            json.dump(self.itos, f)

    @classmethod
    def load(cls, path) -> "Vocab":
        import json
        with open(path) as f:
            return cls(json.load(f))


def process_dict(data: dict, tokenizer: Optional[Tokenizer] = None) -> dict:
    """Reference-compatible document builder (inference.py:95-126):
    {'title': ..., 'body': ...} -> {'text': 'xxxfldtitle <t> xxxfldbody <b>'};
    any failure yields the literal 'xxxUnk' (reference behavior)."""
    tok = tokenizer or Tokenizer()
    try:
        title = str(data["title"]).strip()
        body = str(data["body"]).strip()
        text = f"xxxfldtitle {title} xxxfldbody {body}"
        # run string-level rules only (tokenization happens at numericalize)
        for rule in tok.pre_rules:
            text = rule(text)
        if not text.strip():
            text = "xxxUnk"
    except Exception:
        text = "xxxUnk"
    return {"text": text}
