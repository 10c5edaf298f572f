"""Property-based tests (hypothesis) — randomized invariants on the pure
components: scan recurrence, pooling, tokenizer parity, URL/spec codecs."""
import string

import torch
from hypothesis import given, settings, strategies as st

from code_intelligence_amd.gh.util import (build_issue_spec, build_issue_url,
                                           parse_issue_spec, parse_issue_url)
from code_intelligence_amd.ops.pool import _cpu_concat_pool
from code_intelligence_amd.ops.qrnn import _fo_pool_torch

NAME = st.text(alphabet=string.ascii_lowercase + string.digits + "-_.",
               min_size=1, max_size=20).filter(
    lambda s: not s.startswith(".") and "/" not in s and "#" not in s)


@settings(max_examples=50, deadline=None)
@given(NAME, NAME, st.integers(min_value=1, max_value=10**8))
def test_issue_spec_url_roundtrip(owner, repo, num):
    assert parse_issue_spec(build_issue_spec(owner, repo, num)) == (owner, repo, num)
    assert parse_issue_url(build_issue_url(owner, repo, num)) == (owner, repo, num)


@settings(max_examples=25, deadline=None)
@given(st.integers(1, 4), st.integers(1, 9), st.integers(1, 8),
       st.integers(0, 2**31 - 1))
def test_fo_pool_invariants(B, T, H, seed):
    g = torch.Generator().manual_seed(seed)
    gates = torch.randn(B, T, 3 * H, generator=g, dtype=torch.float64)
    c0 = torch.randn(B, H, generator=g, dtype=torch.float64)
    h, cT = _fo_pool_torch(gates, c0)
    assert h.shape == (B, T, H) and cT.shape == (B, H)
    # c_t is a convex combination of c_{t-1} and z_t in (-1,1) =>
    # elementwise |c_T| <= max(|c0|, 1)
    bound = torch.maximum(c0.abs(), torch.ones_like(c0))
    assert (cT.abs() <= bound + 1e-9).all()
    # output gate in (0,1): |h| <= |c|
    z = torch.tanh(gates[..., :H])
    assert (h.abs() <= 1 + c0.abs().amax() + 1e-9).all()
    # T=1 closed form
    if T == 1:
        f = torch.sigmoid(gates[:, 0, H:2 * H])
        o = torch.sigmoid(gates[:, 0, 2 * H:])
        c1 = f * c0 + (1 - f) * z[:, 0]
        assert torch.allclose(h[:, 0], o * c1, atol=1e-12)


@settings(max_examples=25, deadline=None)
@given(st.integers(1, 3), st.integers(1, 7), st.integers(1, 5),
       st.integers(0, 2**31 - 1))
def test_concat_pool_matches_manual(B, T, H, seed):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(B, T, H, generator=g)
    lens = torch.randint(1, T + 1, (B,), generator=g, dtype=torch.int32)
    out = _cpu_concat_pool(x, lens)
    for b in range(B):
        n = int(lens[b])
        ref = torch.cat([x[b, :n].mean(0), x[b, :n].amax(0), x[b, n - 1]])
        assert torch.allclose(out[b], ref, atol=1e-6)


@settings(max_examples=40, deadline=None)
@given(st.text(alphabet=string.printable, max_size=120))
def test_tokenizer_native_matches_python_ascii(s):
    from code_intelligence_amd.text.tokenizer import Tokenizer
    py = Tokenizer(native=False)
    nat = Tokenizer(native=True)
    assert nat.process_text(s) == py.process_text(s)


@settings(max_examples=30, deadline=None)
@given(st.integers(1, 5), st.integers(1, 7),
       st.lists(st.integers(2, 25), min_size=1, max_size=12),
       st.integers(0, 1000))
def test_lm_loader_invariants(bs, bptt, doc_lens, seed):
    from code_intelligence_amd.data.lm_loader import LMStreamLoader
    docs = [[(seed + i * 31 + j) % 100 for j in range(n)]
            for i, n in enumerate(doc_lens)]
    ld = LMStreamLoader(docs, bs=bs, bptt=bptt, seed=seed)
    batches = list(ld)
    assert len(batches) == len(ld)
    for x, y in batches:
        assert x.shape == y.shape == (bs, bptt)
        # y is x shifted by one within each parallel stream
    if batches:
        xs = torch.cat([x for x, _ in batches], dim=1)
        ys = torch.cat([y for _, y in batches], dim=1)
        assert torch.equal(xs[:, 1:], ys[:, :-1])


@settings(max_examples=60, deadline=None)
@given(st.booleans(), st.booleans(), st.booleans(), st.booleans(),
       st.booleans(), st.booleans())
def test_triage_rules_property(has_kind, has_prio, has_area, p0, has_proj,
                               closed):
    """needs_triage iff open and (missing kind/priority/area, or P0 without
    a project event) — for every combination of label events."""
    from code_intelligence_amd.triage.triage import TriageInfo
    events = []
    t = "2024-01-0{}T00:00:00Z"
    if has_kind:
        events.append({"__typename": "LabeledEvent", "createdAt": t.format(1),
                       "label": {"name": "kind/bug"}})
    prio_name = "priority/p0" if p0 else "priority/p2"
    if has_prio:
        events.append({"__typename": "LabeledEvent", "createdAt": t.format(2),
                       "label": {"name": prio_name}})
    if has_area:
        events.append({"__typename": "LabeledEvent", "createdAt": t.format(3),
                       "label": {"name": "area/ops"}})
    if has_proj:
        events.append({"__typename": "AddedToProjectEvent",
                       "createdAt": t.format(4)})
    issue = {"id": "x", "number": 1, "state": "CLOSED" if closed else "OPEN",
             "closedAt": "2024-02-01T00:00:00Z" if closed else None,
             "labels": {"edges": ([{"node": {"name": prio_name}}]
                                  if has_prio else [])},
             "projectCards": {"edges": []},
             "timelineItems": {"edges": [{"node": e} for e in events]}}
    info = TriageInfo.from_issue(issue)
    expected = (not closed) and (
        not (has_kind and has_prio and has_area)
        or (has_prio and p0 and not has_proj))
    assert info.needs_triage == expected
    if not info.needs_triage:
        assert info.triaged_at is not None


@settings(max_examples=40, deadline=None)
@given(st.lists(st.sampled_from(["alpha", "beta", "gamma", "delta"]),
                min_size=1, max_size=30))
def test_vocab_roundtrip_property(tokens):
    from code_intelligence_amd.text.tokenizer import Vocab, defaults_specials
    v = Vocab(defaults_specials + ["alpha", "beta", "gamma", "delta"])
    ids = v.numericalize(tokens)
    assert v.textify(ids).split() == tokens  # in-vocab roundtrip is exact
