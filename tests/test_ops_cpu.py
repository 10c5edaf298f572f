"""CPU reference-path tests for the op layer (pool, CE, AdamW, dropout)."""
import torch
import torch.nn.functional as F

from code_intelligence_amd.ops.adam import FusedAdamW
from code_intelligence_amd.ops.crossentropy import tied_decoder_ce
from code_intelligence_amd.ops.dropout import variational_dropout
from code_intelligence_amd.ops.pool import concat_pool


def test_concat_pool_reference_semantics():
    """Matches batch_seq_pool (inference.py:232-263): mean/max over the true
    length, 'last' = state at length-1 (not padded tail)."""
    torch.manual_seed(0)
    B, T, H = 3, 6, 4
    h = torch.randn(B, T, H)
    lengths = torch.tensor([6, 3, 1])
    out = concat_pool(h, lengths)
    assert out.shape == (B, 3 * H)
    for b, L in enumerate(lengths.tolist()):
        seg = h[b, :L]
        assert torch.allclose(out[b, :H], seg.mean(0), atol=1e-6)
        assert torch.allclose(out[b, H:2 * H], seg.max(0).values, atol=1e-6)
        assert torch.allclose(out[b, 2 * H:], h[b, L - 1], atol=1e-6)


def test_tied_ce_matches_functional():
    torch.manual_seed(0)
    N, H, V = 10, 8, 50
    h = torch.randn(N, H)
    w = torch.randn(V, H)
    b = torch.randn(V)
    t = torch.randint(0, V, (N,))
    loss = tied_decoder_ce(h, w, b, t)
    ref = F.cross_entropy(F.linear(h, w, b), t)
    assert torch.allclose(loss, ref, atol=1e-5)


def test_fused_adamw_matches_torch_adamw():
    torch.manual_seed(0)
    p1 = torch.nn.Parameter(torch.randn(40))
    p2 = torch.nn.Parameter(p1.detach().clone())
    g = torch.randn(40)
    o1 = FusedAdamW([p1], lr=1e-2, betas=(0.9, 0.99), weight_decay=0.01)
    o2 = torch.optim.AdamW([p2], lr=1e-2, betas=(0.9, 0.99), weight_decay=0.01)
    for _ in range(5):
        p1.grad = g.clone()
        p2.grad = g.clone()
        o1.step()
        o2.step()
    assert torch.allclose(p1, p2, atol=1e-6), (p1 - p2).abs().max()


def test_variational_dropout_locked_mask():
    torch.manual_seed(0)
    x = torch.ones(4, 10, 8)
    y = variational_dropout(x, 0.5, training=True)
    # same mask across timesteps: each (b, h) column is constant over T
    assert torch.equal(y[:, 0, :] != 0, y[:, 5, :] != 0)
    # scaling preserves expectation roughly
    assert abs(y.mean().item() - 1.0) < 0.2
    assert torch.equal(variational_dropout(x, 0.5, training=False), x)


def test_tied_decoder_accuracy():
    from code_intelligence_amd.ops.crossentropy import tied_decoder_accuracy
    torch.manual_seed(0)
    w = torch.eye(6)  # logits = h -> argmax = largest coordinate
    h = torch.zeros(4, 6)
    h[0, 2] = 1; h[1, 5] = 1; h[2, 0] = 1; h[3, 1] = 1
    t = torch.tensor([2, 5, 0, 3])  # 3 of 4 correct
    acc = tied_decoder_accuracy(h, w, None, t, chunk=3)
    assert float(acc) == 0.75
