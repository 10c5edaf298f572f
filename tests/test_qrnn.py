"""QRNN encoder tests (reference --qrnn flag, train.py:43 / lm_tune.py:43).

CPU numerics lock the torch fo-pool path; the HIP scan kernel is compared
against it in tests/test_gpu_kernels.py.
"""
import torch
import pytest

from code_intelligence_amd.models import AWDLSTM, WeightDroppedQRNN
from code_intelligence_amd.ops.qrnn import _fo_pool_torch, qrnn_forward


def test_fo_pool_explicit_recurrence():
    torch.manual_seed(0)
    B, T, H = 3, 7, 5
    gates = torch.randn(B, T, 3 * H, dtype=torch.float64)
    c0 = torch.randn(B, H, dtype=torch.float64)
    h, cT = _fo_pool_torch(gates, c0)
    # independent scalar recurrence
    z = torch.tanh(gates[..., :H])
    f = torch.sigmoid(gates[..., H:2 * H])
    o = torch.sigmoid(gates[..., 2 * H:])
    c = c0.clone()
    for t in range(T):
        c = f[:, t] * c + (1 - f[:, t]) * z[:, t]
        assert torch.allclose(h[:, t], o[:, t] * c, atol=1e-12)
    assert torch.allclose(cT, c, atol=1e-12)


def test_fo_pool_gradcheck():
    torch.manual_seed(1)
    B, T, H = 2, 4, 3
    gates = torch.randn(B, T, 3 * H, dtype=torch.float64, requires_grad=True)
    c0 = torch.randn(B, H, dtype=torch.float64, requires_grad=True)
    assert torch.autograd.gradcheck(
        lambda g, c: _fo_pool_torch(g, c)[0], (gates, c0))


def test_qrnn_window2_chaining():
    """Two BPTT windows must equal one long window (prev_x + c carry)."""
    torch.manual_seed(2)
    B, T, E, H = 2, 6, 8, 10
    layer = WeightDroppedQRNN(E, H, weight_p=0.0, window=2).eval()
    x = torch.randn(B, 2 * T, E)
    c0 = torch.zeros(B, H)
    full, (_, cT_full) = layer(x, (c0, c0))
    layer.reset()
    first, (_, c_mid) = layer(x[:, :T], (c0, c0))
    second, (_, cT_split) = layer(x[:, T:], (c_mid, c_mid))
    assert torch.allclose(torch.cat([first, second], dim=1), full, atol=1e-6)
    assert torch.allclose(cT_split, cT_full, atol=1e-6)


def test_qrnn_model_forward_backward_and_reset():
    torch.manual_seed(3)
    model = AWDLSTM(vocab_sz=50, emb_sz=12, n_hid=16, n_layers=3, qrnn=True)
    x = torch.randint(0, 50, (4, 9))
    logits, raw, out = model(x)
    assert logits.shape == (4, 9, 50)
    assert raw[0].shape == (4, 9, 16) and raw[-1].shape == (4, 9, 12)
    logits.sum().backward()
    assert model.encoder.rnns[0].weight_raw.grad is not None
    # window-2 layer saved its prev_x; reset clears it
    assert model.encoder.rnns[0].prev_x is not None
    model.reset()
    assert model.encoder.rnns[0].prev_x is None


def test_qrnn_memorizes_tiny_sequence():
    torch.manual_seed(4)
    model = AWDLSTM(vocab_sz=20, emb_sz=16, n_hid=24, n_layers=2, qrnn=True,
                    output_p=0, hidden_p=0, input_p=0, embed_p=0, weight_p=0)
    seq = torch.tensor([[5, 9, 3, 14, 7, 2, 11, 5, 9, 3, 14, 7, 2, 11]])
    opt = torch.optim.Adam(model.parameters(), lr=5e-3)
    x, y = seq[:, :-1], seq[:, 1:]
    first = None
    for i in range(150):
        model.reset()
        logits, _, _ = model(x)
        loss = torch.nn.functional.cross_entropy(
            logits.reshape(-1, 20), y.reshape(-1))
        if first is None:
            first = loss.item()
        opt.zero_grad()
        loss.backward()
        opt.step()
    assert loss.item() < first * 0.2, (first, loss.item())


def test_bidir_raises():
    with pytest.raises(NotImplementedError):
        AWDLSTM(vocab_sz=10, bidir=True)


def test_qrnn_window_validation():
    with pytest.raises(ValueError):
        qrnn_forward(torch.randn(1, 2, 4), torch.zeros(1, 3),
                     torch.randn(9, 12), torch.zeros(9), window=3)


def test_gates_gemm_chunking_matches_unchunked():
    from code_intelligence_amd.ops.qrnn import _gates_gemm
    torch.manual_seed(5)
    flat = torch.randn(37, 8)
    w = torch.randn(12, 8)
    b = torch.randn(12)
    ref = torch.addmm(b, flat, w.t())
    # no-grad path (out= chunks)
    with torch.no_grad():
        out = _gates_gemm(flat, w, b, max_rows=10)
    assert torch.allclose(out, ref, atol=1e-6)
    # autograd path (cat chunks) — gradients flow
    fw = flat.clone().requires_grad_()
    ww = w.clone().requires_grad_()
    out2 = _gates_gemm(fw, ww, b, max_rows=10)
    assert torch.allclose(out2, ref, atol=1e-6)
    out2.sum().backward()
    assert fw.grad is not None and ww.grad is not None
    ref2 = torch.addmm(b, fw.detach(), ww.detach().t())
    gref = torch.autograd.functional.vjp(
        lambda a, c: torch.addmm(b, a, c.t()).sum(), (fw.detach(), ww.detach()))[1]
    assert torch.allclose(fw.grad, gref[0], atol=1e-6)
    assert torch.allclose(ww.grad, gref[1], atol=1e-6)
