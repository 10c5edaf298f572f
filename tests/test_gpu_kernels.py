"""GPU numerics tests: every gfx950 HIP kernel vs the plain PyTorch fp32
CPU reference of the same op. Marked gpu; run via
`python -m pytest tests -m gpu` on an MI355X box."""
import os

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def _cpu_lstm_ref(x, h0, c0, w_ih, w_hh, b_ih, b_hh):
    from code_intelligence_amd.ops.lstm import _cpu_lstm_loop
    return _cpu_lstm_loop(x, h0, c0, w_ih, w_hh, b_ih, b_hh)


@pytest.mark.parametrize("mode", ["lib", "fused"])
@pytest.mark.parametrize("shape", [(8, 5, 64, 96), (130, 3, 128, 256)])
def test_lstm_forward_matches_cpu_fp32(mode, shape):
    torch.manual_seed(0)
    B, T, In, H = shape
    os.environ["CI_LSTM_MODE"] = mode
    x32 = torch.randn(B, T, In)
    w_ih = torch.randn(4 * H, In) * 0.1
    w_hh = torch.randn(4 * H, H) * 0.1
    b_ih = torch.randn(4 * H) * 0.05
    b_hh = torch.randn(4 * H) * 0.05
    out_ref, h_ref, c_ref = _cpu_lstm_ref(
        x32, torch.zeros(B, H), torch.zeros(B, H), w_ih, w_hh, b_ih, b_hh)

    from code_intelligence_amd.ops.lstm import lstm_forward
    dt = torch.bfloat16 if mode == "fused" else torch.float32
    dev_args = [t.to(DEV, dt) for t in (x32,)] + \
        [torch.zeros(B, H, device=DEV, dtype=dt), torch.zeros(B, H, device=DEV, dtype=dt)] + \
        [t.to(DEV, dt) for t in (w_ih, w_hh, b_ih, b_hh)]
    out, (hT, cT) = lstm_forward(*dev_args)
    tol = 0.05 if dt == torch.bfloat16 else 2e-4
    assert torch.allclose(out.float().cpu(), out_ref, atol=tol), \
        (out.float().cpu() - out_ref).abs().max()
    assert torch.allclose(hT.float().cpu(), h_ref, atol=tol)


@pytest.mark.parametrize("mode", ["lib", "fused"])
def test_lstm_backward_matches_cpu_fp32(mode):
    torch.manual_seed(1)
    B, T, In, H = 16, 4, 32, 48
    os.environ["CI_LSTM_MODE"] = mode
    dt = torch.bfloat16 if mode == "fused" else torch.float32

    def run(device, dtype):
        torch.manual_seed(2)
        x = torch.randn(B, T, In).to(device, dtype).requires_grad_(True)
        w_ih = (torch.randn(4 * H, In) * 0.1).to(device, dtype).requires_grad_(True)
        w_hh = (torch.randn(4 * H, H) * 0.1).to(device, dtype).requires_grad_(True)
        b_ih = (torch.randn(4 * H) * 0.05).to(device, dtype).requires_grad_(True)
        b_hh = (torch.randn(4 * H) * 0.05).to(device, dtype).requires_grad_(True)
        h0 = torch.zeros(B, H, device=device, dtype=dtype)
        c0 = torch.zeros(B, H, device=device, dtype=dtype)
        from code_intelligence_amd.ops.lstm import lstm_forward
        out, (hT, cT) = lstm_forward(x, h0, c0, w_ih, w_hh, b_ih, b_hh)
        loss = out.float().pow(2).mean()
        loss.backward()
        return {"x": x.grad, "w_ih": w_ih.grad, "w_hh": w_hh.grad,
                "b_ih": b_ih.grad}

    ref = run("cpu", torch.float32)
    got = run(DEV, dt)
    tol = 0.05 if dt == torch.bfloat16 else 1e-3
    for k in ref:
        r, g = ref[k].float(), got[k].float().cpu()
        denom = r.abs().max().clamp_min(1e-6)
        assert ((r - g).abs().max() / denom) < tol, (k, (r - g).abs().max(), denom)


def test_concat_pool_matches_cpu():
    torch.manual_seed(0)
    B, T, H = 33, 40, 256
    h = torch.randn(B, T, H)
    lengths = torch.randint(1, T + 1, (B,))
    from code_intelligence_amd.ops.pool import concat_pool, _cpu_concat_pool
    ref = _cpu_concat_pool(h, lengths)
    got = concat_pool(h.to(DEV, torch.bfloat16), lengths.to(DEV)).float().cpu()
    assert torch.allclose(got, ref, atol=0.03), (got - ref).abs().max()


def test_tied_ce_matches_cpu():
    torch.manual_seed(0)
    N, H, V = 64, 32, 1000
    h = torch.randn(N, H)
    w = torch.randn(V, H) * 0.1
    b = torch.randn(V) * 0.1
    t = torch.randint(0, V, (N,))
    from code_intelligence_amd.ops.crossentropy import tied_decoder_ce
    ref = F.cross_entropy(F.linear(h, w, b), t)

    hd = h.to(DEV).requires_grad_(True)
    wd = w.to(DEV).requires_grad_(True)
    bd = b.to(DEV).requires_grad_(True)
    loss = tied_decoder_ce(hd, wd, bd, t.to(DEV))
    assert abs(float(loss) - float(ref)) < 1e-3
    loss.backward()
    # grads vs autograd reference
    h2 = h.clone().requires_grad_(True)
    w2 = w.clone().requires_grad_(True)
    b2 = b.clone().requires_grad_(True)
    F.cross_entropy(F.linear(h2, w2, b2), t).backward()
    for g, r in ((hd.grad, h2.grad), (wd.grad, w2.grad), (bd.grad, b2.grad)):
        assert torch.allclose(g.float().cpu(), r, atol=1e-3), (g.float().cpu() - r).abs().max()


def test_fused_adamw_gpu_matches_torch():
    torch.manual_seed(0)
    from code_intelligence_amd.ops.adam import FusedAdamW
    p1 = torch.nn.Parameter(torch.randn(1000, device=DEV))
    p2 = torch.nn.Parameter(p1.detach().clone())
    o1 = FusedAdamW([p1], lr=1e-2, weight_decay=0.01, betas=(0.9, 0.99))
    o2 = torch.optim.AdamW([p2], lr=1e-2, weight_decay=0.01, betas=(0.9, 0.99))
    for i in range(5):
        g = torch.randn(1000, device=DEV)
        p1.grad = g.clone()
        p2.grad = g.clone()
        o1.step()
        o2.step()
    assert torch.allclose(p1, p2, atol=1e-5), (p1 - p2).abs().max()


def test_fused_adamw_bf16_master_weights():
    from code_intelligence_amd.ops.adam import FusedAdamW
    p = torch.nn.Parameter(torch.randn(512, device=DEV, dtype=torch.bfloat16))
    o = FusedAdamW([p], lr=1e-2)
    before = p.detach().float().clone()
    p.grad = torch.ones_like(p)
    o.step()
    assert not torch.equal(before, p.detach().float())
    assert o.state[p]["master"].dtype == torch.float32


def test_native_extension_is_loaded():
    """Guard against silent eager fallback: the in-tree .so must be loaded."""
    from code_intelligence_amd.ops import extension
    lib = extension.require()
    assert lib.__file__.endswith(".so")
    assert "code_intelligence_amd" in lib.__file__


@pytest.mark.parametrize("mode", ["lib", "fused"])
def test_lstm_tiny_shapes(mode):
    """Edge grids: B=1 (heavy M-tail), T=1, odd batch."""
    from code_intelligence_amd.ops.lstm import lstm_forward, _cpu_lstm_loop
    os.environ["CI_LSTM_MODE"] = mode
    for B, T, In, H in [(1, 3, 16, 32), (3, 1, 32, 64), (5, 2, 24, 40)]:
        torch.manual_seed(B * 100 + T)
        x = torch.randn(B, T, In)
        w_ih = torch.randn(4 * H, In) * 0.2
        w_hh = torch.randn(4 * H, H) * 0.2
        b_ih = torch.randn(4 * H) * 0.1
        b_hh = torch.randn(4 * H) * 0.1
        ref, h_ref, c_ref = _cpu_lstm_loop(
            x, torch.zeros(B, H), torch.zeros(B, H), w_ih, w_hh, b_ih, b_hh)
        dt = torch.bfloat16 if mode == "fused" else torch.float32
        out, (hT, cT) = lstm_forward(
            x.to(DEV, dt), torch.zeros(B, H, device=DEV, dtype=dt),
            torch.zeros(B, H, device=DEV, dtype=dt),
            w_ih.to(DEV, dt), w_hh.to(DEV, dt),
            b_ih.to(DEV, dt), b_hh.to(DEV, dt))
        tol = 0.06 if dt == torch.bfloat16 else 3e-4
        assert torch.allclose(out.float().cpu(), ref, atol=tol), \
            (B, T, (out.float().cpu() - ref).abs().max())


def test_concat_pool_length_one_and_full():
    from code_intelligence_amd.ops.pool import concat_pool, _cpu_concat_pool
    torch.manual_seed(0)
    h = torch.randn(2, 5, 8)
    lengths = torch.tensor([1, 5])
    ref = _cpu_concat_pool(h, lengths)
    got = concat_pool(h.to(DEV), lengths.to(DEV)).cpu()
    assert torch.allclose(got, ref, atol=1e-5)


def test_ce_chunk_boundary():
    """N not divisible by CHUNK and N < CHUNK both exercised."""
    import code_intelligence_amd.ops.crossentropy as ce
    from code_intelligence_amd.ops.crossentropy import tied_decoder_ce
    import torch.nn.functional as F
    old = ce._FusedCEFunction.CHUNK
    try:
        ce._FusedCEFunction.CHUNK = 7
        os.environ["CI_CE_CHUNK"] = "7"
        torch.manual_seed(0)
        h = torch.randn(23, 16, device=DEV)
        w = torch.randn(50, 16, device=DEV) * 0.3
        t = torch.randint(0, 50, (23,), device=DEV)
        loss = tied_decoder_ce(h, w, None, t)
        ref = F.cross_entropy(F.linear(h, w), t)
        assert abs(float(loss) - float(ref)) < 1e-4
    finally:
        ce._FusedCEFunction.CHUNK = old
        os.environ.pop("CI_CE_CHUNK", None)


@pytest.mark.parametrize("H", [96, 10])  # vector path / scalar-tail path
def test_qrnn_fo_pool_matches_cpu_fp32(H):
    """HIP fo-pool scan fwd+bwd vs the plain-torch reference (fp32)."""
    from code_intelligence_amd.ops.qrnn import _fo_pool_torch, fo_pool
    torch.manual_seed(0)
    B, T = 5, 13
    gates = torch.randn(B, T, 3 * H)
    c0 = torch.randn(B, H)
    gc, cc = gates.clone().requires_grad_(), c0.clone().requires_grad_()
    h_ref, cT_ref = _fo_pool_torch(gc, cc)
    dh = torch.randn_like(h_ref)
    dcT = torch.randn_like(cT_ref)
    (h_ref * dh).sum().add_((cT_ref * dcT).sum()).backward()

    gg = gates.to(DEV).requires_grad_()
    cg = c0.to(DEV).requires_grad_()
    h, cT = fo_pool(gg, cg)
    assert torch.allclose(h.cpu(), h_ref.detach(), atol=1e-5)
    assert torch.allclose(cT.cpu(), cT_ref.detach(), atol=1e-5)
    (h * dh.to(DEV)).sum().add_((cT * dcT.to(DEV)).sum()).backward()
    assert torch.allclose(gg.grad.cpu(), gc.grad, atol=1e-5)
    assert torch.allclose(cg.grad.cpu(), cc.grad, atol=1e-5)


def test_qrnn_layer_gpu_matches_cpu():
    """Whole QRNN layer (GEMM + scan + window-2 shift) GPU bf16 vs CPU fp32."""
    from code_intelligence_amd.models import WeightDroppedQRNN
    torch.manual_seed(1)
    B, T, E, H = 4, 10, 64, 80
    layer = WeightDroppedQRNN(E, H, weight_p=0.0, window=2).eval()
    x = torch.randn(B, T, E)
    c0 = torch.zeros(B, H)
    ref, (_, cT_ref) = layer(x, (c0, c0))
    layer.reset()
    lg = WeightDroppedQRNN(E, H, weight_p=0.0, window=2).to(DEV, torch.bfloat16).eval()
    lg.load_state_dict({k: v.to(DEV, torch.bfloat16)
                        for k, v in layer.state_dict().items()})
    out, (_, cT) = lg(x.to(DEV, torch.bfloat16),
                      (c0.to(DEV, torch.bfloat16),) * 2)
    assert (out.float().cpu() - ref).abs().max() < 0.05
    assert (cT.float().cpu() - cT_ref).abs().max() < 0.05


# ---- K1: embedding gather + row dropout (wired round 2) ------------------

def test_emb_gather_scatter_matches_masked_embedding():
    """emb_gather with an explicit row mask == F.embedding on the masked
    table; emb_scatter == autograd's dW, both vs CPU fp32."""
    from code_intelligence_amd.ops import extension as ext
    lib = ext.require()
    torch.manual_seed(3)
    V, E, B, T, pad = 500, 64, 8, 12, 1
    w = torch.randn(V, E)
    ids = torch.randint(0, V, (B, T))
    ids[0, :3] = pad
    mask = (torch.rand(V) > 0.2).float() / 0.8

    # CPU fp32 reference through autograd
    w_ref = w.clone().requires_grad_(True)
    out_ref = F.embedding(ids, w_ref * mask[:, None], padding_idx=pad)
    g = torch.randn_like(out_ref)
    out_ref.backward(g)

    wg = w.to(DEV, torch.bfloat16)
    out = lib.emb_gather(wg, ids.to(DEV).contiguous(), mask.to(DEV))
    assert out.dtype == torch.bfloat16
    assert torch.allclose(out.float().cpu(), out_ref.detach(), atol=0.03)
    dw = lib.emb_scatter(g.to(DEV, torch.bfloat16), ids.to(DEV).contiguous(),
                         mask.to(DEV), V, pad)
    assert dw.dtype == torch.float32
    assert torch.allclose(dw.cpu(), w_ref.grad, atol=0.05), \
        (dw.cpu() - w_ref.grad).abs().max()


def test_embedding_dropout_module_gpu_matches_cpu_eval():
    """EmbeddingDropout eval path on GPU (kernel gather, no mask) must
    equal the CPU F.embedding path bit-for-bit-ish in bf16 tolerance."""
    from code_intelligence_amd.models.awd_lstm import EmbeddingDropout
    torch.manual_seed(4)
    emb = torch.nn.Embedding(300, 48, padding_idx=1)
    mod = EmbeddingDropout(emb, 0.02).eval()
    ids = torch.randint(0, 300, (4, 9))
    # CPU references FIRST: .to(DEV) mutates the module in place
    ref = mod(ids)
    ref_s = mod(ids, scale=0.5)
    gmod = EmbeddingDropout(emb.to(DEV).to(torch.bfloat16), 0.02).eval()
    out = gmod(ids.to(DEV))
    assert torch.allclose(out.float().cpu(), ref, atol=0.02)
    # scale argument parity (serve path passes scale=None; engine uses it)
    out_s = gmod(ids.to(DEV), scale=0.5)
    assert torch.allclose(out_s.float().cpu(), ref_s, atol=0.02)


def test_embedding_dropout_module_gpu_train_grads():
    """Training path: the fused-row-mask kernel's forward zeros whole word
    rows consistently and backward routes masked grads to only the looked-
    up rows (pad row excluded)."""
    from code_intelligence_amd.models.awd_lstm import EmbeddingDropout
    torch.manual_seed(5)
    V, E, pad = 200, 32, 1
    emb = torch.nn.Embedding(V, E, padding_idx=pad).to(DEV)
    mod = EmbeddingDropout(emb, 0.3).train()
    ids = torch.randint(2, V, (6, 7), device=DEV)
    ids[0, 0] = pad
    out = mod(ids)
    # whole-row consistency: a dropped word is dropped at EVERY position
    flat_ids = ids.reshape(-1)
    flat_out = out.reshape(-1, E)
    for tok in flat_ids.unique():
        rows = flat_out[flat_ids == tok]
        zeroed = (rows.abs().sum(dim=1) == 0)
        assert bool(zeroed.all()) or bool((~zeroed).all())
    loss = out.float().pow(2).sum()
    loss.backward()
    g = emb.weight.grad
    assert g is not None
    assert g[pad].abs().sum() == 0
    looked = set(flat_ids.tolist()) - {pad}
    untouched = [i for i in range(V) if i not in looked and i != pad]
    assert g[untouched].abs().sum() == 0
    assert g[sorted(looked)].abs().sum() > 0


# ---- K3: seeded DropConnect kernel (wired round 2) -----------------------

def test_dropconnect_kernel_stats_and_determinism():
    from code_intelligence_amd.ops import extension as ext
    lib = ext.require()
    torch.manual_seed(6)
    w = torch.randn(4 * 96, 96, device=DEV, dtype=torch.bfloat16)
    p = 0.2
    a = lib.dropconnect_apply(w, 12345, p)
    b = lib.dropconnect_apply(w, 12345, p)
    assert torch.equal(a, b)  # same seed -> same mask
    c = lib.dropconnect_apply(w, 54321, p)
    assert not torch.equal(a, c)
    drop_frac = float((a == 0).float().mean())
    assert abs(drop_frac - p) < 0.02
    kept = a[a != 0]
    ref = w[a != 0].float() / (1 - p)
    assert torch.allclose(kept.float(), ref, atol=0.02)


def test_dropconnect_autograd_grad_uses_same_mask():
    from code_intelligence_amd.ops.dropout import _DropConnectFunction
    torch.manual_seed(7)
    w = torch.randn(64, 48, device=DEV, dtype=torch.float32,
                    requires_grad=True)
    out = _DropConnectFunction.apply(w, 0.25, 777)
    g = torch.randn_like(out)
    out.backward(g.clone())
    dropped = out == 0
    assert torch.all(w.grad[dropped] == 0)
    assert torch.allclose(w.grad[~dropped], g[~dropped] / 0.75, atol=1e-6)


def test_weightdrop_lstm_train_step_runs_with_kernels():
    """A training forward+backward of WeightDroppedLSTM on GPU goes
    through dropconnect_apply (train) and produces finite grads."""
    from code_intelligence_amd.models.awd_lstm import WeightDroppedLSTM
    torch.manual_seed(8)
    m = WeightDroppedLSTM(32, 48, weight_p=0.2).to(DEV, torch.bfloat16).train()
    x = torch.randn(4, 6, 32, device=DEV, dtype=torch.bfloat16)
    h = torch.zeros(4, 48, device=DEV, dtype=torch.bfloat16)
    out, _ = m(x, (h, h.clone()))
    out.float().pow(2).mean().backward()
    for n, p_ in m.named_parameters():
        assert p_.grad is not None and torch.isfinite(p_.grad.float()).all(), n


# ---- deployed-shape correctness ---------------

@pytest.mark.timeout(600)
def test_lstm_deployed_shape_fused_vs_lib_vs_cpu_spot():
    """H=2400, B=512, T=2 (the bench kernel grid incl. edge tiles): fused
    and lib paths must agree with each other everywhere and with a CPU
    fp32 reference on spot rows."""
    from code_intelligence_amd.ops.lstm import lstm_forward, _cpu_lstm_loop
    torch.manual_seed(9)
    B, T, In, H = 512, 2, 800, 2400
    x32 = torch.randn(B, T, In) * 0.5
    w_ih = torch.randn(4 * H, In) * 0.02
    w_hh = torch.randn(4 * H, H) * 0.02
    b_ih = torch.randn(4 * H) * 0.02
    b_hh = torch.randn(4 * H) * 0.02
    spot = [0, 1, 255, 510, 511]
    out_ref, h_ref, c_ref = _cpu_lstm_loop(
        x32[spot], torch.zeros(len(spot), H), torch.zeros(len(spot), H),
        w_ih, w_hh, b_ih, b_hh)

    outs = {}
    for mode in ("lib", "fused"):
        os.environ["CI_LSTM_MODE"] = mode
        args = [x32.to(DEV, torch.bfloat16),
                torch.zeros(B, H, device=DEV, dtype=torch.bfloat16),
                torch.zeros(B, H, device=DEV, dtype=torch.bfloat16),
                w_ih.to(DEV, torch.bfloat16), w_hh.to(DEV, torch.bfloat16),
                b_ih.to(DEV, torch.bfloat16), b_hh.to(DEV, torch.bfloat16)]
        out, (hT, cT) = lstm_forward(*args)
        outs[mode] = (out.float().cpu(), hT.float().cpu())
    os.environ.pop("CI_LSTM_MODE", None)
    # fused vs lib: same bf16 inputs, both fp32-accumulated -> tight
    assert torch.allclose(outs["fused"][0], outs["lib"][0], atol=0.02), \
        (outs["fused"][0] - outs["lib"][0]).abs().max()
    # spot rows vs CPU fp32 (bf16 input rounding dominates)
    for mode in ("lib", "fused"):
        assert torch.allclose(outs[mode][0][spot], out_ref, atol=0.08), \
            (mode, (outs[mode][0][spot] - out_ref).abs().max())
        assert torch.allclose(outs[mode][1][spot], h_ref, atol=0.08), mode


# ---- fp8 CE GEMM path (CI_CE_FP8R, round 2) -------------------------------

def _ce_both_paths(N, H, V, seed=11):
    from code_intelligence_amd.ops.crossentropy import tied_decoder_ce
    torch.manual_seed(seed)
    h0 = torch.randn(N, H) * 0.5
    w0 = torch.randn(V, H) * 0.05
    b0 = torch.randn(V) * 0.01
    t0 = torch.randint(0, V, (N,))
    res = {}
    for mode in ("bf16", "fp8r"):
        os.environ["CI_CE_FP8R"] = "1" if mode == "fp8r" else "0"
        h = h0.to(DEV, torch.bfloat16).requires_grad_(True)
        w = w0.to(DEV, torch.bfloat16).requires_grad_(True)
        b = b0.to(DEV, torch.bfloat16).requires_grad_(True)
        loss = tied_decoder_ce(h, w, b, t0.to(DEV))
        loss.backward()
        res[mode] = (float(loss), h.grad.float().cpu(), w.grad.float().cpu(),
                     b.grad.float().cpu())
    os.environ.pop("CI_CE_FP8R", None)
    return res


@pytest.mark.timeout(300)
def test_ce_fp8_resident_close_to_bf16():
    """fp8-input CE GEMMs + fp8 dh: loss within ~2% of the bf16 path and
    gradients directionally identical (cosine > 0.98)."""
    res = _ce_both_paths(N=2048, H=160, V=4096)
    lb, hb, wb, bb = res["bf16"]
    lf, hf, wf, bf_ = res["fp8r"]
    assert abs(lf - lb) / abs(lb) < 0.02, (lf, lb)

    def cos(a, b):
        return float((a.flatten() @ b.flatten()) /
                     (a.norm() * b.norm() + 1e-30))
    assert cos(hf, hb) > 0.98, cos(hf, hb)
    assert cos(wf, wb) > 0.98, cos(wf, wb)
    assert cos(bf_, bb) > 0.97, cos(bf_, bb)


@pytest.mark.timeout(300)
def test_ce_fp8_resident_tail_chunk():
    """N not a multiple of 16 exercises the bf16-GEMM tail fallback for
    both forward quantize-store and backward dh."""
    os.environ["CI_CE_CHUNK"] = "64"
    try:
        res = _ce_both_paths(N=72 + 9, H=160, V=4096, seed=12)
    finally:
        os.environ.pop("CI_CE_CHUNK", None)
    lb, hb, *_ = res["bf16"]
    lf, hf, *_ = res["fp8r"]
    assert abs(lf - lb) / abs(lb) < 0.03
    assert torch.isfinite(hf).all()


# ---- fused AR/TAR kernel (round 2) ----------------------------------------

@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_artar_fused_matches_eager(dtype):
    from code_intelligence_amd.ops.artar import artar_loss
    torch.manual_seed(13)
    B, T, H = 16, 12, 64
    alpha, beta = 2.0, 1.0
    # r as a transpose view of time-major storage (the trainer's layout)
    r_store = torch.randn(T, B, H, device=DEV, dtype=dtype)
    r = r_store.transpose(0, 1).requires_grad_(True)
    out = (torch.randn(B, T, H, device=DEV, dtype=dtype)
           ).requires_grad_(True)
    reg = artar_loss(out, r, alpha, beta)
    (reg * 1.7).backward()

    out2 = out.detach().clone().float().requires_grad_(True)
    r2 = r.detach().clone().float().requires_grad_(True)
    ref = alpha * out2.pow(2).mean() + \
        beta * (r2[:, 1:] - r2[:, :-1]).pow(2).mean()
    (ref * 1.7).backward()
    tol = 1e-4 if dtype == torch.float32 else 0.02
    assert abs(float(reg) - float(ref)) / float(ref) < tol
    assert torch.allclose(out.grad.float(), out2.grad, atol=tol), \
        (out.grad.float() - out2.grad).abs().max()
    assert torch.allclose(r.grad.float(), r2.grad, atol=tol), \
        (r.grad.float() - r2.grad).abs().max()


def test_artar_loss_trainer_path_gpu():
    """loss_on_batch on GPU engages the fused kernel (transpose-view raw
    outputs) and produces grads matching an eager-formula run."""
    from code_intelligence_amd.models.awd_lstm import AWDLSTM
    from code_intelligence_amd.train.trainer import LMTrainer, TrainConfig
    torch.manual_seed(14)
    m = AWDLSTM(vocab_sz=256, emb_sz=32, n_hid=64, n_layers=2,
                output_p=0, hidden_p=0, input_p=0, embed_p=0, weight_p=0
                ).to(DEV, torch.bfloat16)
    tr = LMTrainer(m, TrainConfig(alpha=2.0, beta=1.0))
    m.train()
    x = torch.randint(0, 256, (4, 16), device=DEV)
    y = torch.roll(x, -1, 1)
    m.reset(4)
    loss_fused = tr.loss_on_batch(x, y)
    loss_fused.backward()
    g_fused = {n: p.grad.float().cpu().clone()
               for n, p in m.named_parameters() if p.grad is not None}
    for p in m.parameters():
        p.grad = None
    # eager formula on the same weights/batch
    m.reset(4)
    enc, dec = m.encoder, m.decoder
    raw, outs = enc(x)
    out = dec.output_dp(outs[-1])
    from code_intelligence_amd.ops.crossentropy import tied_decoder_ce
    loss2 = tied_decoder_ce(out.reshape(-1, out.shape[-1]),
                            dec.decoder.weight, dec.decoder.bias,
                            y.reshape(-1))
    loss2 = loss2 + 2.0 * out.float().pow(2).mean()
    rr = raw[-1].float()
    loss2 = loss2 + 1.0 * (rr[:, 1:] - rr[:, :-1]).pow(2).mean()
    loss2.backward()
    assert abs(float(loss_fused) - float(loss2)) / float(loss2) < 0.02
    for n, p in m.named_parameters():
        if p.grad is None:
            continue
        g2 = p.grad.float().cpu()
        denom = g2.abs().max().clamp_min(1e-5)
        assert ((g_fused[n] - g2).abs().max() / denom) < 0.05, n


def test_artar_engages_on_trainer_layout():
    """The dropout output keeps the LSTM's time-major layout (its C-contig
    base is the (T,B,H) storage). The fused kernel must engage for that
    layout — r2 regression: the first wiring silently fell back to eager
    because out.is_contiguous() was False."""
    from code_intelligence_amd.ops.artar import artar_loss
    torch.manual_seed(15)
    B, T, H = 8, 6, 64
    r_store = torch.randn(T, B, H, device=DEV, dtype=torch.bfloat16)
    r = r_store.transpose(0, 1).requires_grad_(True)
    mask = torch.rand(B, 1, H, device=DEV, dtype=torch.bfloat16)
    out = r.detach() * mask  # trainer layout: time-major base
    assert not out.is_contiguous() and out.transpose(0, 1).is_contiguous()
    out.requires_grad_(True)
    reg = artar_loss(out, r, 2.0, 1.0)
    assert "ARTAR" in type(reg.grad_fn).__name__, type(reg.grad_fn).__name__
    reg.backward()
    out2 = out.detach().float().clone().requires_grad_(True)
    r2 = r.detach().float().clone().requires_grad_(True)
    ref = 2.0 * out2.pow(2).mean() + \
        1.0 * (r2[:, 1:] - r2[:, :-1]).pow(2).mean()
    ref.backward()
    assert abs(float(reg) - float(ref)) / float(ref) < 0.02
    assert torch.allclose(out.grad.float(), out2.grad, atol=0.02)
    assert torch.allclose(r.grad.float(), r2.grad, atol=0.02)


# ---- fp8 LSTM GEMM path (CI_LSTM_FP8, round 2) ----------------------------

@pytest.mark.timeout(300)
def test_lstm_fp8_close_to_bf16():
    """fp8 input-projection + fp8 recurrent GEMM (fixed 1/448 h scale):
    outputs within e4m3 tolerance of the bf16 lib path, grads
    directionally identical."""
    from code_intelligence_amd.ops.lstm import lstm_forward
    torch.manual_seed(21)
    B, T, In, H = 32, 12, 64, 96

    def run(fp8):
        os.environ["CI_LSTM_FP8"] = "1" if fp8 else "0"
        torch.manual_seed(22)
        x = (torch.randn(B, T, In) * 0.5).to(DEV, torch.bfloat16) \
            .requires_grad_(True)
        w_ih = (torch.randn(4 * H, In) * 0.1).to(DEV, torch.bfloat16) \
            .requires_grad_(True)
        w_hh = (torch.randn(4 * H, H) * 0.1).to(DEV, torch.bfloat16) \
            .requires_grad_(True)
        b = torch.zeros(4 * H, device=DEV, dtype=torch.bfloat16)
        h0 = torch.zeros(B, H, device=DEV, dtype=torch.bfloat16)
        out, (hT, cT) = lstm_forward(x, h0, h0.clone(), w_ih, w_hh, b,
                                     b.clone())
        out.float().pow(2).mean().backward()
        return (out.detach().float().cpu(), x.grad.float().cpu(),
                w_hh.grad.float().cpu())

    try:
        out_b, gx_b, gw_b = run(False)
        out_f, gx_f, gw_f = run(True)
    finally:
        os.environ.pop("CI_LSTM_FP8", None)
    # forward: fp8 quantization noise bounded
    assert torch.allclose(out_f, out_b, atol=0.06), \
        (out_f - out_b).abs().max()

    def cos(a, b):
        return float((a.flatten() @ b.flatten()) /
                     (a.norm() * b.norm() + 1e-30))
    assert cos(gx_f, gx_b) > 0.99, cos(gx_f, gx_b)
    assert cos(gw_f, gw_b) > 0.99, cos(gw_f, gw_b)


def test_quantize_e4m3_kernel():
    from code_intelligence_amd.ops import extension as ext
    lib = ext.require()
    torch.manual_seed(23)
    x = torch.randn(1000, device=DEV, dtype=torch.bfloat16) * 3
    s = (x.abs().amax().float() / 448.0)
    q = torch.empty(1000, dtype=torch.float8_e4m3fn, device=DEV)
    lib.quantize_e4m3(x, q, s)
    ref = ((x.float() / s).clamp(-448, 448)).to(torch.float8_e4m3fn)
    assert torch.equal(q.view(torch.uint8), ref.view(torch.uint8))


# ---- persistent whole-sequence GEMV (CI_SERVE_PERSISTENT, round 2) --------

@pytest.mark.timeout(300)
def test_persistent_gemv_matches_per_step():
    from code_intelligence_amd.ops import extension as ext
    lib = ext.require()
    torch.manual_seed(33)
    for B, T, H in [(1, 40, 2400), (4, 17, 96), (2, 3, 50)]:
        xp = torch.randn(T, B, 4 * H, device=DEV, dtype=torch.bfloat16) * 0.3
        bias = torch.randn(4 * H, device=DEV, dtype=torch.float32) * 0.05
        h0 = torch.randn(B, H, device=DEV, dtype=torch.bfloat16) * 0.3
        c0 = torch.randn(B, H, device=DEV, dtype=torch.float32) * 0.3
        w = torch.randn(4 * H, H, device=DEV, dtype=torch.bfloat16) * 0.05

        def bufs():
            return (torch.empty(T, B, H, device=DEV, dtype=torch.bfloat16),
                    torch.empty(T, B, H, device=DEV, dtype=torch.float32),
                    torch.empty(T, B, 4 * H, device=DEV,
                                dtype=torch.bfloat16))

        hs1, cs1, g1 = bufs()
        lib.lstm_seq_forward_gemv(xp, bias, h0, c0, w, hs1, cs1, g1)
        hs2, cs2, g2 = bufs()
        ws = torch.zeros(4, dtype=torch.int32, device=DEV)
        nb = lib.lstm_seq_forward_gemv_persistent(xp, bias, h0, c0, w,
                                                  hs2, cs2, g2, ws)
        torch.cuda.synchronize()
        assert nb > 0, "persistent grid unavailable"
        assert int(ws[2]) == 0, "grid barrier bailed"
        assert torch.equal(hs1, hs2), (B, T, H,
                                       (hs1.float() - hs2.float()).abs().max())
        assert torch.allclose(cs1, cs2, atol=1e-5)
        assert torch.equal(g1, g2)


def test_persistent_gemv_serve_wrapper_path():
    """InferenceWrapper single-request path under CI_SERVE_PERSISTENT=1
    equals the default path's embedding."""
    import numpy as np
    from code_intelligence_amd.models.awd_lstm import AWDLSTM
    from code_intelligence_amd.engine.inference import InferenceWrapper
    from code_intelligence_amd.text.tokenizer import Vocab, defaults_specials
    torch.manual_seed(34)
    words = [f"w{i}" for i in range(3000)]
    v = Vocab(defaults_specials + words)
    model = AWDLSTM(vocab_sz=len(v), emb_sz=64, n_hid=96, n_layers=2)
    w = InferenceWrapper(encoder=model.encoder, vocab=v)
    text = "xxfld 1 crash w7 w8 w9 xxfld 2 " + " ".join(
        f"w{i % 2900}" for i in range(60))
    a = w.get_pooled_features(text).numpy()
    os.environ["CI_SERVE_PERSISTENT"] = "1"
    try:
        b = w.get_pooled_features(text).numpy()
    finally:
        os.environ.pop("CI_SERVE_PERSISTENT", None)
    np.testing.assert_allclose(a, b, atol=1e-3)


@pytest.mark.timeout(300)
def test_ce_fp8_extreme_logits_no_overflow():
    """Large-magnitude hidden states / weights (|logits| in the hundreds)
    must not overflow the fp8 CE path: per-tensor input scales keep the
    GEMM inputs in range and the loss stays finite and close to bf16."""
    res = _ce_both_paths(N=512, H=160, V=2048, seed=41)
    # scale inputs up via a fresh run with big tensors
    from code_intelligence_amd.ops.crossentropy import tied_decoder_ce
    torch.manual_seed(42)
    h = (torch.randn(512, 160) * 8).to(DEV, torch.bfloat16).requires_grad_(True)
    w = (torch.randn(2048, 160) * 2).to(DEV, torch.bfloat16).requires_grad_(True)
    b = (torch.randn(2048) * 4).to(DEV, torch.bfloat16).requires_grad_(True)
    t = torch.randint(0, 2048, (512,), device=DEV)
    os.environ["CI_CE_FP8R"] = "1"
    try:
        loss = tied_decoder_ce(h, w, b, t)
        loss.backward()
    finally:
        os.environ.pop("CI_CE_FP8R", None)
    assert torch.isfinite(loss), loss
    for g in (h.grad, w.grad, b.grad):
        assert torch.isfinite(g.float()).all()
    # bf16 reference on the same tensors
    h2 = h.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    os.environ["CI_CE_FP8R"] = "0"
    try:
        ref = tied_decoder_ce(h2, w2, b2, t)
    finally:
        os.environ.pop("CI_CE_FP8R", None)
    assert abs(float(loss) - float(ref)) / max(abs(float(ref)), 1e-6) < 0.1


def test_ce_fp8_engages_by_default():
    """The DEFAULT environment must route the training CE through the
    fp8 function (CI_CE_FP8R defaults on) — guards against silent
    fallback to the bf16 path in the driver's bench."""
    from code_intelligence_amd.ops.crossentropy import tied_decoder_ce
    assert os.environ.get("CI_CE_FP8R") in (None, "1")
    torch.manual_seed(51)
    h = torch.randn(64, 32, device=DEV, dtype=torch.bfloat16,
                    requires_grad=True)
    w = torch.randn(512, 32, device=DEV, dtype=torch.bfloat16,
                    requires_grad=True)
    b = torch.zeros(512, device=DEV, dtype=torch.bfloat16,
                    requires_grad=True)
    t = torch.randint(0, 512, (64,), device=DEV)
    loss = tied_decoder_ce(h, w, b, t)
    assert "FusedCEFp8" in type(loss.grad_fn).__name__, \
        type(loss.grad_fn).__name__
