"""Randomized GPU kernel fuzz: random shapes through the lib LSTM
fwd+bwd, concat-pool, and both CE paths, each checked against the CPU
fp32 reference. Complements the fixed-shape pytest suite."""
import sys, random
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
import os
import torch
import torch.nn.functional as F
from code_intelligence_amd.ops.lstm import lstm_forward, _cpu_lstm_loop
from code_intelligence_amd.ops.pool import concat_pool, _cpu_concat_pool
from code_intelligence_amd.ops.crossentropy import tied_decoder_ce

dev = "cuda:0"
rng = random.Random(int(sys.argv[1]) if len(sys.argv) > 1 else 0)
fails = 0
for i in range(30):
    B = rng.randint(9, 200)   # >8 keeps us off the serve GEMV branch
    T = rng.randint(1, 10)
    In = rng.choice([8, 16, 24, 40, 64, 120])
    H = rng.choice([8, 16, 32, 56, 96, 160, 250])
    torch.manual_seed(i)
    x32 = torch.randn(B, T, In) * 0.5
    wi = torch.randn(4 * H, In) * 0.1
    wh = torch.randn(4 * H, H) * 0.1
    bi = torch.randn(4 * H) * 0.05
    bh = torch.randn(4 * H) * 0.05
    ref, href, cref = _cpu_lstm_loop(x32, torch.zeros(B, H),
                                     torch.zeros(B, H), wi, wh, bi, bh)
    x = x32.to(dev, torch.bfloat16).requires_grad_(True)
    args = [torch.zeros(B, H, device=dev, dtype=torch.bfloat16),
            torch.zeros(B, H, device=dev, dtype=torch.bfloat16)]
    wcl = [t.to(dev, torch.bfloat16).requires_grad_(True)
           for t in (wi, wh, bi, bh)]
    out, (hT, cT) = lstm_forward(x, *args, *wcl)
    if not torch.allclose(out.float().cpu(), ref, atol=0.08):
        print("lstm fwd", i, B, T, In, H,
              (out.float().cpu() - ref).abs().max()); fails += 1
    out.float().pow(2).mean().backward()  # bwd smoke: finite grads
    for t in [x] + wcl:
        if t.grad is not None and not torch.isfinite(t.grad.float()).all():
            print("lstm bwd nonfinite", i); fails += 1

for i in range(30):
    B, T, H = rng.randint(1, 40), rng.randint(1, 30), rng.randint(1, 300)
    torch.manual_seed(500 + i)
    h = torch.randn(B, T, H)
    lens = torch.randint(1, T + 1, (B,))
    got = concat_pool(h.to(dev, torch.bfloat16), lens.to(dev)).float().cpu()
    ref = _cpu_concat_pool(h, lens)
    if not torch.allclose(got, ref, atol=0.05):
        print("pool", i, B, T, H); fails += 1

for i in range(30):
    N = rng.choice([16, 48, 96, 97, 256, 1000])
    H = rng.choice([16, 32, 64, 160])
    V = rng.choice([64, 1000, 4096, 4111])
    torch.manual_seed(900 + i)
    h32 = torch.randn(N, H) * 0.5
    w32 = torch.randn(V, H) * 0.1
    b32 = torch.randn(V) * 0.05 if i % 2 else None
    t = torch.randint(0, V, (N,))
    ref = F.cross_entropy(F.linear(h32, w32, b32), t)
    for mode in ("0", "1"):
        os.environ["CI_CE_FP8R"] = mode
        hd = h32.to(dev, torch.bfloat16).requires_grad_(True)
        wd = w32.to(dev, torch.bfloat16).requires_grad_(True)
        bd = b32.to(dev, torch.bfloat16).requires_grad_(True) \
            if b32 is not None else None
        loss = tied_decoder_ce(hd, wd, bd, t.to(dev))
        tol = 0.02 if mode == "0" else 0.08
        if abs(float(loss) - float(ref)) / max(float(ref), 1e-6) > tol:
            print("ce", mode, i, N, H, V, float(loss), float(ref)); fails += 1
        loss.backward()
        for g in [hd.grad, wd.grad] + ([bd.grad] if bd is not None else []):
            if not torch.isfinite(g.float()).all():
                print("ce grad nonfinite", mode, i); fails += 1
os.environ.pop("CI_CE_FP8R", None)
print("fuzz fails:", fails)
sys.exit(1 if fails else 0)
