"""GEMM layout/chunk probes for the training-step hot shapes (GPU).

Times hipBLASLt through torch.mm for the backward dh GEMM layouts and the
CE chunk-size choices, so kernel work targets the measured best layout.
Run: python scripts/gemm_probe.py
"""
import sys
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import time

import torch

if not torch.cuda.is_available():
    raise SystemExit(f"{__file__} is a GPU probe workload - run it on an MI355X box (gpurun)")
dev = "cuda:0"
torch.manual_seed(0)


def bench(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def tf(flops, sec):
    return flops / sec / 1e12


B, H = 512, 2400
dt = torch.bfloat16

# --- backward dh GEMM: (B,4H) @ (4H,H) -------------------------------
dg = torch.randn(B, 4 * H, device=dev, dtype=dt)
w = torch.randn(4 * H, H, device=dev, dtype=dt)
w_t = w.t().contiguous()          # (H,4H)
out = torch.empty(B, H, device=dev, dtype=dt)
fl = 2.0 * B * 4 * H * H

t = bench(lambda: torch.mm(dg, w, out=out))
print(f"bwd dh NN   mm(dg, w)           : {t*1e6:8.1f} us  {tf(fl, t):6.0f} TF")
t = bench(lambda: torch.mm(dg, w_t.t(), out=out))
print(f"bwd dh NT   mm(dg, w_t.t())     : {t*1e6:8.1f} us  {tf(fl, t):6.0f} TF")

# --- forward rec GEMM: (B,H) @ (H,4H) --------------------------------
h = torch.randn(B, H, device=dev, dtype=dt)
out2 = torch.empty(B, 4 * H, device=dev, dtype=dt)
t = bench(lambda: torch.mm(h, w.t(), out=out2))
print(f"fwd rec NT  mm(h, w.t())        : {t*1e6:8.1f} us  {tf(fl, t):6.0f} TF")
t = bench(lambda: torch.mm(h, w_t, out=out2))
print(f"fwd rec NN  mm(h, w_t)          : {t*1e6:8.1f} us  {tf(fl, t):6.0f} TF")

# --- CE logits GEMM: (chunk, 800) @ (800, 60k) ------------------------
E, V = 800, 60000
emb = torch.randn(V, E, device=dev, dtype=dt)
for chunk in (4096, 8192, 16384, 32768):
    hh = torch.randn(chunk, E, device=dev, dtype=dt)
    logits = torch.empty(chunk, V, device=dev, dtype=dt)
    fl2 = 2.0 * chunk * E * V
    t = bench(lambda: torch.mm(hh, emb.t(), out=logits), iters=10)
    # scale to the full 262144 rows
    full = t * (262144 / chunk)
    print(f"CE fwd chunk={chunk:6}: {t*1e3:7.2f} ms  {tf(fl2, t):6.0f} TF  (full-batch {full*1e3:6.1f} ms)")
    dlog = logits
    dh_ = torch.empty(chunk, E, device=dev, dtype=dt)
    fl3 = 2.0 * chunk * E * V
    t = bench(lambda: torch.mm(dlog, emb, out=dh_), iters=10)
    print(f"CE dh  chunk={chunk:6}: {t*1e3:7.2f} ms  {tf(fl3, t):6.0f} TF")
    dw_ = torch.empty(V, E, device=dev, dtype=torch.float32)
    t = bench(lambda: torch.mm(dlog.t(), hh, out=None), iters=10)
    print(f"CE dE  chunk={chunk:6}: {t*1e3:7.2f} ms  {tf(fl3, t):6.0f} TF")

# --- dW batched GEMM: (4H, BT) @ (BT, H) ------------------------------
BT = 262144
dga = torch.randn(BT, 4 * H, device=dev, dtype=dt)
hp = torch.randn(BT, H, device=dev, dtype=dt)
fl4 = 2.0 * 4 * H * BT * H
t = bench(lambda: torch.mm(dga.t(), hp), iters=5)
print(f"dW_hh  (4H,BT)x(BT,H)           : {t*1e3:7.2f} ms  {tf(fl4, t):6.0f} TF")
t = bench(lambda: torch.mm(hp.t(), dga), iters=5)
print(f"dW_hh^T (H,BT)x(BT,4H)          : {t*1e3:7.2f} ms  {tf(fl4, t):6.0f} TF")

# --- xp fwd GEMM ------------------------------------------------------
x = torch.randn(BT, H, device=dev, dtype=dt)
fl5 = 2.0 * BT * H * 4 * H
t = bench(lambda: torch.mm(x, w.t()), iters=5)
print(f"xp (BT,H)x(H,4H) NT             : {t*1e3:7.2f} ms  {tf(fl5, t):6.0f} TF")
