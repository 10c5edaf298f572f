"""Probe: would fp8 pay on the LSTM GEMMs?

Shapes (deployed config): per-timestep recurrent (512,2400)x(2400,9600),
input projection (262144,In)x(In,9600), dx backward (262144,9600)x(9600,H).
Compares bf16 torch.mm vs fp8 _scaled_mm->bf16 (per-tensor scales,
quantize cost measured separately).

Run on an MI355X box: python scripts/lstm_fp8_probe.py
"""
import sys
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import time
import torch

if not torch.cuda.is_available():
    raise SystemExit("GPU probe - run under gpurun")
dev = "cuda:0"
torch.manual_seed(0)
f8 = torch.float8_e4m3fn


def bench(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def q(t):
    s = (t.abs().amax().float() / 448.0).clamp_min(1e-12)
    return (t * (1.0 / s)).clamp(-448, 448).to(f8), s


shapes = [
    ("recurrent (per t)", 512, 2400, 9600),
    ("xp l2-4", 262144, 2400, 9600),
    ("xp l1", 262144, 800, 9600),
    ("dx l2-3", 262144, 9600, 2400),
]
for name, M, K, N in shapes:
    a = torch.randn(M, K, device=dev, dtype=torch.bfloat16) * 0.3
    w = torch.randn(N, K, device=dev, dtype=torch.bfloat16) * 0.02  # (N,K) rm
    out = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
    fl = 2.0 * M * K * N / 1e12
    t_bf = bench(lambda: torch.mm(a, w.t(), out=out))
    a8, sa = q(a)
    w8, sw = q(w)
    w8t = w8.t()
    t_f8 = bench(lambda: torch._scaled_mm(a8, w8t, scale_a=sa, scale_b=sw,
                                          out_dtype=torch.bfloat16, out=out))
    t_q = bench(lambda: (a * (1.0 / sa)).clamp(-448, 448).to(f8))
    print(f"{name:18s} M{M} K{K} N{N}: bf16 {t_bf:7.3f} ms ({fl/t_bf*1e3:5.0f} TF)"
          f"  fp8 {t_f8:7.3f} ms ({fl/t_f8*1e3:5.0f} TF)  quantA {t_q:6.3f} ms")

# per-timestep call-overhead check: 64 sequential small scaled_mm calls
a = torch.randn(512, 2400, device=dev, dtype=torch.bfloat16)
w = torch.randn(9600, 2400, device=dev, dtype=torch.bfloat16) * 0.02
a8, sa = q(a)
w8, sw = q(w)
w8t = w8.t()
out = torch.empty(512, 9600, device=dev, dtype=torch.bfloat16)


def seq_bf():
    for _ in range(64):
        torch.mm(a, w.t(), out=out)


def seq_f8():
    for _ in range(64):
        torch._scaled_mm(a8, w8t, scale_a=sa, scale_b=sw,
                         out_dtype=torch.bfloat16, out=out)


print(f"64x sequential: bf16 {bench(seq_bf, 5)/64*1e3:6.1f} us/call   "
      f"fp8 {bench(seq_f8, 5)/64*1e3:6.1f} us/call")
