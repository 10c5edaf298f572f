"""Isolated fp8-resident CE component timings at the bench chunk shape:
which piece pays and which saves vs the bf16-resident path, plus a
numeric proof of the _scaled_mm scale_result dequant convention.

Run on an MI355X box: python scripts/fp8r_probe.py
"""
import sys
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import time
import torch

if not torch.cuda.is_available():
    raise SystemExit("GPU probe - run under gpurun")

from code_intelligence_amd.ops import extension as ext
lib = ext.require()

dev = "cuda:0"
torch.manual_seed(0)
C, E, V = 16384, 800, 60000
f8 = torch.float8_e4m3fn
h = torch.randn(C, E, device=dev, dtype=torch.bfloat16) * 0.5
w = torch.randn(V, E, device=dev, dtype=torch.bfloat16) * 0.05
b32 = torch.randn(V, device=dev, dtype=torch.float32) * 0.01
tgt = torch.randint(0, V, (C,), device=dev)


def bench(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3  # ms


fl = 2.0 * C * E * V / 1e12  # TFLOP
sa = (h.abs().amax().float() / 448).clamp_min(1e-12)
sw = (w.abs().amax().float() / 448).clamp_min(1e-12)
h8 = (h * (1.0 / sa)).clamp(-448, 448).to(f8)
w8 = (w * (1.0 / sw)).clamp(-448, 448).to(f8)
w8_t = w8.t()
s_out = torch.full((), 64.0 * 2 / 448, device=dev)

logits_bf = torch.empty(C, V, device=dev, dtype=torch.bfloat16)
logits8 = torch.empty(C, V, device=dev, dtype=f8)

t = bench(lambda: torch.mm(h, w.t(), out=logits_bf))
print(f"bf16 mm out=:                 {t:8.3f} ms  {fl/t*1e3:6.0f} TF")

t = bench(lambda: torch._scaled_mm(h8, w8_t, scale_a=sa, scale_b=sw,
                                   out_dtype=torch.bfloat16, out=logits_bf))
print(f"fp8 scaled_mm -> bf16 out=:   {t:8.3f} ms  {fl/t*1e3:6.0f} TF")

t = bench(lambda: torch._scaled_mm(h8, w8_t, scale_a=sa, scale_b=sw,
                                   scale_result=s_out, out_dtype=f8,
                                   out=logits8))
print(f"fp8 scaled_mm -> fp8 out=:    {t:8.3f} ms  {fl/t*1e3:6.0f} TF")

# scale_result convention: dequant = stored * s_out ?
ref = torch.mm(h.float(), w.float().t())
got_mul = logits8.float() * s_out
err_mul = (got_mul - ref).abs().max() / ref.abs().max()
got_div = logits8.float() / s_out
err_div = (got_div - ref).abs().max() / ref.abs().max()
print(f"scale_result convention: err(out*s)={float(err_mul):.4f} "
      f"err(out/s)={float(err_div):.4f}  (small one is the convention)")

lse = torch.empty(C, device=dev, dtype=torch.float32)
tl = torch.empty(C, device=dev, dtype=torch.float32)
t = bench(lambda: lib.ce_rowstats(logits_bf, tgt, b32, lse, tl))
print(f"ce_rowstats bf16:             {t:8.3f} ms")
t = bench(lambda: lib.ce_rowstats_fp8(logits8, tgt, b32, s_out, lse, tl))
print(f"ce_rowstats fp8:              {t:8.3f} ms")

scale = torch.full((1,), 1.0 / C, device=dev)
scratch = torch.empty(C, V, device=dev, dtype=torch.bfloat16)
t = bench(lambda: lib.ce_dlogits(logits_bf, tgt, b32, lse, scale))
print(f"ce_dlogits bf16 (in-place):   {t:8.3f} ms")
t = bench(lambda: lib.ce_dlogits_fp8(logits8, tgt, b32, s_out, lse, scale,
                                     scratch, 448.0))
print(f"ce_dlogits fp8 (+bf16 scr):   {t:8.3f} ms")

# dh GEMM: (C,V) x (V,E)
dh = torch.empty(C, E, device=dev, dtype=torch.bfloat16)
fl2 = 2.0 * C * E * V / 1e12
t = bench(lambda: torch.mm(logits_bf, w, out=dh))
print(f"dh bf16 mm:                   {t:8.3f} ms  {fl2/t*1e3:6.0f} TF")
w8_cm = w8.t().contiguous().t()
sc_st = (scale.reshape(()) / 448.0)
t = bench(lambda: torch._scaled_mm(logits8, w8_cm, scale_a=sc_st, scale_b=sw,
                                   out_dtype=torch.bfloat16, out=dh))
print(f"dh fp8 scaled_mm:             {t:8.3f} ms  {fl2/t*1e3:6.0f} TF")

# quantize/copy costs
t = bench(lambda: (h * (1.0 / sa)).clamp(-448, 448).to(f8))
print(f"h chunk quantize:             {t:8.3f} ms")
t = bench(lambda: (w * (1.0 / sw)).clamp(-448, 448).to(f8))
print(f"w quantize (60k x 800):       {t:8.3f} ms")
t = bench(lambda: w8.t().contiguous())
print(f"w8 transpose copy:            {t:8.3f} ms")
wr = torch.empty(C, E, device=dev, dtype=torch.bfloat16)
t = bench(lambda: (h * w[tgt]).sum(dim=1, dtype=torch.float32))
print(f"exact tgt gather-dot:         {t:8.3f} ms")
