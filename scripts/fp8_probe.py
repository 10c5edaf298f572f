import sys; sys.path.insert(0, str(__import__("pathlib").Path(__file__).resolve().parents[1]))
import time, torch
if not torch.cuda.is_available():
    raise SystemExit(f"{__file__} is a GPU probe workload - run it on an MI355X box (gpurun)")
dev = "cuda:0"
torch.manual_seed(0)
C, E, V = 16384, 800, 60000
h = torch.randn(C, E, device=dev, dtype=torch.bfloat16)
w = torch.randn(V, E, device=dev, dtype=torch.bfloat16) * 0.1

def bench(fn, iters=20):
    for _ in range(3): fn()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize(); return (time.perf_counter() - t0) / iters

fl = 2.0 * C * E * V
t = bench(lambda: torch.mm(h, w.t()))
print(f"bf16 mm: {t*1e3:.3f} ms {fl/t/1e12:.0f} TF")
try:
    f8 = torch.float8_e4m3fn
    hs = h.abs().amax() / 448.0
    ws = w.abs().amax() / 448.0
    h8 = (h / hs).to(f8)
    w8 = (w / ws).to(f8)
    out = torch.empty(C, V, device=dev, dtype=torch.bfloat16)
    def run():
        return torch._scaled_mm(h8, w8.t(), scale_a=hs.float(), scale_b=ws.float(), out_dtype=torch.bfloat16)
    r = run()
    t = bench(run)
    print(f"fp8 scaled_mm: {t*1e3:.3f} ms {fl/t/1e12:.0f} TF")
    ref = torch.mm(h, w.t())
    rel = (r.float() - ref.float()).abs().max() / ref.float().abs().max()
    print("max rel err vs bf16 mm:", float(rel))
except Exception as e:
    print("fp8 path failed:", type(e).__name__, str(e)[:300])
