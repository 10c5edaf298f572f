"""Would wavefront layer-pipelining pay? Measures whether N concurrent
per-timestep recurrent GEMMs (latency-bound: 555 TF of 2.5 PF peak) on
separate streams beat the same GEMMs sequentially, and the same for the
cell pointwise kernel. If the concurrency factor is ~1, the chip is
already saturated and the wavefront restructure buys nothing.
"""
import sys
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
import time
import torch
from code_intelligence_amd.ops import extension as ext

lib = ext.require()
dev = "cuda:0"
torch.manual_seed(0)
B, H = 512, 2400
nstreams = 3
hs = [torch.randn(B, H, device=dev, dtype=torch.bfloat16) for _ in range(nstreams)]
ws = [torch.randn(4 * H, H, device=dev, dtype=torch.bfloat16).t().contiguous().t()
      for _ in range(nstreams)]
wts = [w.t() for w in ws]
outs = [torch.empty(B, 4 * H, device=dev, dtype=torch.bfloat16)
        for _ in range(nstreams)]
streams = [torch.cuda.Stream() for _ in range(nstreams)]
REP = 64


def bench(fn, iters=5):
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def seq():
    for _ in range(REP):
        for i in range(nstreams):
            torch.mm(hs[i], wts[i], out=outs[i])


def conc():
    for i, st in enumerate(streams):
        with torch.cuda.stream(st):
            for _ in range(REP):
                torch.mm(hs[i], wts[i], out=outs[i])
    for st in streams:
        torch.cuda.current_stream().wait_stream(st)


t_seq = bench(seq)
t_conc = bench(conc)
print(f"recurrent GEMM x{nstreams}: seq {t_seq:7.2f} ms   "
      f"concurrent {t_conc:7.2f} ms   factor {t_seq/t_conc:.2f}x")

# cell pointwise kernels concurrently with a GEMM stream
xp = torch.randn(1, B, 4 * H, device=dev, dtype=torch.bfloat16)
bias = torch.zeros(4 * H, device=dev, dtype=torch.float32)
c0 = torch.zeros(B, H, device=dev, dtype=torch.float32)
hs_t = torch.empty(1, B, H, device=dev, dtype=torch.bfloat16)
cs_t = torch.empty(1, B, H, device=dev, dtype=torch.float32)
g_t = torch.empty(1, B, 4 * H, device=dev, dtype=torch.bfloat16)


def mixed_seq():
    for _ in range(REP):
        torch.mm(hs[0], wts[0], out=outs[0])
        lib.lstm_seq_forward_lib(xp, bias, hs[1], c0, ws[1], hs_t, cs_t, g_t)


def mixed_conc():
    with torch.cuda.stream(streams[0]):
        for _ in range(REP):
            torch.mm(hs[0], wts[0], out=outs[0])
    with torch.cuda.stream(streams[1]):
        for _ in range(REP):
            lib.lstm_seq_forward_lib(xp, bias, hs[1], c0, ws[1],
                                     hs_t, cs_t, g_t)
    for st in streams[:2]:
        torch.cuda.current_stream().wait_stream(st)


t_ms = bench(mixed_seq)
t_mc = bench(mixed_conc)
print(f"GEMM + cell mixed:     seq {t_ms:7.2f} ms   "
      f"concurrent {t_mc:7.2f} ms   factor {t_ms/t_mc:.2f}x")
