"""Deeper TunableOp search over the LSTM/CE weight-grad GEMM shapes
(the NT K=262144 GEMMs run ~790 TF, 92 ms/step total — is that the
library ceiling?). Writes the tuned solutions to gpurun_out/tune_dw*.csv
for merging into profiles/tunableop_mi355x*.csv if they beat the ship.

Run on an MI355X box: python scripts/dw_tune_probe.py
"""
import os

os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
os.environ["PYTORCH_TUNABLEOP_TUNING"] = "1"
os.environ["PYTORCH_TUNABLEOP_FILENAME"] = "gpurun_out/tune_dw.csv"
os.environ["PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS"] = "1000"
os.environ["PYTORCH_TUNABLEOP_MAX_TUNING_ITERATIONS"] = "300"

import time
import torch

if not torch.cuda.is_available():
    raise SystemExit("GPU probe - run under gpurun")
dev = "cuda:0"
torch.manual_seed(0)


def bench(fn, iters=10):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


shapes = [
    # (name, M, N, K): dW = A(K,M).t() @ B(K,N)
    ("lstm dW_hh big", 9600, 2400, 262144),
    ("lstm dW_ih l1", 9600, 800, 262144),
    ("lstm dW l4", 3200, 2400, 262144),
    ("ce dW", 60000, 800, 16384),
]
for name, M, N, K in shapes:
    a = torch.randn(K, M, device=dev, dtype=torch.bfloat16) * 0.01
    b = torch.randn(K, N, device=dev, dtype=torch.bfloat16) * 0.01
    t = bench(lambda: torch.mm(a.t(), b))
    tf = 2.0 * M * N * K / (t / 1e3) / 1e12
    print(f"{name:18s} ({M}x{N} K={K}): {t:8.3f} ms  {tf:6.0f} TF")
