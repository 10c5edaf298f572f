"""200-step trainer soak at the bench shape (fp8-CE default path):
asserts finite losses, stable device memory, stable host RSS."""
import sys, time, json, os, resource
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
import torch
from code_intelligence_amd.data.synthetic import synthetic_issue_tokens
from code_intelligence_amd.models.awd_lstm import AWDLSTM
from code_intelligence_amd.train.trainer import LMTrainer, TrainConfig

steps = int(sys.argv[1]) if len(sys.argv) > 1 else 200
torch.manual_seed(0)
m = AWDLSTM(vocab_sz=60000, emb_sz=800, n_hid=2400, n_layers=4
            ).to("cuda", torch.bfloat16)
tr = LMTrainer(m, TrainConfig())
m.train(); m.reset(512)
docs = synthetic_issue_tokens(64, 60000, seed=7, mean_len=300)
stream = torch.tensor([t for d in docs for t in d], dtype=torch.int64)
need = 512 * 513 * 8
stream = stream.repeat(need // stream.numel() + 1)[:need].view(-1, 513).to("cuda")
losses, mems, rss = [], [], []
t0 = time.perf_counter()
for i in range(steps):
    s = (i * 512) % (stream.shape[0] - 511)
    win = stream[s:s+512]
    l = tr.train_step(win[:, :-1], win[:, 1:], 1e-3)
    assert l == l and abs(l) < 1e4, (i, l)
    if i % 20 == 0:
        torch.cuda.synchronize()
        mems.append(torch.cuda.memory_allocated() // (1 << 20))
        rss.append(resource.getrusage(resource.RUSAGE_SELF).ru_maxrss // 1024)
    losses.append(l)
torch.cuda.synchronize()
dt = time.perf_counter() - t0
print(json.dumps({
    "steps": steps, "ms_per_step": round(dt / steps * 1e3, 1),
    "tokens_per_s": round(512 * 512 * steps / dt),
    "first_loss": round(losses[0], 3), "last_loss": round(losses[-1], 3),
    "mem_mb_first": mems[0], "mem_mb_last": mems[-1],
    "rss_mb_first": rss[0], "rss_mb_last": rss[-1]}))
assert mems[-1] <= mems[1] + 64, mems   # no device-memory creep after warmup
assert rss[-1] <= rss[1] + 256, rss     # no host RSS creep
print("soak ok")
