"""Concurrent /text throughput with vs without dynamic micro-batching
(CI_SERVE_BATCH_MS) at the deployed shape. 32 client threads hammer the
flask app over HTTP; the batched mode shares GPU batches."""
import sys, time, threading, json, os
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
import numpy as np
import requests as rq
import torch

from code_intelligence_amd.serve.app import create_app
from code_intelligence_amd.engine.inference import InferenceWrapper
from code_intelligence_amd.models.awd_lstm import AWDLSTM
from code_intelligence_amd.text.tokenizer import Vocab, defaults_specials
from code_intelligence_amd.data.synthetic import synthetic_issue_texts

torch.manual_seed(0)
v = Vocab(defaults_specials + [f"w{i}" for i in range(59991)])
m = AWDLSTM(vocab_sz=len(v), emb_sz=800, n_hid=2400, n_layers=4)
docs = synthetic_issue_texts(200, seed=3)
N_CLIENTS, N_REQ = 32, 12  # per client

for batch_ms in (0, 4):
    os.environ["CI_SERVE_BATCH_MS"] = str(batch_ms)
    port = 8310 + batch_ms
    app = create_app(wrapper=InferenceWrapper(encoder=m.encoder, vocab=v))
    threading.Thread(
        target=lambda: app.run(host="127.0.0.1", port=port, debug=False,
                               threaded=(batch_ms > 0)),
        daemon=True).start()
    for _ in range(80):
        try:
            if rq.get(f"http://127.0.0.1:{port}/healthz", timeout=1).ok:
                break
        except Exception:
            time.sleep(0.5)
    # warmup
    for d in docs[:8]:
        rq.post(f"http://127.0.0.1:{port}/text", json=d, timeout=60)
    lat, lock = [], threading.Lock()

    def client(cid):
        for i in range(N_REQ):
            d = docs[(cid * N_REQ + i) % 200]
            t0 = time.perf_counter()
            r = rq.post(f"http://127.0.0.1:{port}/text", json=d, timeout=120)
            dt = time.perf_counter() - t0
            assert r.ok and len(r.content) == 9600
            with lock:
                lat.append(dt)
    t0 = time.perf_counter()
    ths = [threading.Thread(target=client, args=(c,)) for c in range(N_CLIENTS)]
    for t in ths: t.start()
    for t in ths: t.join()
    wall = time.perf_counter() - t0
    lat.sort()
    b = app.config.get("batcher")
    print(json.dumps({
        "batch_ms": batch_ms, "clients": N_CLIENTS,
        "requests": N_CLIENTS * N_REQ,
        "req_per_s": round(N_CLIENTS * N_REQ / wall, 1),
        "p50_ms": round(lat[len(lat) // 2] * 1e3, 1),
        "p95_ms": round(lat[int(len(lat) * 0.95)] * 1e3, 1),
        "batches": getattr(b, "batches", None),
        "batched_reqs": getattr(b, "batched_requests", None)}))
    if b is not None:
        b.close()
