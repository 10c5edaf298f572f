"""Engine (InferenceWrapper) + embedding REST service tests (CPU)."""
import numpy as np
import pandas as pd
import torch

from code_intelligence_amd.engine.inference import InferenceWrapper, save_artifacts
from code_intelligence_amd.models.awd_lstm import AWDLSTM
from code_intelligence_amd.serve.app import create_app
from code_intelligence_amd.text.tokenizer import Vocab, defaults_specials


def _tiny_wrapper(tmp_path):
    torch.manual_seed(0)
    words = [f"w{i}" for i in range(200)]
    vocab = Vocab(defaults_specials + words)
    model = AWDLSTM(vocab_sz=len(vocab), emb_sz=16, n_hid=24, n_layers=2)
    save_artifacts(model, vocab, tmp_path / "artifacts")
    return InferenceWrapper(model_path=str(tmp_path / "artifacts"), device="cpu")


def test_wrapper_roundtrip_and_pool_shape(tmp_path):
    w = _tiny_wrapper(tmp_path)
    emb = w.get_pooled_features("w1 w2 w3 totally unknown words")
    assert emb.shape == (1, 48)  # 3 * emb_sz
    assert torch.isfinite(emb).all()


def test_wrapper_deterministic_and_reset(tmp_path):
    w = _tiny_wrapper(tmp_path)
    a = w.get_pooled_features("w1 w2 w3")
    b = w.get_pooled_features("w1 w2 w3")
    assert torch.allclose(a, b)  # encoder.reset() between requests


def test_df_to_embedding_order_preserved(tmp_path):
    w = _tiny_wrapper(tmp_path)
    df = pd.DataFrame({
        "title": ["w1", "w2 w3 w4 w5 w6 w7 w8", "w9 w10"],
        "body": ["short", "a much longer body " * 5, "mid length body"],
    })
    embs = w.df_to_embedding(df, bs=2)
    assert embs.shape == (3, 48)
    # row i must equal the single-doc embedding of doc i (order preserved
    # through sort-by-length batching — reference asserts this too)
    for i in range(3):
        doc = w.process_dict({"title": df["title"][i], "body": df["body"][i]})
        single = w.get_pooled_features(doc["text"]).numpy()[0]
        assert np.allclose(embs[i], single, atol=1e-4), i


def test_batched_equals_single_regardless_of_padding(tmp_path):
    w = _tiny_wrapper(tmp_path)
    texts = ["w1 w2", "w3 w4 w5 w6 w7 w8 w9 w10 w11 w12"]
    both = w.texts_to_embedding(texts, bs=2)
    one = w.texts_to_embedding([texts[0]], bs=1)
    assert np.allclose(both[0], one[0], atol=1e-4)


def test_flask_text_contract(tmp_path):
    w = _tiny_wrapper(tmp_path)
    app = create_app(wrapper=w)
    client = app.test_client()
    r = client.get("/healthz")
    assert r.status_code == 200
    r = client.post("/text", json={"title": "bug in w1", "body": "w2 w3"})
    assert r.status_code == 200
    vec = np.frombuffer(r.data, dtype="<f4")  # reference client contract
    assert vec.shape == (48,)
    r2 = client.post("/texts", json={"documents": [
        {"title": "bug", "body": "w1"}, {"title": "feat", "body": "w2"}]})
    assert r2.get_json()["shape"] == [2, 48]


def test_oom_backoff_halves_batch(tmp_path, monkeypatch):
    """reference inference.py:214-223: CUDA OOM -> halve bs until it fits."""
    w = _tiny_wrapper(tmp_path)
    calls = []
    real = w._encode_batch

    def flaky(ids, lengths):
        calls.append(ids.shape[0])
        if ids.shape[0] > 1 and len(calls) < 3:
            raise torch.cuda.OutOfMemoryError("fake OOM")
        return real(ids, lengths)

    monkeypatch.setattr(w, "_encode_batch", flaky)
    monkeypatch.setattr(torch.cuda, "empty_cache", lambda: None)
    out = w.texts_to_embedding(["w1 w2", "w3", "w4 w5 w6", "w7"], bs=4)
    assert out.shape == (4, 48)
    assert calls[0] == 4 and calls[1] == 2  # halved after OOM


def test_serve_model_uri_resolution(tmp_path, monkeypatch):
    """gs:// MODEL_PATH downloads artifacts through the object store."""
    from code_intelligence_amd.gh.gcs_util import ObjectStore
    import code_intelligence_amd.gh.gcs_util as gcs
    from code_intelligence_amd.serve.app import _resolve_model_path
    from code_intelligence_amd.engine.inference import save_artifacts
    from code_intelligence_amd.models.awd_lstm import AWDLSTM
    from code_intelligence_amd.text.tokenizer import Vocab, defaults_specials
    store = ObjectStore(root=tmp_path / "store")
    monkeypatch.setattr(gcs, "_default_store", store)
    model = AWDLSTM(vocab_sz=20, emb_sz=8, n_hid=12, n_layers=1)
    local = tmp_path / "art"
    save_artifacts(model, Vocab(defaults_specials), local)
    for name in ("config.json", "vocab.json", "encoder.pth"):
        store.upload(str(local / name), f"gs://models/lm/{name}")
    path = _resolve_model_path("gs://models/lm")
    import json as _json
    assert _json.loads(open(f"{path}/config.json").read())["emb_sz"] == 8


def test_wrapper_qrnn_encoder_roundtrip(tmp_path):
    """--qrnn models serve through the same artifact layout (config.json
    carries the flag; encoder.pth holds the QRNN state dict)."""
    torch.manual_seed(1)
    words = [f"w{i}" for i in range(100)]
    vocab = Vocab(defaults_specials + words)
    model = AWDLSTM(vocab_sz=len(vocab), emb_sz=16, n_hid=24, n_layers=2,
                    qrnn=True)
    save_artifacts(model, vocab, tmp_path / "art")
    w = InferenceWrapper(model_path=str(tmp_path / "art"), device="cpu")
    assert w.encoder.qrnn
    emb = w.get_pooled_features("w1 w2 w3")
    assert emb.shape == (1, 48) and torch.isfinite(emb).all()
    # embedding matches the in-memory model's own encoder output
    model.eval()
    model.reset(1)
    assert torch.allclose(emb, w.get_pooled_features("w1 w2 w3"), atol=1e-6)


def test_wrapper_qrnn_ignores_graph_optin(tmp_path):
    torch.manual_seed(2)
    vocab = Vocab(defaults_specials + [f"w{i}" for i in range(50)])
    model = AWDLSTM(vocab_sz=len(vocab), emb_sz=16, n_hid=24, n_layers=2,
                    qrnn=True)
    save_artifacts(model, vocab, tmp_path / "a")
    w = InferenceWrapper(model_path=str(tmp_path / "a"), device="cpu",
                         use_graphs=True)
    assert not w.use_graphs  # downgraded: capture unsupported for QRNN
    assert w.get_pooled_features("w1 w2").shape == (1, 48)


def test_serve_max_tokens_cap(tmp_path, monkeypatch):
    w = _tiny_wrapper(tmp_path)
    long_text = " ".join(f"w{i % 50}" for i in range(500))
    assert len(w.numericalize(long_text)) == 500
    monkeypatch.setenv("CI_SERVE_MAX_TOKENS", "64")
    assert len(w.numericalize(long_text)) == 64
    emb = w.get_pooled_features(long_text)
    assert emb.shape == (1, 48)


def test_concurrent_requests_serialized_correctly(tmp_path):
    """The wrapper's hidden-state carry must not corrupt results when
    called from multiple threads (encode lock)."""
    import threading
    w = _tiny_wrapper(tmp_path)
    ref = w.get_pooled_features("w1 w2 w3")
    errs = []

    def hammer():
        for _ in range(40):
            if not torch.allclose(w.get_pooled_features("w1 w2 w3"), ref,
                                  atol=1e-5):
                errs.append(1)

    ts = [threading.Thread(target=hammer) for _ in range(3)]
    for t in ts:
        t.start()
    for t in ts:
        t.join()
    assert not errs


def test_wire_level_http_contract(tmp_path):
    """Real HTTP server + requests client: the reference client recipe
    (np.frombuffer(r.content, '<f4')) works over the wire, with the md5
    header, /metrics and the batched /texts route."""
    import threading
    import time
    import requests
    from werkzeug.serving import make_server
    from code_intelligence_amd.serve.app import create_app

    w = _tiny_wrapper(tmp_path)
    app = create_app(wrapper=w)
    srv = make_server("127.0.0.1", 0, app)   # ephemeral port
    port = srv.server_port
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    try:
        base = f"http://127.0.0.1:{port}"
        assert requests.get(f"{base}/healthz", timeout=5).json() == {"ok": True}
        r = requests.post(f"{base}/text",
                          json={"title": "crash", "body": "w1 w2"}, timeout=10)
        emb = np.frombuffer(r.content, dtype="<f4")
        assert emb.shape == (48,) and np.isfinite(emb).all()
        assert len(r.headers["X-Embedding-MD5"]) == 32
        r2 = requests.post(f"{base}/texts", json={"documents": [
            {"title": "a", "body": "w1"}, {"title": "b", "body": "w2"}]},
            timeout=10)
        assert r2.json()["shape"] == [2, 48]
    finally:
        srv.shutdown()


def test_microbatcher_batches_concurrent_requests():
    """Concurrent embed() calls share a GPU batch: 12 callers inside one
    window must produce FEWER batches than requests, every caller gets
    its own correct row, and errors propagate per caller."""
    import threading
    import numpy as np
    from code_intelligence_amd.serve.batcher import MicroBatcher

    calls = []

    class _FakeWrapper:
        def texts_to_embedding(self, texts, bs=64):
            calls.append(list(texts))
            # row i encodes a hash of its text so callers can verify
            return np.stack([np.full(4, float(len(t)), dtype=np.float32)
                             for t in texts])

    mb = MicroBatcher(_FakeWrapper(), window_ms=50.0, max_batch=64)
    try:
        results = {}
        errs = []

        def call(i):
            try:
                results[i] = mb.embed("x" * (i + 1))
            except Exception as e:  # pragma: no cover
                errs.append(e)

        threads = [threading.Thread(target=call, args=(i,))
                   for i in range(12)]
        for t in threads:
            t.start()
        for t in threads:
            t.join(timeout=10)
        assert not errs
        assert len(results) == 12
        for i, emb in results.items():
            assert emb.shape == (1, 4)
            assert float(emb[0, 0]) == i + 1  # own row, not a neighbor's
        assert mb.batches < 12, mb.batches  # actually batched
        assert mb.batched_requests == 12
    finally:
        mb.close()


def test_microbatcher_propagates_errors():
    from code_intelligence_amd.serve.batcher import MicroBatcher
    import pytest as _pytest

    class _Boom:
        def texts_to_embedding(self, texts, bs=64):
            raise RuntimeError("encode failed")

    mb = MicroBatcher(_Boom(), window_ms=1.0)
    try:
        with _pytest.raises(RuntimeError, match="encode failed"):
            mb.embed("hello")
    finally:
        mb.close()


def test_serve_app_batched_text_route(monkeypatch, tmp_path):
    """CI_SERVE_BATCH_MS>0: the /text route returns the same payload
    contract through the batcher."""
    monkeypatch.setenv("CI_SERVE_BATCH_MS", "2")
    app = create_app(wrapper=_tiny_wrapper(tmp_path))
    assert app.config["batcher"] is not None
    client = app.test_client()
    r = client.post("/text", json={"title": "hello", "body": "world"})
    assert r.status_code == 200
    emb = np.frombuffer(r.data, "<f4")
    assert emb.ndim == 1 and emb.size > 0 and np.isfinite(emb).all()
    assert "X-Embedding-MD5" in r.headers
    app.config["batcher"].close()
