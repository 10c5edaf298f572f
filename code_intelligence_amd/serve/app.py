"""Embedding REST service — flask-app contract of the reference
(Issue_Embeddings/flask_app/app.py):

* ``POST /text`` with JSON {"title": ..., "body": ...} -> raw little-endian
  float32 bytes of the (1, 2400) embedding (app.py:49-76; clients do
  ``np.frombuffer(r.content, '<f4')`` — flask_app/README.md:40-41)
* ``GET /healthz`` -> 200 (deployment readiness probe)
* ``POST /texts`` (new, batched): JSON {"documents": [{title, body}, ...]}
  -> {"shape": [N, 2400], "data": base64(f32 bytes)} — the MI355X-native
  bulk path the reference left unfinished (app.py:78-95 ``all_issues``).

Unlike the reference's single-threaded CPU flask replicas (9 of them,
deployments.yaml:6), one MI355X process serves batched requests through
the hipGraph-captured encoder.
"""
from __future__ import annotations

import base64
import hashlib
import logging
import os

from flask import Flask, jsonify, request

from ..engine.inference import InferenceWrapper

log = logging.getLogger(__name__)


def _resolve_model_path(model_path: str | None) -> str:
    """Reference app.py:20-34 downloads the model artifact from GCS at pod
    start; here gs:// URIs resolve through the object store into a local
    cache dir."""
    path = model_path or os.environ.get("MODEL_PATH", "model_files")
    if not path.startswith("gs://"):
        return path
    import tempfile
    from ..gh.gcs_util import default_store
    store = default_store()
    cache = os.path.join(tempfile.gettempdir(), "ci_model_cache")
    os.makedirs(cache, exist_ok=True)
    for name in ("config.json", "vocab.json", "encoder.pth"):
        store.download(f"{path.rstrip('/')}/{name}", os.path.join(cache, name))
    return cache


def create_app(wrapper: InferenceWrapper | None = None,
               model_path: str | None = None) -> Flask:
    app = Flask("issue_embedding_server")
    # bound request bodies: one multi-MB body would otherwise occupy the
    # single-threaded server for an unbounded stretch (advisor r1 finding)
    app.config["MAX_CONTENT_LENGTH"] = int(
        os.environ.get("CI_SERVE_MAX_BYTES", str(16 * 1024 * 1024)))
    if wrapper is None:
        wrapper = InferenceWrapper(
            model_path=_resolve_model_path(model_path),
            use_graphs=os.environ.get("CI_SERVE_GRAPHS", "0") == "1")
    app.config["wrapper"] = wrapper
    # dynamic micro-batching (CI_SERVE_BATCH_MS>0): concurrent /text
    # callers share GPU batches through the bulk path — the reference
    # needed 9 replicas for concurrency; one MI355X process batches it
    batch_ms = float(os.environ.get("CI_SERVE_BATCH_MS", "0"))
    batcher = None
    if batch_ms > 0:
        from .batcher import MicroBatcher
        batcher = MicroBatcher(wrapper, window_ms=batch_ms,
                               max_batch=int(os.environ.get(
                                   "CI_SERVE_BATCH_MAX", "64")))
    app.config["batcher"] = batcher

    @app.route("/healthz", methods=["GET"])
    def healthz():
        return jsonify({"ok": True})

    try:
        from prometheus_client import Counter, Histogram, generate_latest
        # re-registration (tests build several apps per process) -> reuse
        try:
            req_count = Counter("embedding_requests_total", "POST /text requests")
            req_lat = Histogram("embedding_request_seconds", "request latency")
        except ValueError:
            from prometheus_client import REGISTRY
            req_count = REGISTRY._names_to_collectors["embedding_requests_total"]
            req_lat = REGISTRY._names_to_collectors["embedding_request_seconds"]
    except ImportError:  # pragma: no cover
        req_count = req_lat = generate_latest = None

    @app.route("/text", methods=["POST"])
    def text():
        import time as _time
        t0 = _time.perf_counter()
        data = request.get_json(force=True)
        doc = wrapper.process_dict({"title": data.get("title", ""),
                                    "body": data.get("body", "")})
        if batcher is not None:
            emb = batcher.embed(doc["text"]).astype("<f4")
        else:
            emb = wrapper.get_pooled_features(doc["text"]).numpy() \
                .astype("<f4")
        payload = emb.tobytes()
        md5 = hashlib.md5(payload).hexdigest()
        log.debug("embedding md5=%s", md5)
        if req_count is not None:
            req_count.inc()
            req_lat.observe(_time.perf_counter() - t0)
        # md5 echoed for the reference's change-debugging workflow
        # (app.py:73-75 / repo_specific_model.py:179-181)
        return payload, 200, {"Content-Type": "application/octet-stream",
                              "X-Embedding-MD5": md5}

    @app.route("/metrics")
    def metrics():
        if generate_latest is None:
            return "prometheus_client not installed", 501
        return generate_latest(), 200, {"Content-Type": "text/plain"}

    @app.route("/texts", methods=["POST"])
    def texts():
        data = request.get_json(force=True)
        docs = data.get("documents", [])
        texts_ = [wrapper.process_dict(
            {"title": d.get("title", ""), "body": d.get("body", "")})["text"]
            for d in docs]
        emb = wrapper.texts_to_embedding(texts_, bs=int(data.get("bs", 100)))
        emb = emb.astype("<f4")
        return jsonify({"shape": list(emb.shape),
                        "data": base64.b64encode(emb.tobytes()).decode()})

    return app


def main():
    import argparse
    p = argparse.ArgumentParser()
    p.add_argument("--model_path", default=os.environ.get("MODEL_PATH", "model_files"))
    p.add_argument("--host", default="0.0.0.0")
    p.add_argument("--port", type=int, default=8080)
    args = p.parse_args()
    app = create_app(model_path=args.model_path)
    # mirror the reference: debug mode is forbidden in serving
    # (app.py:122-128). Single-threaded unless micro-batching is on —
    # batching needs concurrent request threads to batch across.
    threaded = app.config.get("batcher") is not None
    app.run(host=args.host, port=args.port, debug=False, threaded=threaded)


if __name__ == "__main__":
    main()
