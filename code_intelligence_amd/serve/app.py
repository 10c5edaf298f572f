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

import numpy as np
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
    if wrapper is None:
        wrapper = InferenceWrapper(
            model_path=_resolve_model_path(model_path),
            use_graphs=os.environ.get("CI_SERVE_GRAPHS", "0") == "1")
    app.config["wrapper"] = wrapper

    @app.route("/healthz", methods=["GET"])
    def healthz():
        return jsonify({"ok": True})

    @app.route("/text", methods=["POST"])
    def text():
        data = request.get_json(force=True)
        doc = wrapper.process_dict({"title": data.get("title", ""),
                                    "body": data.get("body", "")})
        emb = wrapper.get_pooled_features(doc["text"]).numpy().astype("<f4")
        payload = emb.tobytes()
        log.debug("embedding md5=%s", hashlib.md5(payload).hexdigest())
        return payload, 200, {"Content-Type": "application/octet-stream"}

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
    # mirror the reference: debug mode is forbidden in serving (app.py:122-128)
    app.run(host=args.host, port=args.port, debug=False, threaded=False)


if __name__ == "__main__":
    main()
