"""Kubeflow-info chatbot webhook (reference: chatbot/pkg/server.go).

Fulfils Dialogflow-style intents about label ownership: loads a
labels-owners.yaml (labels.go:23-46), regex-matches query parameters
'{area}/{value}' against label names as '{prefix}.*/.*{value}.*'
(server.go:164-192) and answers with the owners (server.go:195-221).
Prometheus metrics at /metrics + a heartbeat counter (server.go:26-30,
61-66,152)."""
from __future__ import annotations

import logging
import re
import threading
import time
from typing import List, Optional

import yaml
from flask import Flask, jsonify, request

try:
    from prometheus_client import Counter, generate_latest
    _HAVE_PROM = True
except ImportError:  # pragma: no cover
    _HAVE_PROM = False

log = logging.getLogger(__name__)


class KubeflowLabels:
    """labels-owners.yaml: {labels: [{name: 'area/ops', owners: [...]}, ...]}"""

    def __init__(self, labels: List[dict]):
        self.labels = labels

    @classmethod
    def load(cls, path) -> "KubeflowLabels":
        data = yaml.safe_load(open(path)) or {}
        return cls(data.get("labels", []))

    def match_labels(self, area: str, value: Optional[str] = None) -> List[dict]:
        """server.go:164-192: '{prefix}.*/.*{value}.*' regex match."""
        if value:
            pattern = re.compile(f"{re.escape(area)}.*/.*{re.escape(value)}.*")
        else:
            pattern = re.compile(f"{re.escape(area)}.*")
        return [l for l in self.labels if pattern.match(l.get("name", ""))]


def create_app(labels: KubeflowLabels) -> Flask:
    app = Flask("kubeflow_chatbot")
    heartbeats = None
    requests_total = None
    if _HAVE_PROM:
        heartbeats = Counter("chatbot_heartbeats_total", "heartbeat ticks")
        requests_total = Counter("chatbot_webhook_requests_total", "webhook hits")

        def beat():  # heartbeat every 10 s (server.go:61-66)
            while True:
                heartbeats.inc()
                time.sleep(10)
        threading.Thread(target=beat, daemon=True).start()

    @app.route("/")
    def index():
        return "kubeflow label chatbot"

    @app.route("/dialogflow/webhook", methods=["POST"])
    def webhook():
        if requests_total is not None:
            requests_total.inc()
        body = request.get_json(force=True) or {}
        params = ((body.get("queryResult") or {}).get("parameters")) or {}
        area = params.get("area") or params.get("label") or ""
        value = params.get("value") or params.get("platform")
        matches = labels.match_labels(area, value)
        if not matches:
            text = f"I could not find a label matching '{area}'."
        else:
            parts = []
            for m in matches[:5]:
                owners = ", ".join(m.get("owners") or []) or "nobody (unowned)"
                parts.append(f"label {m['name']} is owned by {owners}")
            text = "; ".join(parts)
        return jsonify({"fulfillmentText": text})

    @app.route("/metrics")
    def metrics():
        if not _HAVE_PROM:
            return "prometheus_client not installed", 501
        return generate_latest(), 200, {"Content-Type": "text/plain"}

    return app


def main():  # pragma: no cover
    import argparse
    p = argparse.ArgumentParser()
    p.add_argument("--labels", required=True, help="labels-owners.yaml path")
    p.add_argument("--port", type=int, default=8095)
    args = p.parse_args()
    app = create_app(KubeflowLabels.load(args.labels))
    app.run(host="0.0.0.0", port=args.port)


if __name__ == "__main__":  # pragma: no cover
    main()
