"""needs-sync / needs-train HTTP server (reference: Label_Microservice/go/
cmd/automl/pkg/server/server.go + kpt.go).

* GET /needsSync — compare the registry's latest trained model for the
  dataset against the value deployed in config (the reference reads the
  kpt setter 'automl-model' from a Kptfile, kpt.go:37-59; here a YAML
  config file with the same setter key). Response:
  {"needsSync": bool, "parameters": {"name": <latest-model>}}
  (server.go:50-112)
* GET /needsTrain — true when no trained model exists or the newest is
  older than --retrainInterval (default 12h, main.go:48; server.go:116-176)
"""
from __future__ import annotations

import datetime
import logging
from pathlib import Path

import yaml
from flask import Flask, jsonify

from .registry import LocalModelRegistry

log = logging.getLogger(__name__)

DEFAULT_RETRAIN_INTERVAL_H = 12.0  # automl main.go:48


def read_deployed_setter(config_path, setter: str = "automl-model") -> str | None:
    """Read the deployed-model 'setter' value from a kpt-style YAML config."""
    p = Path(config_path)
    if not p.exists():
        return None
    data = yaml.safe_load(p.read_text()) or {}
    # kpt Kptfile layout: openAPI.definitions['io.k8s.cli.setters.<name>'].x-k8s-cli.setter.value
    defs = (((data.get("openAPI") or {}).get("definitions")) or {})
    key = f"io.k8s.cli.setters.{setter}"
    if key in defs:
        return (((defs[key].get("x-k8s-cli") or {}).get("setter")) or {}).get("value")
    # plain layout fallback: {'setters': {name: value}}
    return (data.get("setters") or {}).get(setter)


def create_app(registry: LocalModelRegistry, dataset: str, config_path,
               retrain_interval_h: float = DEFAULT_RETRAIN_INTERVAL_H) -> Flask:
    app = Flask("needs_sync_server")

    @app.route("/needsSync")
    def needs_sync():
        latest = registry.latest_trained(dataset)
        deployed = read_deployed_setter(config_path)
        if latest is None:
            return jsonify({"needsSync": False, "reason": "no trained model"})
        needs = latest.name != deployed
        return jsonify({"needsSync": needs,
                        "parameters": {"name": latest.name},
                        "deployed": deployed})

    @app.route("/needsTrain")
    def needs_train():
        latest = registry.latest_trained(dataset)
        if registry.is_training(dataset):
            return jsonify({"needsTrain": False, "reason": "training in progress"})
        if latest is None:
            return jsonify({"needsTrain": True, "reason": "no model"})
        created = datetime.datetime.fromisoformat(latest.create_time)
        age_h = (datetime.datetime.now(datetime.timezone.utc) - created
                 ).total_seconds() / 3600.0
        return jsonify({"needsTrain": age_h > retrain_interval_h,
                        "age_hours": round(age_h, 2)})

    @app.route("/healthz")
    def healthz():
        return jsonify({"ok": True})

    return app


def main():  # pragma: no cover
    import argparse
    p = argparse.ArgumentParser()
    p.add_argument("--registry", required=True)
    p.add_argument("--dataset", required=True)
    p.add_argument("--config", required=True)
    p.add_argument("--retrainInterval", type=float,
                   default=DEFAULT_RETRAIN_INTERVAL_H)
    p.add_argument("--port", type=int, default=8090)
    args = p.parse_args()
    app = create_app(LocalModelRegistry(args.registry), args.dataset,
                     args.config, args.retrainInterval)
    app.run(host="0.0.0.0", port=args.port)


if __name__ == "__main__":  # pragma: no cover
    main()
