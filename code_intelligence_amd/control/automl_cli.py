"""Model-registry CLI (reference: Label_Microservice/go/cmd/automl/main.go —
cobra commands serve | get | deploy | label models | isTraining).

  python -m code_intelligence_amd.control.automl_cli --registry DIR get --dataset D
  ... deploy --name MODEL
  ... is-training --dataset D
  ... serve --dataset D --config Kptfile.yaml [--retrainInterval 12]
  ... train --dataset D --command "python -m ..."   (registers + runs)
"""
from __future__ import annotations

import argparse
import json
import shlex
import subprocess
import sys

from .registry import LocalModelRegistry


def cmd_get(reg: LocalModelRegistry, args):
    latest = reg.latest_trained(args.dataset)
    deployed = reg.latest_deployed(args.dataset)
    print(json.dumps({
        "latest_trained": latest.to_json() if latest else None,
        "latest_deployed": deployed.to_json() if deployed else None,
        "evaluation": reg.evaluation_at_confidence(latest.name)
        if latest else None}, indent=2))


def cmd_models(reg: LocalModelRegistry, args):
    print(json.dumps([r.to_json() for r in reg.list(args.dataset)], indent=2))


def cmd_deploy(reg: LocalModelRegistry, args):
    rec = reg.deploy(args.name)
    print(json.dumps(rec.to_json(), indent=2))


def cmd_is_training(reg: LocalModelRegistry, args):
    print(json.dumps({"isTraining": reg.is_training(args.dataset)}))


def cmd_train(reg: LocalModelRegistry, args):
    rec = reg.create_training(args.dataset)
    print(f"registered training run {rec.name}")
    if args.command:
        rc = subprocess.call(shlex.split(args.command))
        reg.finish_training(rec.name, ok=(rc == 0))
        print(f"run finished rc={rc}; model state={reg.get(rec.name).state}")
        sys.exit(rc)


def cmd_serve(reg: LocalModelRegistry, args):
    from .needs_sync_server import create_app
    app = create_app(reg, args.dataset, args.config, args.retrainInterval)
    app.run(host="0.0.0.0", port=args.port)


def main(argv=None):
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--registry", required=True)
    sub = p.add_subparsers(dest="cmd", required=True)

    g = sub.add_parser("get")
    g.add_argument("--dataset", required=True)
    g.set_defaults(fn=cmd_get)
    m = sub.add_parser("models")
    m.add_argument("--dataset", default=None)
    m.set_defaults(fn=cmd_models)
    d = sub.add_parser("deploy")
    d.add_argument("--name", required=True)
    d.set_defaults(fn=cmd_deploy)
    t = sub.add_parser("is-training")
    t.add_argument("--dataset", required=True)
    t.set_defaults(fn=cmd_is_training)
    tr = sub.add_parser("train")
    tr.add_argument("--dataset", required=True)
    tr.add_argument("--command", default=None)
    tr.set_defaults(fn=cmd_train)
    s = sub.add_parser("serve")
    s.add_argument("--dataset", required=True)
    s.add_argument("--config", required=True)
    s.add_argument("--retrainInterval", type=float, default=12.0)
    s.add_argument("--port", type=int, default=8090)
    s.set_defaults(fn=cmd_serve)

    args = p.parse_args(argv)
    reg = LocalModelRegistry(args.registry)
    args.fn(reg, args)


if __name__ == "__main__":
    main()
