"""Train the universal 3-kind (bug/feature/question) classifier from the
issue archive and save its artifacts (reference: the universal Keras model
training; serving loads via UNIVERSAL_MODEL_PATH).

  python scripts/train_universal.py --org kubeflow --archive /srv/ci/archive \
      --out /srv/ci/universal [--epochs 8] [--prefix kind/]
"""
import sys
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import argparse
import json


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--org", required=True)
    p.add_argument("--archive", required=True)
    p.add_argument("--out", required=True)
    p.add_argument("--epochs", type=int, default=8)
    p.add_argument("--lr", type=float, default=1e-3)
    p.add_argument("--prefix", default="", help="e.g. 'kind/'")
    p.add_argument("--device", default="cpu")
    args = p.parse_args()
    from code_intelligence_amd.label.trainers import train_universal_model
    model = train_universal_model(args.org, archive_root=args.archive,
                                  epochs=args.epochs, lr=args.lr,
                                  prefix=args.prefix, device=args.device)
    model.save(args.out)
    print(json.dumps({"saved": args.out,
                      "thresholds": model.thresholds,
                      "vocab": len(model.vocab)}))


if __name__ == "__main__":
    main()
