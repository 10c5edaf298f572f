"""Pipeline step 2: train the repo-specific MLP + per-label thresholds
from previously-saved issue embeddings (reference: repo_mlp.ipynb train
step of the KFP pipeline).

  python scripts/train_repo_mlp.py --org kubeflow --repo kubeflow \
      [--store /path/to/object_store]
"""
import sys
from pathlib import Path
sys.path.insert(0, str(Path(__file__).resolve().parents[1]))

import argparse
import json


def main():
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--org", required=True)
    p.add_argument("--repo", required=True)
    p.add_argument("--store", default=None)
    args = p.parse_args()
    from code_intelligence_amd.gh.gcs_util import ObjectStore
    from code_intelligence_amd.label.trainers import train_repo_mlp
    store = ObjectStore(root=args.store) if args.store else None
    out = train_repo_mlp(args.org, args.repo, store=store)
    print(json.dumps({k: v for k, v in out.items() if k != "thresholds"},
                     default=str))


if __name__ == "__main__":
    main()
