"""Bulk-embed a repo's archived issues to the object store
(reference: Get-GitHub-Issues.ipynb / issues_loader.py save_issue_embeddings).

  python scripts/embed_repo.py --org kubeflow --repo kubeflow \
      --model_path model_files --archive /path/to/archive \
      [--store /path/to/object_store] [--train-mlp]
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
    p.add_argument("--model_path", required=True)
    p.add_argument("--archive", required=True)
    p.add_argument("--store", default=None)
    p.add_argument("--bs", type=int, default=100)
    p.add_argument("--train-mlp", action="store_true",
                   help="also run pipeline step 2 (repo MLP + thresholds)")
    args = p.parse_args()
    from code_intelligence_amd.engine.inference import InferenceWrapper
    from code_intelligence_amd.gh.gcs_util import ObjectStore
    from code_intelligence_amd.label.trainers import (save_issue_embeddings,
                                                      train_repo_mlp)
    store = ObjectStore(root=args.store) if args.store else None
    wrapper = InferenceWrapper(model_path=args.model_path)
    uri = save_issue_embeddings(args.org, args.repo, wrapper, store=store,
                                archive_root=args.archive, bs=args.bs)
    out = {"embeddings": uri}
    if args.train_mlp:
        out["train"] = {k: v for k, v in train_repo_mlp(
            args.org, args.repo, store=store).items() if k != "thresholds"}
    print(json.dumps(out, default=str))


if __name__ == "__main__":
    main()
