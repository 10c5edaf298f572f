"""Hyper-parameter sweep runner (reference: Issue_Embeddings/hyperparam_sweep —
W&B random/grid/bayes sweeps over emb_sz/n_hid/n_layers/bs/bptt/lr/wd with
one agent per GPU, hp_runner.sh; 538 configs on 24 V100s).

Offline MI355X equivalent: a sweep spec (grid or random samples), one
worker process per GPU (HIP_VISIBLE_DEVICES pinning), results appended to
a shared JSONL leaderboard. No external tracking service.

  python -m code_intelligence_amd.train.sweep --spec sweep.yaml \
      --gpus 8 --trials 64 --out sweeps/run1
spec yaml:
  method: random            # random | grid
  parameters:
    emb_sz:   {values: [500, 800, 900]}
    n_hid:    {min: 1725, max: 3000, type: int}
    lr:       {min: 1.0e-4, max: 5.0e-3, log: true}
    one_cycle: {values: [true]}
"""
from __future__ import annotations

import argparse
import itertools
import json
import math
import os
import random
import subprocess
import sys
import time
from pathlib import Path
from typing import Dict, List

import yaml


def sample_space(spec: dict, n_trials: int, seed: int = 0) -> List[Dict]:
    params = spec.get("parameters", {})
    method = spec.get("method", "random")
    if method == "grid":
        keys = sorted(params)
        values = []
        for k in keys:
            p = params[k]
            if "values" not in p:
                raise ValueError(f"grid sweep needs 'values' for {k}")
            values.append(p["values"])
        combos = [dict(zip(keys, c)) for c in itertools.product(*values)]
        return combos[:n_trials] if n_trials else combos
    rng = random.Random(seed)
    out = []
    for _ in range(n_trials):
        cfg = {}
        for k, p in params.items():
            if "values" in p:
                cfg[k] = rng.choice(p["values"])
            else:
                lo, hi = p["min"], p["max"]
                if p.get("log"):
                    v = math.exp(rng.uniform(math.log(lo), math.log(hi)))
                else:
                    v = rng.uniform(lo, hi)
                cfg[k] = int(round(v)) if p.get("type") == "int" else v
        out.append(cfg)
    return out


def run_trial_inline(cfg: Dict, base_args: List[str], out_dir: Path,
                     trial_id: int, gpu: int | None = None) -> Dict:
    """One trial = one train-CLI subprocess (one agent per GPU, like the
    reference's hp_runner.sh)."""
    env = dict(os.environ)
    if gpu is not None:
        env["HIP_VISIBLE_DEVICES"] = str(gpu)
    args = [sys.executable, "-m", "code_intelligence_amd.train",
            "--model_path", str(out_dir / f"trial{trial_id}")] + base_args
    for k, v in cfg.items():
        args += [f"--{k}", str(v)]
    t0 = time.time()
    proc = subprocess.run(args, capture_output=True, text=True, env=env)
    result = {"trial": trial_id, "config": cfg, "gpu": gpu,
              "elapsed_s": round(time.time() - t0, 1),
              "returncode": proc.returncode}
    for line in reversed(proc.stdout.splitlines()):
        try:
            obj = json.loads(line)
            if "final" in obj:
                result["metrics"] = obj["final"]
                break
        except json.JSONDecodeError:
            continue
    return result


class SweepRunner:
    def __init__(self, spec: dict, out_dir, n_gpus: int = 1,
                 base_args: List[str] | None = None,
                 trial_fn=run_trial_inline):
        self.spec = spec
        self.out = Path(out_dir)
        self.out.mkdir(parents=True, exist_ok=True)
        self.n_gpus = max(1, n_gpus)
        self.base_args = base_args or []
        self.trial_fn = trial_fn
        self.board = self.out / "leaderboard.jsonl"

    def _record(self, result: Dict):
        with open(self.board, "a") as f:
            f.write(json.dumps(result) + "\n")

    def run(self, n_trials: int, seed: int = 0) -> List[Dict]:
        trials = sample_space(self.spec, n_trials, seed)
        results: List[Dict] = []
        # simple round-robin over GPUs, n_gpus trials in flight
        from concurrent.futures import ThreadPoolExecutor
        with ThreadPoolExecutor(max_workers=self.n_gpus) as pool:
            futs = []
            for i, cfg in enumerate(trials):
                futs.append(pool.submit(self.trial_fn, cfg, self.base_args,
                                        self.out, i, i % self.n_gpus))
            for fut in futs:
                r = fut.result()
                self._record(r)
                results.append(r)
        return results

    def best(self, metric: str = "valid_loss") -> Dict | None:
        rows = [json.loads(l) for l in open(self.board)] \
            if self.board.exists() else []
        rows = [r for r in rows if r.get("metrics", {}).get(metric) is not None]
        return min(rows, key=lambda r: r["metrics"][metric], default=None)


def main(argv=None):
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("--spec", required=True)
    p.add_argument("--out", required=True)
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--trials", type=int, default=8)
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--base", default="--data_path synthetic:500 --epochs 1",
                   help="extra train-CLI args common to all trials")
    args = p.parse_args(argv)
    spec = yaml.safe_load(open(args.spec))
    runner = SweepRunner(spec, args.out, n_gpus=args.gpus,
                         base_args=args.base.split())
    runner.run(args.trials, args.seed)
    best = runner.best()
    print(json.dumps({"best": best}, indent=2))


if __name__ == "__main__":
    main()
