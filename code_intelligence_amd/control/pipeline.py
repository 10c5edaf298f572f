"""Multi-step training pipeline runner — the process-native equivalent of
the reference's KFP training pipeline (Training_Pipeline.ipynb: a 2-step
``@dsl.pipeline`` scrape-issues -> train, each step ``set_gpu_limit(1)``,
compiled and submitted to kfp). Here a pipeline is a YAML document of
ordered steps with ``{param}`` templating, executed as subprocesses (or
injected runners for tests) with fail-fast semantics and a JSON status
record per run — the artifact ModelSync triggers instead of a Tekton
PipelineRun.

Pipeline YAML shape (deploy/pipelines/scrape_train_pipeline.yaml):

  name: scrape-train
  params: {org: kubeflow, repo: examples, out: /srv/ci/models}
  steps:
    - name: scrape
      command: [python, scripts/embed_repo.py, "--org", "{org}", ...]
    - name: train
      command: [python, -m, code_intelligence_amd.label.trainers, ...]
"""
from __future__ import annotations

import dataclasses
import json
import subprocess
import time
import uuid
from pathlib import Path
from typing import Callable, Dict, List, Optional

import yaml

__all__ = ["PipelineSpec", "PipelineRunner", "load_pipeline"]


@dataclasses.dataclass
class Step:
    name: str
    command: List[str]


@dataclasses.dataclass
class PipelineSpec:
    name: str
    steps: List[Step]
    params: Dict[str, str] = dataclasses.field(default_factory=dict)


def load_pipeline(path) -> PipelineSpec:
    doc = yaml.safe_load(Path(path).read_text())
    steps = [Step(name=s["name"], command=[str(c) for c in s["command"]])
             for s in doc["steps"]]
    return PipelineSpec(name=doc["name"], steps=steps,
                        params={k: str(v)
                                for k, v in (doc.get("params") or {}).items()})


class PipelineRunner:
    """Execute a pipeline's steps in order, fail-fast, with a JSON status
    record (KFP run page equivalent) under ``run_dir``."""

    def __init__(self, spec: PipelineSpec, run_dir=None,
                 runner: Optional[Callable[[List[str]], int]] = None):
        self.spec = spec
        self.run_id = f"{spec.name}-{uuid.uuid4().hex[:5]}"
        self.run_dir = Path(run_dir) if run_dir else None
        self._runner = runner

    def _render(self, cmd: List[str], params: Dict[str, str]) -> List[str]:
        merged = {**self.spec.params, **params}
        try:
            return [c.format(**merged) for c in cmd]
        except KeyError as e:
            raise ValueError(f"pipeline param {e} not provided") from e

    def run(self, **params) -> dict:
        record = {"run_id": self.run_id, "pipeline": self.spec.name,
                  "params": {**self.spec.params,
                             **{k: str(v) for k, v in params.items()}},
                  "steps": [], "status": "Succeeded"}
        t0 = time.time()
        for step in self.spec.steps:
            cmd = self._render(step.command, record["params"])
            ts = time.time()
            if self._runner is not None:
                rc = self._runner(cmd)
            else:
                rc = subprocess.call(cmd)
            record["steps"].append({
                "name": step.name, "command": cmd, "returncode": rc,
                "seconds": round(time.time() - ts, 3),
                "status": "Succeeded" if rc == 0 else "Failed"})
            if rc != 0:           # fail-fast: later steps never start
                record["status"] = "Failed"
                break
        record["seconds"] = round(time.time() - t0, 3)
        if self.run_dir is not None:
            self.run_dir.mkdir(parents=True, exist_ok=True)
            (self.run_dir / f"{self.run_id}.json").write_text(
                json.dumps(record, indent=1))
        return record


def main(argv=None):
    import argparse
    p = argparse.ArgumentParser(description=__doc__)
    p.add_argument("pipeline", help="pipeline YAML path")
    p.add_argument("--run_dir", default="pipeline_runs")
    p.add_argument("--param", action="append", default=[],
                   help="k=v override (repeatable)")
    args = p.parse_args(argv)
    spec = load_pipeline(args.pipeline)
    overrides = dict(kv.split("=", 1) for kv in args.param)
    rec = PipelineRunner(spec, run_dir=args.run_dir).run(**overrides)
    print(json.dumps({"run_id": rec["run_id"], "status": rec["status"]}))
    return 0 if rec["status"] == "Succeeded" else 1


if __name__ == "__main__":
    raise SystemExit(main())
