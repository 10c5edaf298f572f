"""Local model registry — the offline stand-in for GCP AutoML that the
reference's Go control plane talks to (Label_Microservice/go/cmd/automl/
pkg/automl/automl.go: GetLatestDeployed/GetLatestTrained 83-120,
DeployModel 155-174, GetModelEvaluation picks P/R at 0.5 confidence
177-234, IsTraining 291-363).

Registry state is a directory of JSON records (one per model version), so
training pipelines, the needs-sync server and the reconciler share it
without any cloud dependency."""
from __future__ import annotations

import dataclasses
import datetime
import json
import uuid
from pathlib import Path
from typing import List, Optional


def _now() -> str:
    return datetime.datetime.now(datetime.timezone.utc).isoformat()


@dataclasses.dataclass
class ModelRecord:
    name: str                  # model id, e.g. 'issue-label-20240101-abcde'
    dataset: str               # logical model family, e.g. 'kubeflow-labels'
    create_time: str
    state: str = "trained"     # training | trained | failed
    deployed: bool = False
    evaluation: Optional[dict] = None  # {'precision': ..., 'recall': ..., 'confidence': 0.5}
    artifact_uri: Optional[str] = None

    def to_json(self) -> dict:
        return dataclasses.asdict(self)

    @classmethod
    def from_json(cls, d: dict) -> "ModelRecord":
        return cls(**d)


class LocalModelRegistry:
    def __init__(self, root):
        self.root = Path(root)
        self.root.mkdir(parents=True, exist_ok=True)

    def _path(self, name: str) -> Path:
        return self.root / f"{name}.json"

    def put(self, record: ModelRecord) -> None:
        self._path(record.name).write_text(json.dumps(record.to_json()))

    def get(self, name: str) -> Optional[ModelRecord]:
        p = self._path(name)
        return ModelRecord.from_json(json.loads(p.read_text())) if p.exists() else None

    def list(self, dataset: Optional[str] = None) -> List[ModelRecord]:
        out = []
        for f in sorted(self.root.glob("*.json")):
            r = ModelRecord.from_json(json.loads(f.read_text()))
            if dataset is None or r.dataset == dataset:
                out.append(r)
        return out

    # --- automl.go-equivalent operations -------------------------------
    def create_training(self, dataset: str, artifact_uri: Optional[str] = None
                        ) -> ModelRecord:
        name = f"{dataset}-{datetime.datetime.utcnow():%Y%m%d%H%M%S}-{uuid.uuid4().hex[:5]}"
        rec = ModelRecord(name=name, dataset=dataset, create_time=_now(),
                          state="training", artifact_uri=artifact_uri)
        self.put(rec)
        return rec

    def finish_training(self, name: str, evaluation: Optional[dict] = None,
                        ok: bool = True) -> None:
        rec = self.get(name)
        rec.state = "trained" if ok else "failed"
        rec.evaluation = evaluation
        self.put(rec)

    def latest_trained(self, dataset: str) -> Optional[ModelRecord]:
        trained = [r for r in self.list(dataset) if r.state == "trained"]
        return max(trained, key=lambda r: r.create_time, default=None)

    def latest_deployed(self, dataset: str) -> Optional[ModelRecord]:
        dep = [r for r in self.list(dataset) if r.deployed]
        return max(dep, key=lambda r: r.create_time, default=None)

    def deploy(self, name: str) -> ModelRecord:
        rec = self.get(name)
        if rec is None:
            raise KeyError(name)
        for other in self.list(rec.dataset):
            if other.deployed and other.name != name:
                other.deployed = False
                self.put(other)
        rec.deployed = True
        self.put(rec)
        return rec

    def is_training(self, dataset: str) -> bool:
        return any(r.state == "training" for r in self.list(dataset))

    def evaluation_at_confidence(self, name: str, confidence: float = 0.5
                                 ) -> Optional[dict]:
        """automl.go GetModelEvaluation semantics: the P/R row at 0.5."""
        rec = self.get(name)
        if rec is None or not rec.evaluation:
            return None
        ev = rec.evaluation
        if isinstance(ev, list):  # confidence-indexed rows
            for row in ev:
                if abs(row.get("confidence", -1) - confidence) < 1e-6:
                    return row
            return None
        return ev
