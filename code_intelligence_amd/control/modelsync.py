"""ModelSync reconciler (reference: Label_Microservice/go/controllers/
modelsync_controller.go — a kubebuilder operator reconciling a ModelSync
CRD into Tekton PipelineRuns).

Process-native re-design: a ``ModelSync`` object holds the spec
(needsSyncUrl + parameter name-mapping + a run template + history limits,
modelsync_types.go:29-51) and ``reconcile()`` performs one loop turn:

  1. classify child runs Running/Succeeded/Failed (controller:103-130)
  2. update status.active (132-147)
  3. GC old runs per history limits (159-195, webhook defaults 100)
  4. poll the needs-sync endpoint (197-225; requeue delay on error)
  5. if needsSync and no active run: construct a run ('name-uuid[:5]',
     parameter merge, 227-293) and start it (310-316)

Runs are subprocesses (or injected callables for tests) instead of Tekton
PipelineRuns; a ``reconcile_forever`` loop stands in for the manager."""
from __future__ import annotations

import dataclasses
import json
import logging
import subprocess
import time
import uuid
from typing import Callable, Dict, List, Optional

log = logging.getLogger(__name__)

DEFAULT_HISTORY_LIMIT = 100  # modelsync_webhook.go:49-56 defaulting


@dataclasses.dataclass
class ModelSyncSpec:
    name: str
    needs_sync_url: str
    run_command: List[str]                      # template; params appended
    parameter_mapping: Dict[str, str] = dataclasses.field(default_factory=dict)
    successful_runs_history_limit: int = DEFAULT_HISTORY_LIMIT
    failed_runs_history_limit: int = DEFAULT_HISTORY_LIMIT
    requeue_after_s: float = 60.0               # controller requeues 1 min on error


class PipelineRun:
    """One child run: a subprocess or an injected callable."""

    def __init__(self, name: str, command: List[str],
                 runner: Optional[Callable[[List[str]], int]] = None):
        self.name = name
        self.command = command
        self._proc: Optional[subprocess.Popen] = None
        self._result: Optional[int] = None
        if runner is not None:
            self._result = runner(command)
        else:
            self._proc = subprocess.Popen(command)

    @property
    def status(self) -> str:
        if self._result is None and self._proc is not None:
            rc = self._proc.poll()
            if rc is None:
                return "Running"
            self._result = rc
        return "Succeeded" if self._result == 0 else "Failed"


class ModelSync:
    def __init__(self, spec: ModelSyncSpec, session=None,
                 runner: Optional[Callable[[List[str]], int]] = None):
        self.spec = spec
        self.runner = runner
        if session is None:
            import requests
            session = requests.Session()
        self.session = session
        self.runs: List[PipelineRun] = []
        self.status: Dict = {"active": [], "succeeded": 0, "failed": 0}

    # --- one reconcile turn (controller.Reconcile, :76-326) --------------
    def reconcile(self) -> Dict:
        running = [r for r in self.runs if r.status == "Running"]
        succeeded = [r for r in self.runs if r.status == "Succeeded"]
        failed = [r for r in self.runs if r.status == "Failed"]

        # GC per history limits (oldest first), then report status
        for pool, limit in ((succeeded, self.spec.successful_runs_history_limit),
                            (failed, self.spec.failed_runs_history_limit)):
            while len(pool) > limit:
                victim = pool.pop(0)
                self.runs.remove(victim)
                log.info("gc'd run %s", victim.name)
        self.status["active"] = [r.name for r in running]
        self.status["succeeded"] = len(succeeded)
        self.status["failed"] = len(failed)

        try:
            needs, params = self._poll_needs_sync()
        except Exception as e:
            log.warning("needsSync poll failed (%s); requeue in %.0fs",
                        e, self.spec.requeue_after_s)
            return {**self.status, "requeue_after_s": self.spec.requeue_after_s}

        if not needs:
            return {**self.status, "needs_sync": False}
        if running:  # skip while a run is active (controller:296-298)
            return {**self.status, "needs_sync": True, "skipped": "active run"}

        run = self._construct_run(params)
        self.runs.append(run)
        log.info("created run %s: %s", run.name, run.command)
        return {**self.status, "needs_sync": True, "created": run.name}

    def _poll_needs_sync(self):
        r = self.session.get(self.spec.needs_sync_url)
        r.raise_for_status()
        data = r.json()
        return bool(data.get("needsSync")), data.get("parameters") or {}

    def _construct_run(self, parameters: Dict[str, str]) -> PipelineRun:
        name = f"{self.spec.name}-{uuid.uuid4().hex[:5]}"
        cmd = list(self.spec.run_command)
        for src, dst in self.spec.parameter_mapping.items():
            if src in parameters:
                cmd.append(f"--{dst}={parameters[src]}")
        return PipelineRun(name, cmd, runner=self.runner)

    def reconcile_forever(self, interval_s: float = 60.0,
                          stop: Optional[Callable[[], bool]] = None) -> None:
        while not (stop and stop()):
            self.reconcile()
            time.sleep(interval_s)
