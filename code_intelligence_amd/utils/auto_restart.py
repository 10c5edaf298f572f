"""Dev harness: run a command, restart it when watched files change or
when it dies (reference: py/code_intelligence/run_with_auto_restart.py,
used with skaffold file sync). The reference used the watchdog package;
this implementation polls mtimes (watchdog is not in the MI355X image).

  python -m code_intelligence_amd.utils.auto_restart \
      --watch code_intelligence_amd -- python -m code_intelligence_amd.label.worker
"""
from __future__ import annotations

import argparse
import logging
import subprocess
import sys
import time
from pathlib import Path
from typing import Dict, Iterable, List

log = logging.getLogger(__name__)


def snapshot(paths: Iterable[str], exts=(".py", ".yaml", ".yml", ".json")) -> Dict[str, float]:
    out: Dict[str, float] = {}
    for root in paths:
        p = Path(root)
        files = [p] if p.is_file() else [f for e in exts for f in p.rglob(f"*{e}")]
        for f in files:
            try:
                out[str(f)] = f.stat().st_mtime
            except OSError:
                pass
    return out


class AutoRestarter:
    def __init__(self, command: List[str], watch: List[str], poll_s: float = 1.0):
        self.command, self.watch, self.poll_s = command, watch, poll_s
        self.proc: subprocess.Popen | None = None
        self.restarts = 0

    def _start(self):
        log.info("starting: %s", self.command)
        self.proc = subprocess.Popen(self.command)

    def restart(self):
        if self.proc and self.proc.poll() is None:
            self.proc.terminate()
            try:
                self.proc.wait(timeout=10)
            except subprocess.TimeoutExpired:
                self.proc.kill()
        self.restarts += 1
        self._start()

    def run(self, max_iterations: int | None = None):
        state = snapshot(self.watch)
        self._start()
        it = 0
        while max_iterations is None or it < max_iterations:
            it += 1
            time.sleep(self.poll_s)
            new = snapshot(self.watch)
            if new != state:
                log.info("change detected; restarting")
                state = new
                self.restart()
            elif self.proc.poll() is not None:
                log.warning("process exited rc=%s; restarting", self.proc.returncode)
                self.restart()


def main():
    logging.basicConfig(level=logging.INFO)
    ap = argparse.ArgumentParser()
    ap.add_argument("--watch", action="append", default=[])
    ap.add_argument("--poll", type=float, default=1.0)
    ap.add_argument("command", nargs=argparse.REMAINDER)
    args = ap.parse_args()
    cmd = args.command
    if cmd and cmd[0] == "--":
        cmd = cmd[1:]
    if not cmd:
        sys.exit("no command given")
    AutoRestarter(cmd, args.watch or ["."], args.poll).run()


if __name__ == "__main__":
    main()
