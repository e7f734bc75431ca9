"""Supervisor: respawn failed pipeline stages, resume from bus offsets.

The reference's fault tolerance is container-level `restart: on-failure`
with `startingOffsets: latest` and NO checkpointing — a restarted stage
silently drops everything sent while it was down (SURVEY.md §5 failure
handling). This supervisor restarts crashed stage processes AND resumes
their consumers from persisted bus offsets, so no message is dropped
(at-least-once).

Usage:
    python -m tskd_amd.parallel.supervisor --stage processstream -- \
        --bus-dir /dev/shm/tskd_bus --speed 60
"""

from __future__ import annotations

import argparse
import json
import logging
import os
import subprocess
import sys
import time
from typing import Dict, List, Optional

log = logging.getLogger("supervisor")


def save_offsets(path: str, positions: Dict[str, int]) -> None:
    tmp = path + ".tmp"
    with open(tmp, "w") as f:
        json.dump(positions, f)
    os.replace(tmp, path)  # atomic


def load_offsets(path: str) -> Dict[str, int]:
    if not os.path.exists(path):
        return {}
    try:
        with open(path) as f:
            return {k: int(v) for k, v in json.load(f).items()}
    except (ValueError, OSError):
        return {}


def restore_consumer(consumer, path: str) -> int:
    """Seek a tskd Consumer to persisted offsets; returns #partitions."""
    pos = load_offsets(path)
    n = 0
    for key, off in pos.items():
        topic, _, part = key.rpartition("/")
        consumer.seek(topic, int(part), off)
        n += 1
    return n


class Supervisor:
    """Run a stage command, restart on failure (bounded backoff)."""

    def __init__(self, cmd: List[str], max_restarts: int = 10,
                 backoff_s: float = 0.5):
        self.cmd = cmd
        self.max_restarts = max_restarts
        self.backoff_s = backoff_s
        self.restarts = 0
        self.proc: Optional[subprocess.Popen] = None

    def run(self) -> int:
        while True:
            log.info("starting: %s", " ".join(self.cmd))
            self.proc = subprocess.Popen(self.cmd)
            rc = self.proc.wait()
            if rc == 0:
                return 0
            self.restarts += 1
            log.warning("stage exited rc=%d (restart %d/%d)", rc,
                        self.restarts, self.max_restarts)
            if self.restarts >= self.max_restarts:
                return rc
            time.sleep(self.backoff_s * min(self.restarts, 8))


STAGES = {
    "sendstream": "tskd_amd.cli.sendstream",
    "processstream": "tskd_amd.cli.processstream",
    "predictstream": "tskd_amd.cli.predictstream",
    "plotdata": "tskd_amd.cli.plotdata",
}


def main(argv=None) -> None:
    logging.basicConfig(level=logging.INFO,
                        format="%(asctime)s %(name)s %(levelname)s %(message)s")
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--stage", required=True, choices=sorted(STAGES))
    ap.add_argument("--max-restarts", type=int, default=10)
    ap.add_argument("rest", nargs=argparse.REMAINDER,
                    help="arguments passed to the stage (after --)")
    args = ap.parse_args(argv)
    rest = [a for a in args.rest if a != "--"]
    sup = Supervisor([sys.executable, "-m", STAGES[args.stage], *rest],
                     max_restarts=args.max_restarts)
    sys.exit(sup.run())


if __name__ == "__main__":
    main()
