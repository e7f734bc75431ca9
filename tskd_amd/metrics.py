"""Per-stage observability: latency/throughput counters.

The reference has only AverageMeter wall-clock prints and the Spark web UI
(SURVEY.md §5 tracing). Here every pipeline stage can report structured
counters — windows/s, messages/s, p50/p99 stage latency — which are also the
benchmark outputs (BASELINE.json: whole-node windows/s + p50 end-to-end
latency).
"""

from __future__ import annotations

import json
import time
from collections import deque
from typing import Dict, Optional


class StageTimer:
    """Rolling latency/throughput stats for one pipeline stage."""

    def __init__(self, name: str, window: int = 512):
        self.name = name
        self.lat = deque(maxlen=window)
        self.items = 0
        self.calls = 0
        self.t_start = time.time()
        self._t0: Optional[float] = None

    def __enter__(self):
        self._t0 = time.perf_counter()
        return self

    def __exit__(self, *exc):
        self.lat.append(time.perf_counter() - self._t0)
        self.calls += 1
        return False

    def add_items(self, n: int) -> None:
        self.items += n

    def _pct(self, p: float) -> float:
        if not self.lat:
            return 0.0
        xs = sorted(self.lat)
        return xs[min(int(p * len(xs)), len(xs) - 1)]

    def snapshot(self) -> Dict:
        elapsed = max(time.time() - self.t_start, 1e-9)
        return {
            "stage": self.name,
            "calls": self.calls,
            "items": self.items,
            "items_per_sec": self.items / elapsed,
            "p50_ms": self._pct(0.50) * 1e3,
            "p99_ms": self._pct(0.99) * 1e3,
            "last_ms": (self.lat[-1] * 1e3) if self.lat else 0.0,
        }

    def log_line(self) -> str:
        return json.dumps(self.snapshot())


class PipelineMetrics:
    """A registry of stage timers with one-line JSON reporting."""

    def __init__(self):
        self.stages: Dict[str, StageTimer] = {}

    def stage(self, name: str) -> StageTimer:
        if name not in self.stages:
            self.stages[name] = StageTimer(name)
        return self.stages[name]

    def report(self) -> str:
        return json.dumps({k: v.snapshot() for k, v in self.stages.items()})


def prometheus_text(timers) -> str:
    """Render StageTimer snapshots in the Prometheus text exposition format
    (the reference has no scrapeable metrics — SURVEY.md §5 observability).
    `timers` is an iterable of StageTimer (or a PipelineMetrics)."""
    if isinstance(timers, PipelineMetrics):
        timers = timers.stages.values()
    lines = [
        "# HELP tskd_stage_calls_total Stage invocations.",
        "# TYPE tskd_stage_calls_total counter",
        "# HELP tskd_stage_items_total Items (windows/messages) processed.",
        "# TYPE tskd_stage_items_total counter",
        "# HELP tskd_stage_latency_ms Stage latency quantiles (rolling).",
        "# TYPE tskd_stage_latency_ms gauge",
    ]
    for t in timers:
        snap = t.snapshot()
        lbl = f'stage="{snap["stage"]}"'
        lines.append(f'tskd_stage_calls_total{{{lbl}}} {snap["calls"]}')
        lines.append(f'tskd_stage_items_total{{{lbl}}} {snap["items"]}')
        for q in ("p50", "p99"):
            lines.append(f'tskd_stage_latency_ms{{{lbl},quantile="{q}"}} '
                         f'{snap[q + "_ms"]:.6f}')
    return "\n".join(lines) + "\n"
