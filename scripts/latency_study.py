#!/usr/bin/env python3
"""Serving-latency distribution: per-trigger wall time (sync after each
replay) for the fused hipGraph serving step at production stream counts."""
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch  # noqa: E402


def study(S, n_triggers=1000):
    from tskd_amd.engine import StreamEngine
    from tskd_amd.engine.stream_engine import TriggerGraph
    from tskd_amd.models import build_model
    from tskd_amd.ops import GraphedForward, MyCNNEngine
    fs = 125.0
    me = MyCNNEngine(build_model("MyCNN5").eval(), device="cuda")
    se = StreamEngine(S, 10, ring_grid=2048, fs=fs, device="cuda")
    raw = torch.randn(S, 8, int(fs * 60), device="cuda", dtype=torch.bfloat16)
    cm = list(range(8))
    while se.nproc == 0 or se.nproc < se.head - se.win_buckets + 1:
        se.ingest_dense(raw, chan_map=cm)
    torch.cuda.synchronize()
    gf = GraphedForward(me, s=S, n=1, dtype=torch.bfloat16, timelast=True,
                        capture=False)
    tg = TriggerGraph(se, raw, cm, gf, stride=12)
    for _ in range(20):
        tg.replay()
    torch.cuda.synchronize()
    lat = []
    for _ in range(n_triggers):
        t0 = time.perf_counter()
        tg.replay()
        torch.cuda.synchronize()
        lat.append((time.perf_counter() - t0) * 1e3)
    lat.sort()
    q = lambda p: lat[min(int(p * len(lat)), len(lat) - 1)]
    print(f"S={S}: p50={statistics.median(lat):.3f} ms  p99={q(0.99):.3f} ms "
          f"p99.9={q(0.999):.3f} ms  max={lat[-1]:.3f} ms  "
          f"({S/statistics.median(lat)*1000:.0f} windows/s/GPU)")


if __name__ == "__main__":
    assert torch.cuda.is_available()
    for S in (4096, 16384, 65536):
        study(S, 600 if S == 65536 else 1000)
