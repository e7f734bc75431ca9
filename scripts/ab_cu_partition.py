#!/usr/bin/env python3
"""CU-partitioned ingest/model co-scheduling A/B (NOTES_R2 lever #8,
VERDICT r1 #6).

Round-1 finding: two-stream overlap (ingest T+1 over model chain T) was
NEUTRAL because every stage fills all 256 CUs — the streams serialize.
Round-2 hypothesis: the ingest kernel is HBM-bound and (per the round-1
read probe) saturates bandwidth from ~2048 blocks, so capping its grid
(TSKD_INGEST_GRID) leaves wave slots free and lets the model chain
actually co-schedule. Win condition: step -> max(ingest, model) instead
of ingest + model (~1.31 + 0.56 ms at S=65536).

Within-process sequential configs on one box (cross-run noise is ±3-5%;
expected effects are >10%). Each config captures a FRESH TriggerGraph
(the env knob is baked into the captured launch), replays N triggers
timed, then frees everything.

Usage: python scripts/ab_cu_partition.py [--streams 65536] [--reps 60]
"""
from __future__ import annotations

import argparse
import json
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402


def run_config(S: int, reps: int, overlap: bool, grid: int | None) -> dict:
    from tskd_amd.engine import StreamEngine
    from tskd_amd.engine.stream_engine import TriggerGraph
    from tskd_amd.models import build_model
    from tskd_amd.ops import GraphedForward, MyCNNEngine
    if grid is None:
        os.environ.pop("TSKD_INGEST_GRID", None)
    else:
        os.environ["TSKD_INGEST_GRID"] = str(grid)
    fs = 125.0
    torch.manual_seed(3)
    me = MyCNNEngine(build_model("MyCNN5").eval(), device="cuda")
    se = StreamEngine(S, 10, ring_grid=2048, fs=fs, device="cuda")
    raw = torch.randn(S, 8, int(fs * 60), device="cuda", dtype=torch.bfloat16)
    cm = list(range(8))
    while se.nproc == 0 or se.nproc < se.head - se.win_buckets + 1:
        se.ingest_dense(raw, chan_map=cm)
    torch.cuda.synchronize()
    gf = GraphedForward(me, s=S, n=1, dtype=torch.bfloat16, timelast=True,
                        capture=False)
    tg = TriggerGraph(se, raw, cm, gf, stride=12, overlap=overlap)
    for _ in range(10):
        out = tg.replay()
    torch.cuda.synchronize()
    probs_ref = out.clone()
    # THROUGHPUT: back-to-back replays, sync only at the ends — a sync per
    # replay would serialize ingest(T+1) behind model(T) and erase the
    # overlap being measured
    t0 = time.perf_counter()
    for _ in range(reps):
        tg.replay()
    torch.cuda.synchronize()
    total = time.perf_counter() - t0
    # LATENCY: synced single-trigger probe (serialized by construction)
    lat = []
    for _ in range(20):
        s0 = time.perf_counter()
        tg.replay()
        torch.cuda.synchronize()
        lat.append((time.perf_counter() - s0) * 1e3)
    # parity: same raw every trigger -> steady-state output must be stable
    err = (tg.replay() - probs_ref).abs().max().item()
    torch.cuda.synchronize()
    lat.sort()
    rec = {
        "overlap": overlap, "ingest_grid": grid or "default(32768)",
        "interval_ms": round(total / reps * 1e3, 4),
        "windows_per_s_M": round(S * reps / total / 1e6, 2),
        "latency_p50_ms": round(statistics.median(lat), 4),
        "out_drift": err,
    }
    del tg, gf, se, me, raw
    import gc
    gc.collect()
    torch.cuda.empty_cache()
    return rec


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--streams", type=int, default=65536)
    ap.add_argument("--reps", type=int, default=60)
    args = ap.parse_args()
    assert torch.cuda.is_available()
    if os.environ.get("TSKD_AB_FINE"):
        # round-2 fine sweep after the coarse result (2048 best at +6.5%):
        # bracket the ingest-BW/model-slot tradeoff
        configs = [
            ("baseline serial", False, None),
            ("overlap grid=1792", True, 1792),
            ("overlap grid=2048", True, 2048),
            ("overlap grid=2304", True, 2304),
            ("overlap grid=2560", True, 2560),
            ("overlap grid=3072", True, 3072),
            ("overlap grid=3584", True, 3584),
            ("overlap grid=2048 (repeat)", True, 2048),
            ("baseline serial (repeat)", False, None),
        ]
    else:
        configs = [
            ("baseline serial", False, None),
            ("overlap full-grid (r1 neutral)", True, None),
            ("overlap grid=8192", True, 8192),
            ("overlap grid=4096", True, 4096),
            ("overlap grid=2048", True, 2048),
            ("overlap grid=1024", True, 1024),
            ("serial grid=2048 (BW check)", False, 2048),
            ("baseline serial (repeat)", False, None),
        ]
    out = []
    for name, ov, grid in configs:
        rec = run_config(args.streams, args.reps, ov, grid)
        rec["name"] = name
        out.append(rec)
        print(json.dumps(rec), flush=True)


if __name__ == "__main__":
    main()
