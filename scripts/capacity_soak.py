#!/usr/bin/env python3
"""Ring-buffer capacity soak at BASELINE config-4 scale.

Allocates ~200 GB of per-patient ring buffers on ONE MI355X (S=65536
streams x C=10 channels x G=24576 grid points x 12 B = 193 GB bucket+proc
state, plus the 7.9 GB raw trigger buffer), then soaks the fused
TriggerGraph across MULTIPLE ring wraps while a CPU mirror engine (stream
0 only, same G) checks window parity every K triggers — wrap correctness
at capacity, not just at the 16 GB bench shape.

Memory plan per GPU (288 GB HBM3E):
  bucket (sum,cnt) packed f32x2 : S*C*G*8  B
  processed grid f32            : S*C*G*4  B
  raw trigger (S,8,7500) bf16   : 7.9 GB
  windows (S,1,120,10) bf16     : 0.16 GB
  model/graph workspace         : < 1 GB
At G=24576: 193.3 + 8.1 = ~201 GB (70% of HBM), ring retention =
24576 grid points x 5 s = 34.1 h of history per stream.

Usage: python scripts/capacity_soak.py [--streams 65536] [--grid 24576]
       [--triggers 2600] [--check-every 200]
"""
from __future__ import annotations

import argparse
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--streams", type=int, default=65536)
    ap.add_argument("--grid", type=int, default=24576)
    ap.add_argument("--triggers", type=int, default=2600)
    ap.add_argument("--check-every", type=int, default=200)
    args = ap.parse_args()
    assert torch.cuda.is_available()
    from tskd_amd.engine import StreamEngine
    from tskd_amd.engine.stream_engine import TriggerGraph
    from tskd_amd.models import build_model
    from tskd_amd.ops import GraphedForward, MyCNNEngine

    S, G = args.streams, args.grid
    fs = 125.0
    ring_gb = S * 10 * G * 12 / 2**30
    print(f"[capacity] S={S} G={G}: ring state {ring_gb:.1f} GiB "
          f"({G * 5 / 3600:.1f} h retention/stream)", flush=True)
    t0 = time.perf_counter()
    me = MyCNNEngine(build_model("MyCNN5").eval(), device="cuda")
    se = StreamEngine(S, 10, ring_grid=G, fs=fs, device="cuda")
    torch.cuda.synchronize()
    print(f"[capacity] alloc+zero in {time.perf_counter() - t0:.2f}s; "
          f"reserved {torch.cuda.memory_reserved() / 2**30:.1f} GiB",
          flush=True)
    torch.manual_seed(7)
    raw = torch.randn(S, 8, int(fs * 60), device="cuda", dtype=torch.bfloat16)
    cm = list(range(8))
    # CPU mirror of stream 0: same G, fed the same raw row each trigger
    mirror = StreamEngine(1, 10, ring_grid=G, fs=fs, device="cpu")
    raw0 = raw[0:1].float().cpu()

    def check_parity(tag: str) -> float:
        assert se.nproc == mirror.nproc, (se.nproc, mirror.nproc)
        w_gpu = se.windows(batch=4, stride=12, dtype=torch.float32)[0]
        w_cpu = mirror.windows(batch=4, stride=12, dtype=torch.float32)[0]
        err = (w_gpu.cpu() - w_cpu).abs().max().item()
        assert err < 1e-3, f"{tag}: window mismatch {err}"
        return err

    # warm to steady state (eager triggers) with the mirror in lockstep
    while se.nproc == 0 or se.nproc < se.head - se.win_buckets + 1:
        se.ingest_dense(raw, chan_map=cm)
        mirror.ingest_dense(raw0, chan_map=cm)
    torch.cuda.synchronize()
    err0 = check_parity("steady-state")
    gf = GraphedForward(me, s=S, n=1, dtype=torch.bfloat16, timelast=True,
                        capture=False)
    tg = TriggerGraph(se, raw, cm, gf, stride=12)
    mirror.ingest_dense(raw0, chan_map=cm)  # TriggerGraph warm run = 1 trigger

    mem0 = torch.cuda.memory_allocated()
    lat = []
    errs = [err0]
    t0 = time.perf_counter()
    for i in range(args.triggers):
        s0 = time.perf_counter()
        out = tg.replay()
        torch.cuda.synchronize()
        lat.append((time.perf_counter() - s0) * 1e3)
        mirror.ingest_dense(raw0, chan_map=cm)  # untimed (host mirror)
        if (i + 1) % args.check_every == 0:
            assert torch.isfinite(out).all(), f"non-finite at {i}"
            mem = torch.cuda.memory_allocated()
            assert mem <= mem0 + (64 << 20), f"memory grew {mem0}->{mem}"
            errs.append(check_parity(f"trigger {i + 1}"))
            print(f"[capacity] {i + 1}/{args.triggers} triggers, "
                  f"{(i + 1) * 12 / G:.2f} ring turns, window err "
                  f"{errs[-1]:.2e}, p50 {statistics.median(lat):.2f} ms",
                  flush=True)
    gpu_s = sum(lat) / 1e3  # GPU trigger time only (mirror is untimed)
    lat.sort()
    turns = args.triggers * 12 / G
    dt = gpu_s
    print(f"[capacity] DONE: {args.triggers} triggers ({turns:.2f} ring "
          f"turns at {ring_gb:.0f} GiB), p50 "
          f"{statistics.median(lat):.3f} ms p99 "
          f"{lat[int(0.99 * len(lat))]:.3f} ms max {lat[-1]:.3f} ms; "
          f"max window err vs CPU mirror {max(errs):.2e}; "
          f"throughput {S * args.triggers / dt / 1e6:.1f} M windows/s",
          flush=True)


if __name__ == "__main__":
    main()
