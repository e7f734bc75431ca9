#!/usr/bin/env python3
"""Flagship serving benchmark: waveform windows/sec, MyCNN5, MI355X.

Measures the BASELINE.json headline metric — waveform windows/sec (whole
node) + p50 step latency for the MyCNN5 8-channel serving path — on
synthetic data (no network: synthetic 8-active-channel windows, random-init
weights; BASELINE.md: the reference publishes no quantitative numbers, so
vs_baseline is null).

One step = one serving macro-batch per GPU: S sequences x B=1024-window
batches through the fused HIP conv+LSTM+head+sigmoid path (reference
semantics per batch, incl. the LSTM batch-axis-as-time quirk).

Usage: python bench.py [--gpus N] [--steps K] [--warmup W]
  For N>1 the driver launches this under torch.distributed.run with one rank
  per GPU (RCCL over xGMI); we read RANK/LOCAL_RANK/WORLD_SIZE from the env.
"""

from __future__ import annotations

import argparse
import json
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch  # noqa: E402


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--seqs", type=int, default=2048,
                   help="concurrent patient sequences per GPU")
    p.add_argument("--batch", type=int, default=1024,
                   help="windows per sequence batch (reference batch semantics)")
    p.add_argument("--variant", default="MyCNN5")
    p.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    args = p.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    dist = None
    if world > 1:
        import torch.distributed as dist_mod
        dist = dist_mod
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        dist.init_process_group("nccl")
    assert torch.cuda.is_available(), "bench.py requires an MI355X"
    torch.cuda.set_device(local_rank)
    device = torch.device("cuda", local_rank)

    from tskd_amd.models import build_model
    from tskd_amd.ops import MyCNNEngine

    torch.manual_seed(1234 + rank)
    model = build_model(args.variant).eval()
    eng = MyCNNEngine(model, device=device)

    S, B = args.seqs, args.batch
    dtype = torch.bfloat16 if args.dtype == "bf16" else torch.float32
    # Synthetic 8-active-channel 125 Hz-derived windows (2 of the 10 wire
    # channels absent -> zero, as in the MIMIC numerics records).
    x = torch.randn(S, B, 10, 120, device=device, dtype=dtype)
    x[:, :, 8:, :] = 0
    age = torch.full((S, B), 65.0, device=device)

    def step() -> None:
        eng.forward(x, age, apply_sigmoid=True)

    for _ in range(args.warmup):
        step()
    torch.cuda.synchronize()
    if dist:
        dist.barrier()
    torch.cuda.synchronize()

    lat = []
    t0 = time.perf_counter()
    for _ in range(args.steps):
        s0 = time.perf_counter()
        step()
        torch.cuda.synchronize()
        lat.append(time.perf_counter() - s0)
    if dist:
        dist.barrier()
    torch.cuda.synchronize()
    t1 = time.perf_counter()

    elapsed = torch.tensor([t1 - t0], device=device, dtype=torch.float64)
    if dist:
        dist.all_reduce(elapsed, op=dist.ReduceOp.MAX)
    elapsed_s = float(elapsed.item())

    windows_per_step_per_gpu = S * B
    total_windows = windows_per_step_per_gpu * args.steps * world
    value = total_windows / elapsed_s
    ms_per_step = elapsed_s / args.steps * 1000.0

    if rank == 0:
        print(json.dumps({
            "metric": "waveform_windows_per_sec",
            "value": value,
            "unit": "windows/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "p50_step_ms": statistics.median(lat) * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": "synthetic 8-active-ch windows, random-init weights",
            "config": {
                "model": args.variant,
                "global_batch": windows_per_step_per_gpu * world,
                "seq_len": 120,
                "batch_per_seq": B,
                "seqs_per_gpu": S,
                "parallelism": f"dp{world}",
            },
        }), flush=True)

    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
