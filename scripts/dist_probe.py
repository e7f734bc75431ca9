#!/usr/bin/env python3
"""RCCL world>1 feasibility + numerics probe (VERDICT round-1 item #1).

Launched under torchrun with world_size ranks. On a 1-GPU box all ranks
share cuda:0 (RCCL permitting); on an 8-GPU node each rank takes its
LOCAL_RANK device. Verifies:
  - init_process_group("nccl") succeeds at world>1 on real hardware
  - all_gather_predictions returns every rank's vector bit-exactly
  - all_reduce MAX (the bench's elapsed-time reduction) is correct
  - per-rank HBM footprint of a production-shape DPServing step
Prints one JSON line per rank to stdout (rank tag included).
"""
from __future__ import annotations

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist


def main() -> None:
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local = int(os.environ.get("LOCAL_RANK", "0"))
    n_dev = torch.cuda.device_count()
    dev_idx = local % max(n_dev, 1)
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    from tskd_amd.parallel.dist import all_gather_predictions, pick_backend
    backend = os.environ.get("PROBE_BACKEND") or pick_backend(world)
    t0 = time.perf_counter()
    dist.init_process_group(backend, rank=rank, world_size=world)
    torch.cuda.set_device(dev_idx)
    device = torch.device("cuda", dev_idx)
    init_s = time.perf_counter() - t0

    S = int(os.environ.get("PROBE_STREAMS", "4096"))
    torch.manual_seed(100 + rank)
    probs = torch.rand(S, device=device)
    t0 = time.perf_counter()
    gathered = all_gather_predictions(probs)
    torch.cuda.synchronize()
    ag_ms = (time.perf_counter() - t0) * 1e3
    # every rank regenerates every other rank's vector to check bit-exactness
    ok = True
    for r in range(world):
        torch.manual_seed(100 + r)
        expect = torch.rand(S, device=device)
        ok = ok and bool(torch.equal(gathered[r], expect))

    red_dev = device if backend == "nccl" else "cpu"
    t = torch.tensor([float(rank + 1)], dtype=torch.float64, device=red_dev)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    ok = ok and (float(t.item()) == float(world))

    mem0 = torch.cuda.memory_allocated(device)
    step_ms = None
    if os.environ.get("PROBE_FULL", "0") == "1":
        # one production-shape serving trigger per rank (ring alloc + step)
        from tskd_amd.parallel.dist import DPServing
        Sfull = int(os.environ.get("PROBE_FULL_STREAMS", "16384"))
        srv = DPServing(Sfull, device=f"cuda:{dev_idx}")
        raw = torch.randn(Sfull, 8, 7500, device=device, dtype=torch.bfloat16)
        for _ in range(3):
            out = srv.step(raw, chan_map=list(range(8)))
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        out = srv.step(raw, chan_map=list(range(8)))
        torch.cuda.synchronize()
        step_ms = (time.perf_counter() - t0) * 1e3
        ok = ok and out.shape == (world, Sfull) and bool(
            torch.isfinite(out).all())

    print(json.dumps({
        "probe": "rccl_world_gt1",
        "backend": backend,
        "rank": rank, "world": world, "device": dev_idx,
        "n_visible_devices": n_dev,
        "init_s": round(init_s, 3),
        "all_gather_ms": round(ag_ms, 3),
        "numerics_ok": ok,
        "mem_alloc_mb": round(mem0 / 2**20, 1),
        "mem_reserved_mb": round(
            torch.cuda.memory_reserved(device) / 2**20, 1),
        "full_step_ms": step_ms,
    }), flush=True)
    dist.barrier()
    dist.destroy_process_group()
    if not ok:
        sys.exit(1)


if __name__ == "__main__":
    main()
