#!/usr/bin/env python3
"""Within-probe interleaved A/B benchmark harness.

Cross-run bench noise on this pool is ±3-5%, so any perf claim under ~10%
needs both variants interleaved in ONE process (guide §5.4 rules 13/24).
Variants are expressed as env knobs read by the kernel launchers at launch
time (e.g. TSKD_CONVBWD_GRID), so they can flip per round without reload.

Usage:
  python scripts/ab_bench.py --mode train --knob TSKD_CONVBWD_GRID \
      --a 2048 --b 8192 --rounds 8 --steps 8
Reports per-variant median/min step time and the median delta.
"""

import argparse
import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402


def build_step(mode: str, device, seqs: int, batch: int, streams: int):
    # NOTE: A/B verdicts are SHAPE-SPECIFIC (the conv WG-vs-1wave schedules
    # measured equal at 4096x256 and 17% apart at 8192x1024). Default
    # shapes here match bench.py's production shapes; override with
    # --seqs/--batch/--streams only to study scaling.
    from tskd_amd.models import build_model
    from tskd_amd.ops import MyCNNEngine
    torch.manual_seed(0)
    model = build_model("MyCNN5").eval()
    if mode == "train":
        from tskd_amd.train.hip_trainer import MyCNNHipTrainer
        S = seqs or 1024
        tr = MyCNNHipTrainer(model, device=device, lr=1e-5)
        x = torch.randn(S, 64, 10, 120, device=device)
        age = torch.full((S, 64), 65.0, device=device)
        y = (torch.rand(S, 64, device=device) < 0.3).float()
        return lambda: tr.step(x, age, y)
    if mode == "infer":
        from tskd_amd.ops import alloc_windows
        S, B = seqs or 8192, batch or 1024
        eng = MyCNNEngine(model, device=device)
        x = alloc_windows(S, B, 10, timelast=True, dtype=torch.bfloat16,
                          device=device)
        x.copy_(torch.randn(S, B, 120, 10, device=device,
                            dtype=torch.bfloat16))
        age = torch.full((S, B), 65.0, device=device)
        return lambda: eng.forward(x, age, apply_sigmoid=True)
    if mode == "pipeline":
        from tskd_amd.engine import StreamEngine
        S = streams or 16384
        eng = MyCNNEngine(model, device=device)
        se = StreamEngine(S, 10, ring_grid=2048, fs=125.0, device=device)
        raw = torch.randn(S, 8, 7500, device=device, dtype=torch.bfloat16)
        age = torch.full((S, 1), 65.0, device=device)

        def step():
            se.ingest_dense(raw, chan_map=list(range(8)))
            # timelast: the serving engine's production window layout
            w = se.windows(batch=1, stride=12, dtype=torch.bfloat16,
                           timelast=True)
            eng.forward(w, age, apply_sigmoid=True)
        return step
    raise ValueError(mode)


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--mode", default="train",
                    choices=["train", "infer", "pipeline"])
    ap.add_argument("--knob", required=True)
    ap.add_argument("--a", required=True)
    ap.add_argument("--b", required=True)
    ap.add_argument("--rounds", type=int, default=8)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=6)
    ap.add_argument("--seqs", type=int, default=0)
    ap.add_argument("--batch", type=int, default=0)
    ap.add_argument("--streams", type=int, default=0)
    args = ap.parse_args()
    assert torch.cuda.is_available()
    step = build_step(args.mode, "cuda", args.seqs, args.batch, args.streams)
    for _ in range(args.warmup):
        step()
    torch.cuda.synchronize()

    times = {"A": [], "B": []}
    for r in range(args.rounds):
        for tag, val in (("A", args.a), ("B", args.b)):
            os.environ[args.knob] = str(val)
            step()  # one unmeasured step after the knob flip
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(args.steps):
                step()
            torch.cuda.synchronize()
            times[tag].append((time.perf_counter() - t0) / args.steps)
    ma, mb = statistics.median(times["A"]), statistics.median(times["B"])
    print(f"A ({args.knob}={args.a}): median {ma*1e3:.4f} ms  "
          f"min {min(times['A'])*1e3:.4f} ms")
    print(f"B ({args.knob}={args.b}): median {mb*1e3:.4f} ms  "
          f"min {min(times['B'])*1e3:.4f} ms")
    print(f"delta (B vs A): {(mb/ma-1)*100:+.2f}% median  "
          f"[rounds={args.rounds}, steps={args.steps}]")


if __name__ == "__main__":
    main()
