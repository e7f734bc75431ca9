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


def build_step(mode: str, device):
    from tskd_amd.models import build_model
    from tskd_amd.ops import MyCNNEngine
    torch.manual_seed(0)
    model = build_model("MyCNN5").eval()
    if mode == "train":
        from tskd_amd.train.hip_trainer import MyCNNHipTrainer
        tr = MyCNNHipTrainer(model, device=device, lr=1e-5)
        x = torch.randn(512, 64, 10, 120, device=device)
        age = torch.full((512, 64), 65.0, device=device)
        y = (torch.rand(512, 64, device=device) < 0.3).float()
        return lambda: tr.step(x, age, y)
    if mode == "infer":
        eng = MyCNNEngine(model, device=device)
        x = torch.randn(4096, 256, 10, 120, device=device,
                        dtype=torch.bfloat16)
        age = torch.full((4096, 256), 65.0, device=device)
        return lambda: eng.forward(x, age, apply_sigmoid=True)
    if mode == "pipeline":
        from tskd_amd.engine import StreamEngine
        eng = MyCNNEngine(model, device=device)
        se = StreamEngine(16384, 10, ring_grid=2048, fs=125.0, device=device)
        raw = torch.randn(16384, 8, 7500, device=device, dtype=torch.bfloat16)
        age = torch.full((16384, 1), 65.0, device=device)

        def step():
            se.ingest_dense(raw, chan_map=list(range(8)))
            w = se.windows(batch=1, stride=12, dtype=torch.bfloat16)
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
    args = ap.parse_args()
    assert torch.cuda.is_available()
    step = build_step(args.mode, "cuda")
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
