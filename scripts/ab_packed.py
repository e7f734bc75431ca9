#!/usr/bin/env python3
"""Within-process A/B of the packed (sum,cnt) bucket layout.

The layout is chosen at StreamEngine construction, so the generic knob
harness (ab_bench) can't flip it per round — instead both engines are
built once (packed and split) and the SAME serving step runs on each in
interleaved rounds.
"""

import os
import statistics
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402


def main(S=None, rounds=None, steps=None) -> None:
    import argparse
    ap = argparse.ArgumentParser()
    ap.add_argument("--streams", type=int, default=16384)
    ap.add_argument("--rounds", type=int, default=7)
    ap.add_argument("--steps", type=int, default=8)
    a = ap.parse_args()
    S, rounds, steps = a.streams, a.rounds, a.steps
    from tskd_amd.engine import StreamEngine
    from tskd_amd.models import build_model
    from tskd_amd.ops import MyCNNEngine
    assert torch.cuda.is_available()
    torch.manual_seed(0)
    me = MyCNNEngine(build_model("MyCNN5").eval(), device="cuda")
    raw = torch.randn(S, 8, 7500, device="cuda", dtype=torch.bfloat16)
    age = torch.full((S, 1), 65.0, device="cuda")
    engines = {}
    for tag, env in (("split", "0"), ("packed", "1")):  # split FIRST

        os.environ["TSKD_PACKED_BUCKETS"] = env
        engines[tag] = StreamEngine(S, 10, ring_grid=2048, fs=125.0,
                                    device="cuda")

    def step(se):
        se.ingest_dense(raw, chan_map=list(range(8)))
        w = se.windows(batch=1, stride=12, dtype=torch.bfloat16,
                       timelast=True)
        me.forward(w, age, apply_sigmoid=True)

    for se in engines.values():
        for _ in range(6):
            step(se)
    torch.cuda.synchronize()
    times = {t: [] for t in engines}
    for _ in range(rounds):
        for tag, se in engines.items():
            step(se)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(steps):
                step(se)
            torch.cuda.synchronize()
            times[tag].append((time.perf_counter() - t0) / steps)
    mp = statistics.median(times["packed"])
    ms = statistics.median(times["split"])
    print(f"packed: median {mp*1e3:.4f} ms  min {min(times['packed'])*1e3:.4f}")
    print(f"split:  median {ms*1e3:.4f} ms  min {min(times['split'])*1e3:.4f}")
    print(f"delta (split vs packed): {(ms/mp-1)*100:+.2f}%")


if __name__ == "__main__":
    main()
