#!/usr/bin/env python3
"""Sparse (event) ingest micro-benchmark — the real numerics-record path.

The dense kernel serves the synthetic headline; production vitals arrive as
irregular events (fs=1/60 Hz numerics via the bus). This measures
`StreamEngine.ingest_events` + the preprocess/window stages at serving
scale: S streams x C channels x one event/channel/second for a 60-s
trigger (a HIGH event rate — real numerics are 60x sparser).
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402


def main(S=16384, C=8, triggers=30, ev_per_chan_s=1.0) -> None:
    from tskd_amd.engine import StreamEngine
    from tskd_amd.models import build_model
    from tskd_amd.ops import MyCNNEngine
    assert torch.cuda.is_available()
    dev = "cuda"
    se = StreamEngine(S, 10, ring_grid=2048, fs=125.0, device=dev)
    me = MyCNNEngine(build_model("MyCNN5").eval(), device=dev)
    age = torch.full((S, 1), 65.0, device=dev)
    n_ev = int(S * C * 60 * ev_per_chan_s)
    g = torch.Generator().manual_seed(0)
    si = torch.randint(0, S, (n_ev,), generator=g)
    ci = torch.randint(0, C, (n_ev,), generator=g)
    toff = torch.rand(n_ev, generator=g, dtype=torch.float64) * 60.0
    vv = torch.randn(n_ev, generator=g)

    def trigger(t):
        se.ingest_events(si, ci, toff + 60.0 * t, vv,
                         advance_to=60.0 * (t + 1))
        if se.ready:
            w = se.windows(batch=1, stride=12, dtype=torch.bfloat16,
                           timelast=True)
            me.forward(w, age, apply_sigmoid=True)

    for t in range(12):  # warm to steady state
        trigger(t)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for t in range(12, 12 + triggers):
        trigger(t)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / triggers
    print(f"[event_bench] S={S} C={C} events/trigger={n_ev} "
          f"({n_ev / 60:.0f}/s stream-wide): {dt * 1e3:.3f} ms/trigger, "
          f"{n_ev / dt / 1e6:.1f} M events/s, {S / dt / 1e6:.2f} M windows/s")


if __name__ == "__main__":
    import argparse
    ap = argparse.ArgumentParser()
    ap.add_argument("--streams", type=int, default=16384)
    ap.add_argument("--triggers", type=int, default=30)
    a = ap.parse_args()
    main(S=a.streams, triggers=a.triggers)
