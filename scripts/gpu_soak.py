#!/usr/bin/env python3
"""Serving-stability soak (run on an MI355X): long TriggerGraph replay with
ring wraps + repeated CLI-stage triggers; asserts no memory growth and
finite outputs throughout."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402


def soak_trigger_graph(n_triggers=2000, S=4096):
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
    gf = GraphedForward(me, s=S, n=1, dtype=torch.bfloat16, timelast=True)
    tg = TriggerGraph(se, raw, cm, gf, stride=12)
    torch.cuda.synchronize()
    mem0 = torch.cuda.memory_allocated()
    t0 = time.perf_counter()
    for i in range(n_triggers):
        out = tg.replay()
        if i % 200 == 0:
            torch.cuda.synchronize()
            assert torch.isfinite(out).all(), f"non-finite at trigger {i}"
            mem = torch.cuda.memory_allocated()
            assert mem <= mem0 + (64 << 20), f"memory grew: {mem0}->{mem}"
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    # ring wrapped n_triggers*12/2048 times; state indices still consistent
    assert se.head == se.nproc + se.win_buckets - 1
    print(f"[soak] trigger-graph: {n_triggers} replays ok, "
          f"{dt / n_triggers * 1e3:.3f} ms/trigger, "
          f"{n_triggers * 12 / 2048:.1f} ring turns, mem stable")


def soak_cli_stages(n_triggers=30):
    import numpy as np

    from tskd_amd.bus import Bus, Producer
    from tskd_amd.cli.predictstream import PredictStream
    from tskd_amd.cli.processstream import ProcessStream
    from tskd_amd.config import GlobalConfig
    from tskd_amd.store import PredictionStore
    import json
    import tempfile
    cfg = GlobalConfig()
    cfg.channel_names = ["HR", "RESP", "PULSE", "SpO2"]
    d = tempfile.mkdtemp()
    bus = Bus(d + "/bus")
    prod = Producer(bus)
    for c in cfg.channel_names:
        bus.create_topic(cfg.topic_for_channel(c))
    ps = ProcessStream(bus, cfg, max_streams=8, device="cuda",
                       starting="earliest")
    store = PredictionStore(d + "/pred.log")
    pr = PredictStream(bus, cfg, store, device="cuda", starting="earliest")
    rng = np.random.default_rng(0)
    t_stream = 0.0
    for trig in range(n_triggers):
        # one minute of numerics for 4 patients at 1/60 Hz
        for pid_i in range(4):
            pid = f"p{pid_i:06d}"
            for ci, name in enumerate(cfg.channel_names):
                prod.produce(cfg.topic_for_channel(name), pid,
                             json.dumps([ci, float(rng.normal(80, 5))]),
                             ts_us=int(t_stream * 1e6))
        t_stream += 60.0
        ps.trigger()
        pr.trigger()
    torch.cuda.synchronize()
    n = store.count()
    assert n > 0, "no predictions produced"
    print(f"[soak] cli-stages: {n_triggers} triggers, {n} predictions ok")


if __name__ == "__main__":
    assert torch.cuda.is_available()
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 2000
    S = int(sys.argv[2]) if len(sys.argv) > 2 else 4096
    soak_trigger_graph(n, S=S)
    if S <= 8192:  # CLI-stage soak is stream-count independent
        soak_cli_stages()
    print("[soak] ALL OK")
