#!/usr/bin/env python3
"""Wire-to-store serving latency + ingest-edge throughput.

Measures the FULL serving path the reference's headline behavioral metric
covers (reference README.md:39-41, predictStream.py:251-263): producer
append -> bus -> native poll+parse+sid (C++) -> event H2D -> GPU ring
ingest -> fused sliding-mean/fill -> window gather -> MFMA conv+LSTM ->
sigmoid -> D2H -> prediction-store insert. Latency per trigger is
  (wall clock after the store insert) - (produce wall clock of the newest
   sample consumed by that trigger)
with event-time = produce wall time (live operation). The reference's
watermark (10/speed s) is a SEMANTIC delay on top and is reported as a
constant, not measured noise.

Modes per scenario:
  live   — producer thread emits at a target events/s; server triggers at
           a fixed cadence. Reports p50/p99/p99.9 wire->store latency.
  drain  — bus pre-filled with a backlog; one timed trigger drains
           max_msgs. Reports edge throughput (events/s, MB/s through the
           full bus->HBM path) broken into poll/ingest/model/persist.

Usage: python scripts/serving_latency.py [--device cuda] [--json OUT]
"""
from __future__ import annotations

import argparse
import json
import os
import statistics
import sys
import tempfile
import threading
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from tskd_amd.bus import Bus, Producer  # noqa: E402
from tskd_amd.cli.serve import FusedServer  # noqa: E402
from tskd_amd.config import get_global_config  # noqa: E402
from tskd_amd.store import PredictionStore  # noqa: E402


def make_server(tmp: str, device: str, max_streams: int,
                ring_grid: int = 4096) -> tuple:
    cfg = get_global_config()
    bus = Bus(os.path.join(tmp, "bus"))
    store = PredictionStore(os.path.join(tmp, "pred.log"))
    srv = FusedServer(bus, cfg, store, device=device,
                      max_streams=max_streams, ring_grid=ring_grid,
                      starting="earliest")
    return bus, store, srv, cfg


def fill_backlog(bus, cfg, n_patients: int, n_events: int,
                 base_us: int = 1_000_000) -> float:
    """Pre-produce n_events across patients/channels with monotonically
    increasing event time; returns produce throughput (events/s)."""
    prod = Producer(bus)
    topics = [cfg.topic_for_channel(c) for c in cfg.channel_names[:8]]
    for t in topics:
        bus.create_topic(t)
    pids = [f"p{i:06d}" for i in range(n_patients)]
    t0 = time.perf_counter()
    base = base_us
    k = 0
    for e in range(n_events):
        pid = pids[e % n_patients]
        ch = (e // n_patients) % 8
        ts_us = base + (e // (n_patients * 8)) * 5_000_000  # 5 s grid
        prod.produce(topics[ch], pid, f"[{ch}, {96.5 + (e % 7)}]",
                     ts_us=ts_us)
        k += 1
    prod.flush()
    dt = time.perf_counter() - t0
    return n_events / dt


def scenario_drain(device: str, n_patients: int, n_events: int) -> dict:
    """Saturated edge: one pre-filled backlog, timed trigger-by-trigger
    drain. Stage breakdown via the server's StageTimer + manual probes."""
    with tempfile.TemporaryDirectory() as tmp:
        bus, store, srv, cfg = make_server(tmp, device, n_patients)
        prod_rate = fill_backlog(bus, cfg, n_patients, n_events)
        drained = 0
        t_poll = t_ingest = t_model = 0.0
        if device != "cpu":
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        while drained < n_events:
            p0 = time.perf_counter()
            sa, ca, va, ta, new_keys = srv.consumer.poll_samples_sid(
                max_msgs=131072, timeout_ms=0, rank=srv.rank,
                world=srv.world, max_streams=srv.max_streams)
            for kk, sid in new_keys:
                srv.pid_index[kk] = sid
                srv.pids.append(kk)
            if not len(sa):
                break
            p1 = time.perf_counter()
            import numpy as np
            srv.hwm = max(srv.hwm, float(ta.max()))
            srv.se.ingest_events_chunked(
                torch.from_numpy(sa.astype(np.int64)),
                torch.from_numpy(ca.astype(np.int64)),
                torch.from_numpy(ta), torch.from_numpy(va),
                advance_to=max(0.0, srv.hwm - srv.watermark_s))
            if device != "cpu":
                torch.cuda.synchronize()
            p2 = time.perf_counter()
            drained += len(sa)
            t_poll += p1 - p0
            t_ingest += p2 - p1
        # one full trigger (model + persist) on the hot rings — feed one
        # more grid point of fresh data (ahead of the consumed backlog)
        # so the trigger actually scores every stream
        fill_backlog(bus, cfg, n_patients, n_patients * 8,
                     base_us=int(srv.hwm * 1e6) + 5_000_000)
        srv.watermark_s = 0.0  # drain everything just produced
        srv.se.force_ready()
        m0 = time.perf_counter()
        n_pred = srv.trigger()
        if device != "cpu":
            torch.cuda.synchronize()
        t_model = time.perf_counter() - m0
        total = time.perf_counter() - t0
        return {
            "scenario": "drain", "device": device,
            "n_patients": n_patients, "n_events": n_events,
            "produce_rate_eps": round(prod_rate),
            "drain_rate_eps": round(drained / (t_poll + t_ingest)),
            "poll_parse_rate_eps": round(drained / t_poll) if t_poll else None,
            "ingest_rate_eps": round(drained / t_ingest) if t_ingest else None,
            "events_drained": drained,
            "trigger_model_persist_ms": round(t_model * 1e3, 3),
            "n_predictions": n_pred,
            "total_s": round(total, 3),
        }


def scenario_live(device: str, n_patients: int, rate_eps: int,
                  n_triggers: int, trigger_period_s: float,
                  poll_thread: bool = False) -> dict:
    """Producer thread at a target event rate with event-time = wall time;
    server triggers on a cadence; wire->store latency per trigger."""
    with tempfile.TemporaryDirectory() as tmp:
        bus, store, srv, cfg = make_server(tmp, device, n_patients)
        if poll_thread:
            srv.start_poll_thread()
        topics = [cfg.topic_for_channel(c) for c in cfg.channel_names[:8]]
        for t in topics:
            bus.create_topic(t)
        # live mode: event time == produce wall time, so the engine's
        # watermark semantics run on the same clock we measure with
        srv.watermark_s = 0.5
        pids = [f"p{i:06d}" for i in range(n_patients)]
        stop = threading.Event()
        produced = [0]

        def producer_loop():
            prod = Producer(bus)
            e = 0
            next_t = time.perf_counter()
            while not stop.is_set():
                now = time.time()
                pid = pids[e % n_patients]
                ch = (e // n_patients) % 8
                prod.produce(topics[ch], pid, f"[{ch}, {96.5 + (e % 7)}]",
                             ts_us=int(now * 1e6))
                e += 1
                produced[0] = e
                # pace to rate_eps (batch sleep every 256 events)
                if e % 256 == 0:
                    next_t += 256.0 / rate_eps
                    dt = next_t - time.perf_counter()
                    if dt > 0:
                        time.sleep(dt)

        th = threading.Thread(target=producer_loop, daemon=True)
        th.start()
        # prime: let data accumulate, then declare the server warm — the
        # reference's ~13-min first-window ramp is a startup constant, not
        # steady-state serving latency
        time.sleep(min(3.0, 8 * trigger_period_s))
        srv.trigger()
        srv.se.force_ready()
        srv.stage_profile = True  # poll/ingest/model/persist decomposition
        lats = []
        for i in range(n_triggers):
            t_start = time.perf_counter()
            newest_before = time.time()
            srv.trigger()           # poll->ingest->model->store, synchronous
            done = time.time()
            # newest consumed sample is at most `hwm` (event time == produce
            # wall time); latency to durable store:
            if srv.hwm > 0:
                lats.append((done - srv.hwm) * 1e3)
            dt = trigger_period_s - (time.perf_counter() - t_start)
            if dt > 0:
                time.sleep(dt)
        stop.set()
        th.join(timeout=2)
        srv.stop_poll_thread()
        lats = [x for x in lats if x >= 0]
        lats.sort()

        def q(p):
            return lats[min(int(p * len(lats)), len(lats) - 1)] if lats \
                else None
        return {
            "scenario": "live", "device": device,
            "poll_thread": poll_thread,
            "n_patients": n_patients, "target_rate_eps": rate_eps,
            "produced_events": produced[0],
            "n_triggers": n_triggers,
            "trigger_period_s": trigger_period_s,
            "wire_to_store_ms": {
                "p50": round(statistics.median(lats), 3) if lats else None,
                "p99": round(q(0.99), 3) if lats else None,
                "max": round(lats[-1], 3) if lats else None,
                "n": len(lats),
            },
            "stage_p50_ms": {
                k: (round(statistics.median(v), 3) if v else None)
                for k, v in srv.stage_ms.items()},
            "note": ("latency = store-insert wall time minus produce wall "
                     "time of the newest consumed sample (event time == "
                     "wall time); excludes the semantic watermark delay "
                     f"({srv.watermark_s}s) and trigger cadence "
                     "by construction only for the newest sample"),
            "n_predictions": srv.n_predictions,
        }


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--device", default="cuda" if torch.cuda.is_available()
                    else "cpu")
    ap.add_argument("--json", default=None)
    ap.add_argument("--quick", action="store_true",
                    help="small shapes (CPU CI smoke)")
    ap.add_argument("--big-live", action="store_true",
                    help="one extra live scenario at 16384 patients")
    args = ap.parse_args()
    dev = args.device
    out = []
    def run(r):
        out.append(r)
        print(json.dumps(r), flush=True)  # stream results scenario-by-scenario
        return r

    if args.quick:
        run(scenario_drain(dev, n_patients=32, n_events=20_000))
        # >= 30 triggers x 0.25 s: the measurement window must span at
        # least one 5-s bucket boundary or no trigger produces new grid
        # points (and therefore no predictions) — phase-dependent flake
        run(scenario_live(dev, n_patients=32, rate_eps=2_000,
                          n_triggers=30, trigger_period_s=0.25))
    else:
        run(scenario_drain(dev, n_patients=1024, n_events=2_000_000))
        run(scenario_live(dev, n_patients=1024, rate_eps=20_000,
                          n_triggers=120, trigger_period_s=0.25))
        run(scenario_live(dev, n_patients=1024, rate_eps=100_000,
                          n_triggers=120, trigger_period_s=0.25))
        run(scenario_live(dev, n_patients=1024, rate_eps=100_000,
                          n_triggers=120, trigger_period_s=0.25,
                          poll_thread=True))
        if args.big_live:
            run(scenario_live(dev, n_patients=16384,
                              rate_eps=100_000, n_triggers=120,
                              trigger_period_s=0.25))
            run(scenario_live(dev, n_patients=16384,
                              rate_eps=100_000, n_triggers=120,
                              trigger_period_s=0.25, poll_thread=True))
    if args.json:
        with open(args.json, "w") as f:
            json.dump(out, f, indent=1)


if __name__ == "__main__":
    main()
