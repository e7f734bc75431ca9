#!/usr/bin/env python3
"""Integration soak of the serving DAEMON (cli.serve.main) — the pieces
that only meet in production run together: live producer (wall-clock
event time) -> bus -> background poll thread -> GPU rings -> fused model
-> store, plus a mid-run hot model reload and the Prometheus metrics
endpoint, all through the real CLI entry.

Usage: python scripts/serve_soak.py [--device cuda] [--seconds 60]
Exits 0 iff predictions kept flowing, the reload happened, and /metrics
served throughout.
"""
from __future__ import annotations

import argparse
import json
import os
import subprocess
import sys
import tempfile
import threading
import time
import urllib.request

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--device", default=None)
    ap.add_argument("--seconds", type=float, default=60.0)
    ap.add_argument("--rate-eps", type=int, default=20_000)
    ap.add_argument("--patients", type=int, default=512)
    ap.add_argument("--metrics-port", type=int, default=5981)
    args = ap.parse_args()
    import torch

    from tskd_amd.bus import Bus, Producer
    from tskd_amd.config import get_global_config
    from tskd_amd.models import build_model, save_checkpoint
    from tskd_amd.store import PredictionStore
    dev = args.device or ("cuda" if torch.cuda.is_available() else "cpu")
    tmp = tempfile.mkdtemp(prefix="serve_soak_")
    bus_dir = os.path.join(tmp, "bus")
    store_path = os.path.join(tmp, "pred.log")
    ckpt = os.path.join(tmp, "model.pth")
    torch.manual_seed(0)
    save_checkpoint(build_model("MyCNN5").eval(), ckpt)

    cfg = get_global_config()
    bus = Bus(bus_dir)
    topics = [cfg.topic_for_channel(c) for c in cfg.channel_names[:8]]
    for t in topics:
        bus.create_topic(t)
    stop = threading.Event()
    produced = [0]

    SPEED = 240.0  # event-time compression, matching `serve --speed 240`
    t_base = time.time()

    def producer_loop():
        prod = Producer(bus)
        e = 0
        next_t = time.perf_counter()
        while not stop.is_set():
            # compressed event time, like the reference's sped-up replay:
            # 780 s of event time (first full model window) passes in
            # ~3.3 s of wall time at SPEED=240
            ev_s = (time.time() - t_base) * SPEED
            pid = f"p{e % args.patients:06d}"
            ch = (e // args.patients) % 8
            prod.produce(topics[ch], pid, f"[{ch}, {97.0 + e % 5}]",
                         ts_us=int(ev_s * 1e6))
            e += 1
            produced[0] = e
            if e % 256 == 0:
                next_t += 256.0 / args.rate_eps
                dt = next_t - time.perf_counter()
                if dt > 0:
                    time.sleep(dt)

    th = threading.Thread(target=producer_loop, daemon=True)
    th.start()

    # ~4 triggers/s at the reference slide of 60 s => --speed 240
    srv = subprocess.Popen(
        [sys.executable, "-m", "tskd_amd.cli.serve",
         "--bus-dir", bus_dir, "--store-path", store_path,
         "--model-path", ckpt, "--device", dev,
         "--max-streams", str(args.patients), "--starting", "earliest",
         "--poll-thread", "--hot-reload", "--speed", "240",
         "--metrics-port", str(args.metrics_port)],
        cwd=REPO, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        text=True)
    store = PredictionStore(store_path)
    t_end = time.time() + args.seconds
    reloaded = False
    counts = []
    metrics_ok = 0
    try:
        while time.time() < t_end:
            time.sleep(3.0)
            counts.append(store.count())
            try:
                body = urllib.request.urlopen(
                    f"http://127.0.0.1:{args.metrics_port}/metrics",
                    timeout=3).read().decode()
                if "tskd_stage_calls_total" in body:
                    metrics_ok += 1
            except Exception:
                pass
            if not reloaded and time.time() > t_end - args.seconds / 2:
                # mid-run hot reload: new random weights, same file
                torch.manual_seed(1)
                save_checkpoint(build_model("MyCNN5").eval(), ckpt)
                reloaded = True
            if srv.poll() is not None:
                break
    finally:
        stop.set()
        srv.terminate()
        try:
            out, _ = srv.communicate(timeout=15)
        except subprocess.TimeoutExpired:
            srv.kill()
            out, _ = srv.communicate()

    grew = sum(1 for a, b in zip(counts, counts[1:]) if b > a)
    hot = "hot-reloaded model" in out
    rec = {
        "device": dev, "seconds": args.seconds,
        "produced_events": produced[0],
        "prediction_counts": counts[-5:],
        "count_growth_intervals": grew,
        "metrics_scrapes_ok": metrics_ok,
        "hot_reload_logged": hot,
        "server_rc": srv.returncode,
    }
    print(json.dumps(rec), flush=True)
    ok = (counts and counts[-1] > 0 and grew >= max(2, len(counts) // 3)
          and metrics_ok >= 2 and hot)
    if not ok:
        print("---- server output tail ----")
        print(out[-3000:])
        sys.exit(1)


if __name__ == "__main__":
    main()
