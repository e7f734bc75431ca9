"""Mock stream — reference bin/mock-stream.py: push random channel data to
the bus so the dashboard can be exercised without the pipeline."""

from __future__ import annotations

import argparse
import json
import time

import numpy as np

from tskd_amd.bus import Bus, Producer
from tskd_amd.config import get_global_config


def main(argv=None) -> None:
    cfg = get_global_config()
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--bus-dir", default=None)
    ap.add_argument("--patients", nargs="*", default=["p000194"])
    ap.add_argument("--rate-hz", type=float, default=2.0)
    ap.add_argument("--n", type=int, default=0, help="samples (0 = forever)")
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args(argv)

    bus = Bus(args.bus_dir)
    prod = Producer(bus)
    rng = np.random.default_rng(args.seed)
    topics = [cfg.topic_for_channel(c) for c in cfg.channel_names]
    for t in topics:
        bus.create_topic(t)
    i = 0
    while args.n == 0 or i < args.n:
        t_us = int(time.time() * 1e6)
        for pid in args.patients:
            for ci, topic in enumerate(topics):
                prod.produce(topic, pid,
                             json.dumps([ci, float(rng.normal(80, 10))]),
                             ts_us=t_us)
        prod.flush()
        i += 1
        time.sleep(1.0 / args.rate_hz)


if __name__ == "__main__":
    main()
