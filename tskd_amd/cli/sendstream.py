"""Replay producer — reference bin/sendStream.py rebuilt on the tskd bus.

One thread per patient record replays WFDB numerics samples to the bus in
pseudo-real-time: topic = channel name (spaces -> '_'), key = patient id,
value = JSON [channel_index, value] (reference sendStream.py:59-64), sleeping
1/(fs * frequency * speed) between sample rows (:72) and flushing per row
(:71). Event time is carried explicitly in the message timestamp as
*stream seconds* x 1e6 so downstream event-time windows are exact at any
--speed.
"""

from __future__ import annotations

import argparse
import json
import logging
import math
import threading
import time
from typing import List, Optional

import numpy as np

from tskd_amd.bus import Bus, Producer
from tskd_amd.config import get_global_config
from tskd_amd.io import get_waveform_path, rdrecord

log = logging.getLogger("sendStream")


def send_record_data(bus: Bus, record_id: str, signal_list: Optional[List[str]],
                     speed: float, frequency: float, cfg,
                     producer_factory=None) -> int:
    producer = producer_factory() if producer_factory else \
        Producer(bus, retries=cfg.producer_retries)
    patient_id = record_id[0:7]
    path = get_waveform_path(record_id, cfg)
    record = rdrecord(path, channel_names=signal_list or cfg.channel_names)
    fs = record.fs * frequency
    sent = 0
    for row in range(record.p_signal.shape[0]):
        t_stream_s = row / record.fs  # event time in stream seconds
        for ci, name in enumerate(record.sig_name):
            val = record.p_signal[row, ci]
            if math.isnan(val):
                continue
            topic = cfg.topic_for_channel(name)
            chan_index = cfg.channel_index(name) if name in cfg.channel_names \
                else ci
            producer.produce(topic, patient_id,
                             json.dumps([chan_index, float(val)]),
                             ts_us=int(t_stream_s * 1e6))
            sent += 1
        producer.flush()  # reference flushes every sample row
        if speed > 0:
            time.sleep(1.0 / (fs * speed))
    log.info("record %s: sent %d samples", record_id, sent)
    return sent


def send_csv_data(bus: Bus, csv_path: str, topic: str, key: str,
                  speed: float, cfg, producer_factory=None) -> int:
    """data.csv replay mode (the upstream demo lineage: timestamp,value)."""
    producer = producer_factory() if producer_factory else \
        Producer(bus, retries=cfg.producer_retries)
    data = np.genfromtxt(csv_path, delimiter=",", names=True,
                         dtype=None, encoding="utf-8")
    sent = 0
    t_prev = None
    for rec in data:
        ts = float(rec[0]) if np.isscalar(rec[0]) else 0.0
        try:
            val = float(rec[1])
        except (ValueError, TypeError):
            continue
        producer.produce(topic, key, json.dumps([0, val]),
                         ts_us=int(ts * 1e6))
        sent += 1
        if t_prev is not None and speed > 0:
            time.sleep(max(0.0, (ts - t_prev) / speed))
        t_prev = ts
    producer.flush()
    return sent


def main(argv=None) -> None:
    logging.basicConfig(level=logging.INFO,
                        format="%(asctime)s %(name)s %(levelname)s %(message)s")
    cfg = get_global_config()
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--signal-list", nargs="*", default=None,
                    help="channel names to stream (default: config)")
    ap.add_argument("--speed", type=float, default=5.0,
                    help="time-compression factor (reference default 5)")
    ap.add_argument("--frequency", type=float, default=1.0,
                    help="sampling-frequency multiplier")
    ap.add_argument("--records", nargs="*", default=None,
                    help="patient records (default: config PATIENTRECORDS)")
    ap.add_argument("--bus-dir", default=None)
    ap.add_argument("--csv", default=None, help="CSV replay mode input file")
    ap.add_argument("--log-file", default=None,
                    help="also log to this file (the reference writes "
                         "${DATAPATH}/producer.log, sendStream.py:15-21)")
    ap.add_argument("--topic", default="data", help="[csv mode] topic")
    ap.add_argument("--relay", default=None, metavar="HOST:PORT",
                    help="produce over TCP to a remote node's bus relay "
                         "instead of a local bus directory")
    args = ap.parse_args(argv)
    if args.log_file:
        fh = logging.FileHandler(args.log_file)
        fh.setFormatter(logging.Formatter(
            "%(asctime)s %(name)s %(levelname)s %(message)s"))
        logging.getLogger().addHandler(fh)

    factory = None
    bus = None
    if args.relay:
        from tskd_amd.bus.relay import RelayProducer
        host, _, port = args.relay.rpartition(":")
        factory = lambda: RelayProducer(host or "127.0.0.1", int(port))  # noqa: E731
    else:
        bus = Bus(args.bus_dir)
    if args.csv:
        n = send_csv_data(bus, args.csv, args.topic, "csv", args.speed, cfg,
                          producer_factory=factory)
        log.info("csv replay done: %d messages", n)
        return

    records = args.records or cfg.patient_records
    threads = [
        threading.Thread(target=send_record_data,
                         args=(bus, r, args.signal_list, args.speed,
                               args.frequency, cfg),
                         kwargs={"producer_factory": factory}, daemon=True)
        for r in records
    ]
    for t in threads:
        t.start()
    for t in threads:
        t.join()


if __name__ == "__main__":
    main()
