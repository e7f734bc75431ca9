"""Stream preprocessor — reference bin/processStream.py rebuilt on the tskd
bus + StreamEngine (the Spark Structured Streaming replacement).

Consumes the raw channel topics, applies the event-time 180 s/5 s sliding
raw-sample mean with ffill/bfill/zero-fill (reference processStream.py:
105-218) inside the GPU/CPU StreamEngine ring buffers, and emits each
trigger's NEW processed points to `call-stream` in the reference wire format:
key = "{patientid}_{channel_index}", value = JSON array of floats
(processStream.py:126-165). `--speed` compresses wall-clock triggers; event
time arrives in message timestamps as stream-seconds so window semantics are
speed-invariant.
"""

from __future__ import annotations

import argparse
import json
import logging
import signal
import time
from typing import Dict

import torch

from tskd_amd.bus import Bus, Consumer, Producer
from tskd_amd.config import get_global_config
from tskd_amd.engine import StreamEngine

log = logging.getLogger("processStream")


class ProcessStream:
    def __init__(self, bus: Bus, cfg, max_streams: int = 64,
                 device: str = "cpu", starting: str = "latest",
                 watermark_s: float = 10.0, out_topic: str = "call-stream"):
        self.bus = bus
        self.cfg = cfg
        self.out_topic = out_topic
        self.engine = StreamEngine(max_streams, cfg.n_channels,
                                   ring_grid=4096, device=device)
        self.consumer = Consumer(bus, starting=starting)
        topics = [cfg.topic_for_channel(c) for c in cfg.channel_names]
        for t in topics:
            bus.create_topic(t)
        self.consumer.subscribe(topics)
        bus.create_topic(out_topic)
        self.producer = Producer(bus, retries=cfg.producer_retries)
        self.pid_index: Dict[str, int] = {}
        self.watermark_s = watermark_s
        self.max_streams = max_streams
        self.hwm = 0.0

    def _sid(self, pid: str) -> int:
        if pid not in self.pid_index:
            if len(self.pid_index) >= self.max_streams:
                raise RuntimeError("max_streams exceeded")
            self.pid_index[pid] = len(self.pid_index)
        return self.pid_index[pid]

    def trigger(self) -> int:
        """One micro-batch: drain bus (native C++ wire-format parse), ingest
        events, emit new points."""
        keys, _topics, chans, vals, ts_arr = self.consumer.poll_samples(
            max_msgs=65536, timeout_ms=0)
        si = [self._sid(k) for k in keys]
        n_in = len(si)
        nproc_before = self.engine.nproc
        if n_in:
            self.hwm = max(self.hwm, float(ts_arr.max()))
        advance = max(0.0, self.hwm - self.watermark_s)
        if si:
            self.engine.ingest_events_chunked(
                torch.tensor(si, dtype=torch.long),
                torch.from_numpy(chans).long(),
                torch.from_numpy(ts_arr),
                torch.from_numpy(vals),
                advance_to=advance)
        else:
            # watermark can still advance processed points
            new_head = int(advance / self.engine.bucket_s)
            if new_head > self.engine.head:
                self.engine._clear_ahead(new_head)
                self.engine.head = new_head
                self.engine._refill()
        np_new = self.engine.nproc - nproc_before
        if np_new <= 0:
            return 0
        # catch-up past ring capacity: only the retained tail is emittable
        max_emit = self.engine.G - self.engine.win_buckets - 1
        if np_new > max_emit:
            log.warning("catch-up produced %d points; emitting only the "
                        "retained last %d", np_new, max_emit)
            nproc_before = self.engine.nproc - max_emit
            np_new = max_emit
        # emit the new processed points per (patient, channel) — one
        # vectorized gather, then host lists (no per-point .item() calls)
        gidx = torch.tensor([(nproc_before + j) % self.engine.G
                             for j in range(np_new)], dtype=torch.long,
                            device=self.engine.device)
        block = self.engine.proc.index_select(2, gidx).cpu()
        emitted = 0
        for pid, sid in self.pid_index.items():
            rows = block[sid].tolist()
            for c in range(self.cfg.n_channels):
                self.producer.produce(self.out_topic, f"{pid}_{c}",
                                      json.dumps(rows[c]),
                                      ts_us=int(self.hwm * 1e6))
                emitted += 1
        self.producer.flush(self.out_topic)
        log.info("trigger: %d msgs in, %d new points x %d keys",
                 n_in, np_new, emitted)
        return emitted


def main(argv=None) -> None:
    logging.basicConfig(level=logging.INFO,
                        format="%(asctime)s %(name)s %(levelname)s %(message)s")
    cfg = get_global_config()
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--signal-list", nargs="*", default=None)
    ap.add_argument("--speed", type=float, default=5.0)
    ap.add_argument("--model-call-topic", default="call-stream")
    ap.add_argument("--bus-dir", default=None)
    ap.add_argument("--device", default="cuda" if torch.cuda.is_available()
                    else "cpu")
    ap.add_argument("--max-streams", type=int, default=64)
    ap.add_argument("--starting", default="latest",
                    choices=["latest", "earliest"])
    ap.add_argument("--metrics-port", type=int, default=0,
                    help="expose Prometheus /metrics on 127.0.0.1:PORT")
    ap.add_argument("--max-triggers", type=int, default=0,
                    help="stop after N triggers (0 = run forever)")
    ap.add_argument("--offsets-file", default=None,
                    help="persist/restore consumer offsets (resume-on-restart)")
    ap.add_argument("--trim-consumed", action="store_true",
                    help="retention: reclaim bus storage behind THIS "
                         "consumer's position after each trigger (only safe "
                         "when no other consumer needs the raw topics)")
    args = ap.parse_args(argv)
    if args.signal_list:
        cfg.channel_names = args.signal_list

    bus = Bus(args.bus_dir)
    ps = ProcessStream(bus, cfg, max_streams=args.max_streams,
                       device=args.device, starting=args.starting,
                       watermark_s=cfg.watermark_s,
                       out_topic=args.model_call_topic)
    trigger_period = cfg.preprocess_trigger_s / args.speed
    stop = []
    signal.signal(signal.SIGTERM, lambda *a: stop.append(1))
    if args.offsets_file:
        from tskd_amd.parallel.supervisor import restore_consumer, save_offsets
        restored = restore_consumer(ps.consumer, args.offsets_file)
        if restored:
            log.info("resumed %d partition offsets from %s", restored,
                     args.offsets_file)
    from tskd_amd.metrics import StageTimer
    timer = StageTimer(ap.prog or "stage")
    if args.metrics_port:
        from tskd_amd.cli.serve import start_metrics_server
        start_metrics_server([timer], args.metrics_port)
        log.info("metrics on 127.0.0.1:%d/metrics", args.metrics_port)
    n = 0
    while not stop:
        t0 = time.time()
        with timer:
            out = ps.trigger()
        timer.add_items(out)
        if args.offsets_file:
            save_offsets(args.offsets_file, ps.consumer.positions())
        if args.trim_consumed:
            for key, off in ps.consumer.positions().items():
                topic, _, part = key.rpartition("/")
                bus.trim_topic(topic, int(part), off)
        n += 1
        if n % 10 == 0:
            log.info("metrics %s", timer.log_line())
        if args.max_triggers and n >= args.max_triggers:
            break
        time.sleep(max(0.0, trigger_period - (time.time() - t0)))


if __name__ == "__main__":
    main()
