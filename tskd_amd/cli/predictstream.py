"""Inference service — reference bin/predictStream.py rebuilt on the tskd
bus + MyCNNEngine + embedded prediction store.

Consumes `call-stream` (key "{patientid}_{channel_index}", value JSON array
of processed points), maintains a rolling 120-point context per (patient,
channel), assembles (1, 10, 120) model inputs (missing channels zero), looks
up patient age (65.0 default), runs MyCNN5 + sigmoid, and inserts
(SUBJECT_ID, PRED_TIME, RISK_SCORE) into the prediction store — the
reference's four MySQL interactions (predictStream.py:23-33, 96-190).

DIVERGENCE (deliberate, SURVEY.md §7 hard parts): the reference's tensor
assembly has a stale-variable bug (predictStream.py:113-139 writes every
present channel from the LAST channel's data); we assemble each channel from
its own data. All ready patients are batched into ONE fused GPU call per
trigger (S patients x N=1 window) instead of a per-patient python loop.
"""

from __future__ import annotations

import argparse
import json
import logging
import os
import signal
import time
from collections import defaultdict, deque
from typing import Deque, Dict, Optional, Tuple

import torch

from tskd_amd.bus import Bus, Consumer, Producer
from tskd_amd.config import get_global_config
from tskd_amd.models import build_model, load_checkpoint
from tskd_amd.ops import MyCNNEngine
from tskd_amd.store import AgeTable, PredictionStore

log = logging.getLogger("predictStream")


class PredictStream:
    def __init__(self, bus: Bus, cfg, store: PredictionStore,
                 ages: Optional[AgeTable] = None, model=None,
                 device: str = "cpu", call_topic: str = "call-stream",
                 response_topic: Optional[str] = None,
                 starting: str = "latest"):
        self.cfg = cfg
        self.store = store
        self.ages = ages or AgeTable()
        self.engine = MyCNNEngine(model or build_model("MyCNN5").eval(),
                                  device=device)
        self.device = device
        bus.create_topic(call_topic)
        self.consumer = Consumer(bus, starting=starting)
        self.consumer.subscribe([call_topic])
        self.response_topic = response_topic
        self.producer = Producer(bus) if response_topic else None
        if response_topic:
            bus.create_topic(response_topic)
        self.win = cfg.window_size  # 120
        # rolling context per (patient, channel)
        self.ctx: Dict[Tuple[str, int], Deque[float]] = defaultdict(
            lambda: deque(maxlen=self.win))
        self.last_ts_us: Dict[str, int] = {}
        self.n_predictions = 0
        self.model_path: Optional[str] = None  # set to enable hot reload
        self._model_mtime = 0.0

    def maybe_reload_model(self) -> bool:
        """Hot-reload the checkpoint when the file changes (the reference
        loads once at process start — predictStream.py:36)."""
        if not self.model_path or not os.path.exists(self.model_path):
            return False
        mtime = os.path.getmtime(self.model_path)
        if mtime <= self._model_mtime:
            return False
        try:
            model = load_checkpoint(self.model_path)
        except Exception as e:
            log.warning("hot reload failed (%s); keeping current model", e)
            return False
        self.engine = MyCNNEngine(model, device=self.device)
        self._model_mtime = mtime
        log.info("hot-reloaded model from %s", self.model_path)
        return True

    def trigger(self) -> int:
        msgs = self.consumer.poll(max_msgs=65536, timeout_ms=0)
        touched = set()
        for m in msgs:
            key = m.key.decode()
            pid, _, chan_s = key.rpartition("_")
            try:
                chan = int(chan_s)
                pts = json.loads(m.value)
            except (ValueError, TypeError):
                continue
            if not pid or chan >= self.cfg.n_channels:
                continue
            self.ctx[(pid, chan)].extend(float(p) for p in pts)
            self.last_ts_us[pid] = max(self.last_ts_us.get(pid, 0), m.ts_us)
            touched.add(pid)
        # a patient is ready when every channel WITH data has >= 120 points
        ready = []
        for pid in sorted(touched):
            chans = [c for c in range(self.cfg.n_channels)
                     if (pid, c) in self.ctx and len(self.ctx[(pid, c)]) > 0]
            if chans and all(len(self.ctx[(pid, c)]) >= self.win
                             for c in chans):
                ready.append(pid)
        if not ready:
            return 0
        # model input always has the MODEL's channel count (zero-filled
        # beyond the configured channels), like the reference's fixed
        # (1, 10, 120) tensor (predictStream.py:105).
        cin = self.engine.cin
        x = torch.zeros(len(ready), 1, cin, self.win)
        age = torch.zeros(len(ready), 1)
        for i, pid in enumerate(ready):
            for c in range(min(self.cfg.n_channels, cin)):
                d = self.ctx.get((pid, c))
                if d and len(d) >= self.win:
                    x[i, 0, c, :] = torch.tensor(list(d)[-self.win:])
            age[i, 0] = self.ages.get(pid)
        dev = torch.device(self.device)
        probs = self.engine.forward(x.to(dev), age.to(dev),
                                    apply_sigmoid=True).reshape(-1).cpu()
        for i, pid in enumerate(ready):
            t_us = self.last_ts_us.get(pid, int(time.time() * 1e6))
            self.store.insert(pid, t_us, float(probs[i]))
            log.info("prediction %s @ %.1fs risk=%.4f", pid, t_us / 1e6,
                     probs[i])
            if self.producer:
                self.producer.produce(self.response_topic, pid,
                                      json.dumps({"t_us": t_us,
                                                  "risk": float(probs[i])}))
        self.n_predictions += len(ready)
        return len(ready)


def load_model_for_serving(path: Optional[str]):
    if path and os.path.exists(path):
        return load_checkpoint(path)
    if path:
        log.warning("model path %s missing; using random-init MyCNN5", path)
    return build_model("MyCNN5").eval()


def main(argv=None) -> None:
    logging.basicConfig(level=logging.INFO,
                        format="%(asctime)s %(name)s %(levelname)s %(message)s")
    cfg = get_global_config()
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--model-call-topic", default="call-stream")
    ap.add_argument("--model-response-topic", default=None,
                    help="optional topic for prediction events (the reference "
                         "accepted this flag but never used it)")
    ap.add_argument("--speed", type=float, default=5.0)
    ap.add_argument("--bus-dir", default=None)
    ap.add_argument("--store-path", default="predictions.log")
    ap.add_argument("--model-path", default=cfg.model_path)
    ap.add_argument("--age-table", default=None,
                    help="cohort csv (SUBJECT_ID,dob) or saved age table")
    ap.add_argument("--device", default="cuda" if torch.cuda.is_available()
                    else "cpu")
    ap.add_argument("--starting", default="latest",
                    choices=["latest", "earliest"])
    ap.add_argument("--metrics-port", type=int, default=0,
                    help="expose Prometheus /metrics on 127.0.0.1:PORT")
    ap.add_argument("--max-triggers", type=int, default=0)
    ap.add_argument("--offsets-file", default=None,
                    help="persist/restore consumer offsets (resume-on-restart)")
    ap.add_argument("--hot-reload", action="store_true",
                    help="reload the checkpoint when the file changes")
    args = ap.parse_args(argv)

    bus = Bus(args.bus_dir)
    store = PredictionStore(args.store_path)
    ages = AgeTable()
    if args.age_table and os.path.exists(args.age_table):
        try:
            ages.load_cohort_csv(args.age_table)
        except (ValueError, IndexError):
            ages.load(args.age_table)
    ps = PredictStream(bus, cfg, store, ages,
                       model=load_model_for_serving(args.model_path),
                       device=args.device,
                       call_topic=args.model_call_topic,
                       response_topic=args.model_response_topic,
                       starting=args.starting)
    if args.hot_reload and args.model_path:
        ps.model_path = args.model_path
        if os.path.exists(args.model_path):
            ps._model_mtime = os.path.getmtime(args.model_path)
    trigger_period = cfg.predict_slide_s / args.speed
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
        ps.maybe_reload_model()
        with timer:
            out = ps.trigger()
        timer.add_items(out)
        if args.offsets_file:
            save_offsets(args.offsets_file, ps.consumer.positions())
        n += 1
        if n % 10 == 0:
            log.info("metrics %s", timer.log_line())
        if args.max_triggers and n >= args.max_triggers:
            break
        time.sleep(max(0.0, trigger_period - (time.time() - t0)))


if __name__ == "__main__":
    main()
