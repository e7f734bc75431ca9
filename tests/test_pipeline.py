"""End-to-end pipeline integration: sendStream -> bus -> processStream ->
call-stream -> predictStream -> prediction store -> dashboard API.

Exercises the reference's full dataflow in one process with a
synthetic WFDB numerics record at the real MIMIC rate (fs = 1/60 Hz).
"""

import json
import os
import threading
import time

import numpy as np
import pytest
import torch

from tskd_amd.bus import Bus, Consumer
from tskd_amd.cli.predictstream import PredictStream
from tskd_amd.cli.processstream import ProcessStream
from tskd_amd.cli.sendstream import send_csv_data, send_record_data
from tskd_amd.config import GlobalConfig
from tskd_amd.store import AgeTable, PredictionStore


def _write_wfdb_record(root: str, record: str, sig_names, fs: float,
                       nsamp: int, seed=0):
    """Synthesize a multiplexed fmt-16 record under the reference's
    pXX/pXXXXXX directory layout."""
    pid = record[0:7]
    d = os.path.join(root, pid[0:3], pid)
    os.makedirs(d, exist_ok=True)
    rng = np.random.default_rng(seed)
    nsig = len(sig_names)
    adc = (rng.normal(700, 80, size=(nsamp, nsig))).astype(np.int16)
    adc.tofile(os.path.join(d, f"{record}.dat"))
    with open(os.path.join(d, f"{record}.hea"), "w") as f:
        f.write(f"{record} {nsig} {fs:.13f} {nsamp} 14:34:23.221 23/05/2112\n")
        for name in sig_names:
            f.write(f"{record}.dat 16 10/bpm 16 0 736 0 0 {name}\n")
    return adc / 10.0


@pytest.fixture
def cfg(tmp_path):
    c = GlobalConfig()
    c.wavef_path = str(tmp_path / "wavef")
    c.channel_names = ["HR", "RESP", "PULSE", "SpO2"]
    c.patient_records = ["p000194-test"]
    return c


class TestEndToEnd:
    def test_full_pipeline(self, tmp_path, cfg):
        bus = Bus(str(tmp_path / "bus"))
        nsamp = 32  # 32 min of stream time at fs=1/60
        phys = _write_wfdb_record(cfg.wavef_path, "p000194-test",
                                  cfg.channel_names, 1 / 60, nsamp)

        # stage 1: replay producer (speed so high sleeps are ~0)
        sent = send_record_data(bus, "p000194-test", None, speed=1e6,
                                frequency=1.0, cfg=cfg)
        assert sent == nsamp * len(cfg.channel_names)

        # stage 2: preprocessor (from earliest, one micro-batch)
        ps = ProcessStream(bus, cfg, max_streams=4, device="cpu",
                           starting="earliest")
        emitted = ps.trigger()
        assert emitted > 0
        assert ps.engine.nproc >= 120  # enough grid for a model window

        # the call-stream wire format: key pid_chan, value JSON float array
        cc = Consumer(bus, starting="earliest")
        cc.subscribe(["call-stream"])
        msgs = cc.poll(max_msgs=1024, timeout_ms=500)
        keys = {m.key.decode() for m in msgs}
        assert "p000194_0" in keys
        pts = json.loads(msgs[0].value)
        assert isinstance(pts, list) and len(pts) == ps.engine.nproc

        # stage 3: inference -> store
        store = PredictionStore(str(tmp_path / "pred.log"))
        ages = AgeTable()
        ages.set("p000194", 47.0)
        pr = PredictStream(bus, cfg, store, ages, device="cpu",
                           starting="earliest")
        n = pr.trigger()
        assert n == 1
        assert store.count() == 1
        t, risk = store.latest("p000194")
        assert 0.0 <= risk <= 1.0

        # grid values should reflect the synthetic signal's scale
        hr = np.array(json.loads(
            [m for m in msgs if m.key == b"p000194_0"][0].value))
        assert abs(hr[hr != 0].mean() - phys[:, 0].mean()) < 5.0

    def test_incremental_triggers_produce_more_predictions(self, tmp_path, cfg):
        bus = Bus(str(tmp_path / "bus"))
        _write_wfdb_record(cfg.wavef_path, "p000194-test", cfg.channel_names,
                           1 / 60, 60)
        store = PredictionStore(str(tmp_path / "pred.log"))
        ps = ProcessStream(bus, cfg, max_streams=4, starting="earliest")
        pr = PredictStream(bus, cfg, store, device="cpu", starting="earliest")
        send_record_data(bus, "p000194-test", None, 1e6, 1.0, cfg)
        total = 0
        for _ in range(3):
            ps.trigger()
            total += pr.trigger()
        assert total >= 1
        assert store.count() == total

    def test_csv_replay_mode(self, tmp_path, cfg):
        from tskd_amd.cli.makedata import make_data
        p = str(tmp_path / "data.csv")
        n = make_data(p, seed=42, hours=0.25)
        bus = Bus(str(tmp_path / "bus"))
        bus.create_topic("data")
        c = Consumer(bus, starting="earliest")
        c.subscribe(["data"])
        sent = send_csv_data(bus, p, "data", "csv", speed=1e9, cfg=cfg)
        assert sent == n
        msgs = c.poll(max_msgs=10, timeout_ms=500)
        assert len(msgs) == 10
        ch, val = json.loads(msgs[0].value)
        assert ch == 0 and 0 <= val <= 100

    def test_makedata_deterministic(self, tmp_path):
        from tskd_amd.cli.makedata import make_data
        p1, p2 = str(tmp_path / "a.csv"), str(tmp_path / "b.csv")
        make_data(p1, seed=42, hours=0.1)
        make_data(p2, seed=42, hours=0.1)
        assert open(p1).read() == open(p2).read()


class TestDashboard:
    def test_api_endpoints(self, tmp_path, cfg):
        fastapi = pytest.importorskip("fastapi")  # noqa: F841
        from fastapi.testclient import TestClient

        from tskd_amd.cli.plotdata import DashState, build_app
        bus = Bus(str(tmp_path / "bus"))
        _write_wfdb_record(cfg.wavef_path, "p000194-test", cfg.channel_names,
                           1 / 60, 20)
        store = PredictionStore(str(tmp_path / "pred.log"))
        store.insert("p000194", time.time(), 0.42)
        state = DashState(bus, cfg, store, starting="earliest")
        send_record_data(bus, "p000194-test", None, 1e6, 1.0, cfg)
        t = threading.Thread(target=state.pump, daemon=True)
        t.start()
        time.sleep(0.6)
        state._stop = True
        client = TestClient(build_app(state))
        assert "tskd" in client.get("/").text
        pats = client.get("/api/patients").json()
        assert "p000194" in pats
        raw = client.get("/api/raw/p000194").json()
        assert "0" in raw and len(raw["0"]) == 20
        preds = client.get("/api/predictions").json()
        assert preds and preds[0]["patient"] == "p000194"
        assert abs(preds[0]["risk"] - 0.42) < 1e-6
        # Prometheus text exposition
        body = client.get("/metrics").text
        assert "tskd_stage_calls_total{stage=\"plotdata.pump\"}" in body
        assert 'tskd_stage_latency_ms{stage="plotdata.pump",quantile="p50"}' \
            in body
        h = client.get("/health").json()
        assert h["status"] == "ok" and h["predictions"] == 1
        # SSE push stream: first event carries a full snapshot + predictions
        with client.stream("GET", "/api/stream?limit=1") as r:
            assert r.headers["content-type"].startswith("text/event-stream")
            for line in r.iter_lines():
                if line.startswith("data: "):
                    snap = json.loads(line[len("data: "):])
                    break
        assert "p000194" in snap["patients"]
        assert len(snap["raw"]["p000194"]["0"]) == 20
        assert snap["preds"][0]["patient"] == "p000194"


@pytest.mark.gpu
class TestEndToEndGPU:
    def test_full_pipeline_gpu_engines(self, tmp_path, cfg):
        """The CLI stages with their GPU engines: bus -> ProcessStream(cuda)
        -> call-stream -> PredictStream(cuda) -> store."""
        from tskd_amd.bus import Bus
        from tskd_amd.cli.predictstream import PredictStream
        from tskd_amd.cli.processstream import ProcessStream
        from tskd_amd.cli.sendstream import send_record_data
        from tskd_amd.store import PredictionStore
        bus = Bus(str(tmp_path / "bus"))
        _write_wfdb_record(cfg.wavef_path, "p000194-test", cfg.channel_names,
                           1 / 60, 40)
        send_record_data(bus, "p000194-test", None, 1e6, 1.0, cfg)
        ps = ProcessStream(bus, cfg, max_streams=4, device="cuda",
                           starting="earliest")
        emitted = ps.trigger()
        torch.cuda.synchronize()
        assert emitted > 0
        store = PredictionStore(str(tmp_path / "pred.log"))
        pr = PredictStream(bus, cfg, store, device="cuda",
                           starting="earliest")
        n = pr.trigger()
        torch.cuda.synchronize()
        assert n == 1 and store.count() == 1
        _, risk = store.latest("p000194")
        assert 0.0 <= risk <= 1.0

    def test_gpu_cpu_processstream_equivalence(self, tmp_path, cfg):
        """ProcessStream on cuda and cpu produce the same call-stream points."""
        import json as _json

        from tskd_amd.bus import Bus, Consumer
        from tskd_amd.cli.processstream import ProcessStream
        from tskd_amd.cli.sendstream import send_record_data
        out = {}
        for dev in ("cpu", "cuda"):
            bus = Bus(str(tmp_path / f"bus_{dev}"))
            _write_wfdb_record(cfg.wavef_path, "p000194-test",
                               cfg.channel_names, 1 / 60, 36, seed=3)
            send_record_data(bus, "p000194-test", None, 1e6, 1.0, cfg)
            ps = ProcessStream(bus, cfg, max_streams=4, device=dev,
                               starting="earliest")
            ps.trigger()
            torch.cuda.synchronize()
            c = Consumer(bus, starting="earliest")
            c.subscribe(["call-stream"])
            msgs = c.poll(max_msgs=1024, timeout_ms=500)
            out[dev] = {m.key.decode(): _json.loads(m.value) for m in msgs}
        assert out["cpu"].keys() == out["cuda"].keys()
        for k in out["cpu"]:
            np.testing.assert_allclose(out["cpu"][k], out["cuda"][k],
                                       rtol=1e-4, atol=1e-5)


class TestFusedServe:
    def test_fused_server_cpu(self, tmp_path, cfg):
        """serve.py: bus -> fused ingest+preprocess+infer -> store, one
        process (the production arrangement of processStream+predictStream)."""
        from tskd_amd.cli.serve import FusedServer
        bus = Bus(str(tmp_path / "bus"))
        _write_wfdb_record(cfg.wavef_path, "p000194-test", cfg.channel_names,
                           1 / 60, 40)
        store = PredictionStore(str(tmp_path / "pred.log"))
        ages = AgeTable()
        ages.set("p000194", 47.0)
        srv = FusedServer(bus, cfg, store, ages, device="cpu",
                          max_streams=8, ring_grid=1024, starting="earliest")
        send_record_data(bus, "p000194-test", None, 1e6, 1.0, cfg)
        n = srv.trigger()
        assert n == 1
        assert store.count() == 1
        _, risk = store.latest("p000194")
        assert 0.0 <= risk <= 1.0
        # second trigger with no new data -> no duplicate prediction
        assert srv.trigger() == 0
        assert store.count() == 1

    def test_fused_matches_two_stage(self, tmp_path, cfg):
        """The fused daemon's risk score == the two-stage pipeline's score
        for the same record (same windows, same model weights)."""
        import torch as T
        from tskd_amd.cli.serve import FusedServer
        from tskd_amd.models import build_model
        T.manual_seed(77)
        model = build_model("MyCNN5").eval()
        _write_wfdb_record(cfg.wavef_path, "p000194-test", cfg.channel_names,
                           1 / 60, 40, seed=9)

        # two-stage
        bus1 = Bus(str(tmp_path / "bus1"))
        send_record_data(bus1, "p000194-test", None, 1e6, 1.0, cfg)
        ps = ProcessStream(bus1, cfg, max_streams=4, starting="earliest")
        ps.trigger()
        st1 = PredictionStore(str(tmp_path / "p1.log"))
        pr = PredictStream(bus1, cfg, st1, model=model, device="cpu",
                           starting="earliest")
        pr.trigger()
        _, risk_two_stage = st1.latest("p000194")

        # fused
        bus2 = Bus(str(tmp_path / "bus2"))
        send_record_data(bus2, "p000194-test", None, 1e6, 1.0, cfg)
        st2 = PredictionStore(str(tmp_path / "p2.log"))
        srv = FusedServer(bus2, cfg, st2, model=model, device="cpu",
                          max_streams=8, ring_grid=1024, starting="earliest")
        srv.trigger()
        _, risk_fused = st2.latest("p000194")
        assert abs(risk_fused - risk_two_stage) < 1e-4

    @pytest.mark.gpu
    @pytest.mark.parametrize("pipelined", [False, True])
    def test_fused_server_gpu(self, tmp_path, cfg, pipelined):
        from tskd_amd.cli.serve import FusedServer
        bus = Bus(str(tmp_path / "bus"))
        _write_wfdb_record(cfg.wavef_path, "p000194-test", cfg.channel_names,
                           1 / 60, 40)
        store = PredictionStore(str(tmp_path / "pred.log"))
        srv = FusedServer(bus, cfg, store, device="cuda", max_streams=8,
                          ring_grid=1024, starting="earliest",
                          pipelined=pipelined)
        send_record_data(bus, "p000194-test", None, 1e6, 1.0, cfg)
        n = srv.trigger()
        srv.flush()  # pipelined: persist the deferred trigger (no-op else)
        torch.cuda.synchronize()
        assert n == 1 and store.count() == 1
        _, risk = store.latest("p000194")
        assert 0.0 <= risk <= 1.0

    def test_fused_emit_processed_and_response(self, tmp_path, cfg):
        from tskd_amd.cli.serve import FusedServer
        bus = Bus(str(tmp_path / "bus"))
        _write_wfdb_record(cfg.wavef_path, "p000194-test", cfg.channel_names,
                           1 / 60, 40)
        store = PredictionStore(str(tmp_path / "pred.log"))
        srv = FusedServer(bus, cfg, store, device="cpu", max_streams=8,
                          ring_grid=1024, starting="earliest",
                          response_topic="model-response",
                          emit_processed="call-stream")
        cc = Consumer(bus, starting="earliest")
        cc.subscribe(["call-stream", "model-response"])
        send_record_data(bus, "p000194-test", None, 1e6, 1.0, cfg)
        assert srv.trigger() == 1
        msgs = cc.poll(max_msgs=256, timeout_ms=500)
        topics = {m.topic for m in msgs}
        assert topics == {"call-stream", "model-response"}
        resp = [m for m in msgs if m.topic == "model-response"]
        assert resp[0].key == b"p000194"
        payload = json.loads(resp[0].value)
        assert 0.0 <= payload["risk"] <= 1.0
        # call-stream carries the two-stage wire contract
        proc_keys = {m.key.decode() for m in msgs if m.topic == "call-stream"}
        assert "p000194_0" in proc_keys

    def test_fused_emit_catchup_clamp(self, tmp_path, cfg):
        """A backlog wider than the proc ring must not wrap the emitted
        processed-point arrays: the server clamps emission to the retained
        tail (G - win_buckets - 1 points), mirroring processstream's clamp."""
        from tskd_amd.cli.serve import FusedServer
        bus = Bus(str(tmp_path / "bus"))
        # 600 samples at fs=1/60 span 36000 s = 7200 buckets >> ring G=256
        _write_wfdb_record(cfg.wavef_path, "p000194-test", cfg.channel_names,
                           1 / 60, 600)
        store = PredictionStore(str(tmp_path / "pred.log"))
        srv = FusedServer(bus, cfg, store, device="cpu", max_streams=8,
                          ring_grid=256, starting="earliest",
                          emit_processed="call-stream")
        cc = Consumer(bus, starting="earliest")
        cc.subscribe(["call-stream"])
        send_record_data(bus, "p000194-test", None, 1e6, 1.0, cfg)
        assert srv.trigger() >= 1
        max_emit = srv.se.G - srv.se.win_buckets - 1
        msgs = cc.poll(max_msgs=4096, timeout_ms=500)
        assert msgs, "no processed points emitted"
        for m in msgs:
            pts = json.loads(m.value)
            assert len(pts) == max_emit
            assert all(np.isfinite(pts))

    def test_pipelined_mode_defers_then_matches(self, tmp_path, cfg):
        """pipelined=True: the bus poll of trigger T+1 overlaps trigger T's
        GPU work, so persistence lags one trigger — and after flush() the
        scores equal the unpipelined server's exactly."""
        from tskd_amd.cli.serve import FusedServer
        bus = Bus(str(tmp_path / "bus"))
        _write_wfdb_record(cfg.wavef_path, "p000194-test", cfg.channel_names,
                           1 / 60, 40)
        torch.manual_seed(3)
        from tskd_amd.models import build_model
        model = build_model("MyCNN5").eval()
        s1 = FusedServer(bus, cfg, PredictionStore(str(tmp_path / "a.log")),
                         model=model, device="cpu", max_streams=8,
                         ring_grid=1024, starting="earliest")
        s2 = FusedServer(bus, cfg, PredictionStore(str(tmp_path / "b.log")),
                         model=model, device="cpu", max_streams=8,
                         ring_grid=1024, starting="earliest", pipelined=True)
        send_record_data(bus, "p000194-test", None, 1e6, 1.0, cfg)
        assert s1.trigger() == 1 and s1.store.count() == 1
        assert s2.trigger() == 1
        assert s2.store.count() == 0          # deferred one trigger
        s2.flush()
        assert s2.store.count() == 1
        _, r1 = s1.store.latest("p000194")
        _, r2 = s2.store.latest("p000194")
        assert abs(r1 - r2) < 1e-6

    def test_hot_reload(self, tmp_path, cfg):
        """Swapping the checkpoint file between triggers changes the served
        model (the reference loads once at start and never reloads)."""
        from tskd_amd.cli.serve import FusedServer
        from tskd_amd.models import build_model, save_checkpoint
        bus = Bus(str(tmp_path / "bus"))
        _write_wfdb_record(cfg.wavef_path, "p000194-test", cfg.channel_names,
                           1 / 60, 52)
        store = PredictionStore(str(tmp_path / "pred.log"))
        ckpt = str(tmp_path / "model.pth")
        torch.manual_seed(1)
        save_checkpoint(build_model("MyCNN5").eval(), ckpt)
        from tskd_amd.models import load_checkpoint
        srv = FusedServer(bus, cfg, store, model=load_checkpoint(ckpt),
                          device="cpu", max_streams=8, ring_grid=1024,
                          starting="earliest")
        srv.model_path = ckpt
        import os as _os
        srv._model_mtime = _os.path.getmtime(ckpt)
        send_record_data(bus, "p000194-test", None, 1e6, 1.0, cfg)
        assert not srv.maybe_reload_model()  # unchanged file
        assert srv.trigger() == 1
        _, risk1 = store.latest("p000194")
        # retrain/replace the model on disk with different weights
        torch.manual_seed(99)
        save_checkpoint(build_model("MyCNN5").eval(), ckpt)
        _os.utime(ckpt, (time.time() + 2, time.time() + 2))
        assert srv.maybe_reload_model()
        # more data -> a new prediction from the NEW model
        _write_wfdb_record(cfg.wavef_path, "p000194-test", cfg.channel_names,
                           1 / 60, 64, seed=5)
        send_record_data(bus, "p000194-test", None, 1e6, 1.0, cfg)
        assert srv.trigger() == 1
        _, risk2 = store.latest("p000194")
        assert abs(risk1 - risk2) > 1e-6  # different weights, different score


class TestMetricsServer:
    def test_prometheus_scrape(self):
        import urllib.request
        from tskd_amd.cli.serve import start_metrics_server
        from tskd_amd.metrics import StageTimer
        t = StageTimer("serve")
        with t:
            pass
        t.add_items(3)
        srv = start_metrics_server([t], 0)  # port 0 = ephemeral
        port = srv.server_address[1]
        try:
            body = urllib.request.urlopen(
                f"http://127.0.0.1:{port}/metrics", timeout=5).read().decode()
            assert 'tskd_stage_items_total{stage="serve"} 3' in body
            import urllib.error
            with pytest.raises(urllib.error.HTTPError):
                urllib.request.urlopen(f"http://127.0.0.1:{port}/nope",
                                       timeout=5)
        finally:
            srv.shutdown()
