"""utils + report coverage: offline demo runner, record helpers, eval
reports, sklearn baselines."""

import datetime as dt

import numpy as np
import pytest
import torch

from tskd_amd.models import build_model
from tskd_amd.train.data import make_synthetic_labeled_windows
from tskd_amd.train.report import (classification_metrics, per_patient_f1,
                                   score_model, sklearn_baselines)
from tskd_amd.utils import get_arr, get_ending_time, run_offline_demo


class TestUtils:
    def test_run_offline_demo_synthetic(self):
        out = run_offline_demo(n_synthetic_min=120, seed=0)
        assert out["n_windows"] >= 1
        assert ((out["scores"] >= 0) & (out["scores"] <= 1)).all()
        # 120 minutes of 1/60 Hz -> (119*12+1) grid points
        assert out["grid_points"] == 119 * 12 + 1

    def test_run_offline_demo_reference_record(self, reference_dir, monkeypatch):
        import tskd_amd.config as C
        cfg = C.GlobalConfig()
        cfg.wavef_path = (reference_dir +
                          "/data/waveform/physionet.org/files/"
                          "mimic3wdb-matched/1.0")
        out = run_offline_demo(record_name="p000194-2112-05-23-14-34n",
                               cfg=cfg)
        assert out["n_windows"] >= 1

    def test_get_ending_time(self, reference_dir):
        import tskd_amd.config as C
        from tskd_amd.utils import get_record
        cfg = C.GlobalConfig()
        cfg.wavef_path = (reference_dir +
                          "/data/waveform/physionet.org/files/"
                          "mimic3wdb-matched/1.0")
        rec = get_record("p000194-2112-05-23-14-34n", cfg)
        end = get_ending_time(rec)
        assert end - rec.base_datetime == dt.timedelta(seconds=1625 * 60)

    def test_get_arr(self):
        a = np.zeros((3, 2))
        assert (get_arr(a, 1) == np.ones(3)).all()
        assert get_arr(np.zeros((1, 2)), 5).shape == (1,)

    def test_plot_waveform(self, reference_dir, tmp_path):
        pytest.importorskip("matplotlib")
        import tskd_amd.config as C
        from tskd_amd.utils import plot_waveform
        cfg = C.GlobalConfig()
        cfg.wavef_path = (reference_dir +
                          "/data/waveform/physionet.org/files/"
                          "mimic3wdb-matched/1.0")
        out = str(tmp_path / "w.png")
        plot_waveform("p000194-2112-05-23-14-34n", cfg, out_path=out)
        import os
        assert os.path.getsize(out) > 1000


class TestReport:
    def test_metrics_and_baselines(self):
        x, age, y = make_synthetic_labeled_windows(400, seed=5, pos_frac=0.3)
        m = build_model("MyCNN5").eval()
        probs = score_model(m, x, age)
        rep = classification_metrics(y, probs)
        assert 0.0 <= rep["roc_auc"] <= 1.0
        assert "precision" in rep["report"]
        pf = per_patient_f1(y, probs, ["p%02d" % (i % 4) for i in range(400)])
        assert len(pf) == 4
        # the synthetic signal is linearly separable enough for logreg
        base = sklearn_baselines(x[:300], y[:300], x[300:], y[300:])
        assert base["logreg"]["test_auc"] > 0.8


class TestMetrics:
    def test_stage_timer(self):
        import time as _time

        from tskd_amd.metrics import PipelineMetrics
        pm = PipelineMetrics()
        t = pm.stage("preprocess")
        for _ in range(5):
            with t:
                _time.sleep(0.001)
            t.add_items(12)
        snap = t.snapshot()
        assert snap["calls"] == 5 and snap["items"] == 60
        assert snap["p50_ms"] >= 1.0
        assert "preprocess" in pm.report()


class TestCsvDemo:
    def test_config1_csv_to_mycnn2_cpu(self, tmp_path):
        """BASELINE config 1: data.csv replay -> MyCNN2 eager CPU."""
        from tskd_amd.cli.makedata import make_data
        from tskd_amd.utils import run_csv_demo
        p = str(tmp_path / "data.csv")
        make_data(p, seed=42, hours=2.0)
        out = run_csv_demo(p, variant="MyCNN2")
        assert out["n_windows"] >= 1
        assert ((out["scores"] >= 0) & (out["scores"] <= 1)).all()
        # 2 h at 1 Hz with 50% dropout -> full 5-s grid coverage
        assert out["grid_points"] > 1000


class TestScriptsImportable:
    def test_scripts_parse(self):
        """The operational scripts must at least import/compile on CPU."""
        import py_compile
        for f in ("scripts/ab_bench.py", "scripts/gpu_soak.py",
                  "scripts/demo_e2e.py"):
            py_compile.compile(f, doraise=True)


class TestAuxCLIs:
    def test_download_script_builder(self, tmp_path, monkeypatch):
        """reference bin/download.py:5-25 — wget -r command per patient."""
        from tskd_amd.cli.download import build_commands, main
        cmds = build_commands(["p000194", "p044083"], "data/waveform")
        assert len(cmds) == 2
        assert cmds[0][0] == "wget" and "-r" in cmds[0]
        assert cmds[0][-1].endswith("/p00/p000194/")
        assert cmds[1][-1].endswith("/p04/p044083/")
        monkeypatch.chdir(tmp_path)
        main(["--patients", "p000194", "--script", "dl.sh"])
        body = (tmp_path / "dl.sh").read_text()
        assert body.startswith("#!/bin/sh") and "wget" in body

    def test_mockstream_produces_wire_format(self, tmp_path):
        """reference bin/mock-stream.py — random per-channel samples in the
        [chan_idx, value] wire format on every channel topic."""
        import json
        from tskd_amd.bus import Bus, Consumer
        from tskd_amd.cli.mockstream import main
        from tskd_amd.config import get_global_config
        cfg = get_global_config()
        bus_dir = str(tmp_path / "bus")
        main(["--bus-dir", bus_dir, "--n", "3", "--rate-hz", "1000",
              "--patients", "p000194"])
        bus = Bus(bus_dir)
        cc = Consumer(bus, starting="earliest")
        cc.subscribe([cfg.topic_for_channel(c) for c in cfg.channel_names])
        msgs = cc.poll(max_msgs=256, timeout_ms=200)
        assert len(msgs) == 3 * cfg.n_channels
        seen = set()
        for m in msgs:
            assert m.key == b"p000194"
            ci, val = json.loads(m.value)
            seen.add(ci)
            assert isinstance(val, float)
        assert seen == set(range(cfg.n_channels))

    def test_prometheus_text(self):
        from tskd_amd.metrics import PipelineMetrics, prometheus_text
        pm = PipelineMetrics()
        with pm.stage("ingest"):
            pass
        pm.stage("ingest").add_items(7)
        body = prometheus_text(pm)
        assert body.endswith("\n")
        assert 'tskd_stage_items_total{stage="ingest"} 7' in body
        assert "# TYPE tskd_stage_latency_ms gauge" in body
