"""Shared helpers — the reference bin/utils.py surface not covered elsewhere.

Coverage map (reference utils.py -> here / elsewhere):
  get_global_config :17-65        -> tskd_amd.config
  build_spark_session :69-100     -> (no Spark: tskd_amd.engine.StreamEngine)
  AverageMeter / accuracy :104-134-> tskd_amd.train.metrics
  train/evaluate :183-275         -> tskd_amd.train.loop
  load_dataset :365-384           -> tskd_amd.train.data
  producer/consumer cfg :409-428  -> tskd_amd.bus
  get_waveform_path/get_record    -> tskd_amd.io.wfdb + here
  get_base_time/get_ending_time   -> here
  plot_waveform :475-509          -> here (matplotlib optional)
  create_batch/get_arr :513-541   -> tskd_amd.engine.windowing + here
  run_model_dummy :567-669        -> run_offline_demo here
"""

from __future__ import annotations

import datetime as _dt
from typing import Optional, Sequence

import numpy as np
import torch

from tskd_amd.config import GlobalConfig, get_global_config
from tskd_amd.io import get_waveform_path, rdrecord


def get_record(record_name: str, cfg: Optional[GlobalConfig] = None,
               channel_names: Optional[Sequence[str]] = None):
    """reference utils.py:435-437."""
    cfg = cfg or get_global_config()
    return rdrecord(get_waveform_path(record_name, cfg),
                    channel_names=channel_names or cfg.channel_names)


def get_base_time(record) -> Optional[_dt.datetime]:
    """reference utils.py:439-443."""
    return record.base_datetime


def get_ending_time(record) -> Optional[_dt.datetime]:
    """reference utils.py:445-473: base time + sig_len / fs seconds."""
    bt = record.base_datetime
    if bt is None:
        return None
    return bt + _dt.timedelta(seconds=record.sig_len / record.fs)


def get_arr(arr: np.ndarray, y) -> np.ndarray:
    """reference utils.py:540-541 / notebook get_arr."""
    if arr.shape[0] > 1:
        return np.array([y] * arr.shape[0]).squeeze()
    return np.array([y] * arr.shape[0])


def plot_waveform(record_name: str, cfg: Optional[GlobalConfig] = None,
                  out_path: Optional[str] = None):
    """reference utils.py:475-509 (matplotlib; headless-safe)."""
    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt
    rec = get_record(record_name, cfg)
    fig, axes = plt.subplots(rec.n_sig, 1, figsize=(10, 2 * rec.n_sig),
                             sharex=True)
    if rec.n_sig == 1:
        axes = [axes]
    t = np.arange(rec.p_signal.shape[0]) / rec.fs / 60.0
    for i, ax in enumerate(axes):
        ax.plot(t, rec.p_signal[:, i], lw=0.7)
        ax.set_ylabel(rec.sig_name[i])
    axes[-1].set_xlabel("minutes")
    fig.suptitle(record_name[0:7])
    if out_path:
        fig.savefig(out_path, dpi=72)
        plt.close(fig)
    return fig


def run_offline_demo(model=None, record_name: Optional[str] = None,
                     cfg: Optional[GlobalConfig] = None,
                     n_synthetic_min: int = 120, seed: int = 0) -> dict:
    """Offline end-to-end runner without bus/engine (reference
    run_model_dummy, utils.py:567-669): record (or synthetic waveform) ->
    pandas ETL -> 120-sample windows -> model -> sigmoid scores."""
    from tskd_amd.engine.windowing import sliding_windows
    from tskd_amd.models import build_model
    from tskd_amd.train.data import record_to_training_frame
    cfg = cfg or get_global_config()
    model = model or build_model("MyCNN5").eval()
    if record_name is not None:
        rec = get_record(record_name, cfg)
        sig, fs, names = rec.p_signal, rec.fs, rec.sig_name
    else:
        rng = np.random.default_rng(seed)
        fs = 1 / 60
        n = n_synthetic_min
        names = list(cfg.channel_names[:7])
        sig = rng.normal(80, 10, size=(n, len(names)))
    df = record_to_training_frame(sig, fs, names, cfg.channel_names)
    wins = sliding_windows(df.values, cfg.window_size, cfg.record_overlap)
    if len(wins) == 0:
        return {"n_windows": 0, "scores": np.empty(0)}
    x = torch.from_numpy(wins).float()
    age = torch.full((x.shape[0],), 65.0)
    with torch.no_grad():
        scores = torch.sigmoid(model(x, age)).numpy()
    return {"n_windows": len(wins), "scores": scores,
            "grid_points": len(df)}


def run_csv_demo(csv_path: str, variant: str = "MyCNN2",
                 model=None) -> dict:
    """BASELINE config 1: single-patient CSV replay (timestamp,value) ->
    offline preprocess -> MyCNN2 eager PyTorch on CPU. No bus, no GPU.

    The CSV stream is a single channel; it lands in wire channel 0 with the
    remaining model channels zero (the reference's missing-channel rule).
    """
    from tskd_amd.engine.windowing import (MODEL_WIN, preprocess_series_oracle,
                                           sliding_windows)
    from tskd_amd.models import build_model
    rows = np.genfromtxt(csv_path, delimiter=",", names=True)
    ts = np.asarray(rows["timestamp"], float)
    vals = np.asarray(rows["value"], float)
    n_buckets = int(ts.max() // 5) + 1
    series = preprocess_series_oracle(ts, vals, n_buckets)
    model = model or build_model(variant)
    model = model.eval()
    cin = int(model.IN_CHANNELS)
    grid = np.zeros((len(series), cin))
    grid[:, 0] = series
    wins = sliding_windows(grid, MODEL_WIN, 0.4)
    if len(wins) == 0:
        return {"n_windows": 0, "scores": np.empty(0)}
    with torch.no_grad():
        x = torch.from_numpy(wins).float()
        scores = torch.sigmoid(model(x, torch.full((len(wins),), 65.0)))
    return {"n_windows": len(wins), "grid_points": len(series),
            "scores": scores.numpy()}
