"""Training dataset ETL — reference explore_torch.ipynb cells 1-2, 19-25.

The offline twin of the streaming preprocess: per record, resample to the
5-s grid, 3-min rolling mean, linear interpolation, label windows around the
cardiac-arrest time (positive = last 2 h before CA), 120-sample windows with
40% overlap, class-rebalancing samplers (the reference used imblearn's
Random{Under,Over}Sampler; re-implemented here in numpy).
"""

from __future__ import annotations

from typing import Optional, Sequence, Tuple

import numpy as np
import pandas as pd
import torch
from torch.utils.data import TensorDataset

from tskd_amd.engine.windowing import sliding_windows

create_batch = sliding_windows  # reference name (explore_torch.ipynb cell 1)

POSITIVE_HORIZON_S = 2 * 3600  # last 2 h before CA are positive


def record_to_training_frame(p_signal: np.ndarray, fs: float,
                             sig_names: Sequence[str],
                             channel_names: Sequence[str],
                             start_s: float = 0.0,
                             ca_time_s: Optional[float] = None) -> pd.DataFrame:
    """Reference get_record_df (explore_torch.ipynb cell 2): to 5-s grid via
    resample().first(), rolling('3min').mean(), interpolate(linear); columns
    ordered by the configured channel list with missing channels = 0."""
    n = p_signal.shape[0]
    idx = pd.to_timedelta(np.arange(n) / fs, unit="s")
    df = pd.DataFrame(p_signal, index=idx, columns=list(sig_names))
    # clip [start, CA time]
    lo = pd.to_timedelta(start_s, unit="s")
    hi = pd.to_timedelta(ca_time_s, unit="s") if ca_time_s is not None \
        else idx[-1]
    df = df[(df.index >= lo) & (df.index <= hi)]
    # missing configured channels as 0 columns; channel order = wire order
    for c in channel_names:
        if c not in df.columns:
            df[c] = 0.0
    df = df[list(channel_names)]
    df = df.resample("5s").first()
    df = df.rolling("3min").mean()
    df = df.interpolate(method="linear")
    return df


def label_windows(df: pd.DataFrame, ca_time_s: float,
                  window_size: int = 120, overlap_pct: float = 0.4
                  ) -> Tuple[np.ndarray, np.ndarray]:
    """Split the frame at CA-2h: windows entirely before are negative,
    windows in the last 2 h before CA are positive (cell 21)."""
    split = pd.to_timedelta(ca_time_s - POSITIVE_HORIZON_S, unit="s")
    neg_df = df[df.index < split]
    pos_df = df[df.index >= split]
    xs, ys = [], []
    for sub, y in ((neg_df, 0), (pos_df, 1)):
        if len(sub) >= window_size:
            w = sliding_windows(sub.values, window_size, overlap_pct)
            if len(w):
                xs.append(w)
                ys.append(np.full(len(w), y))
    if not xs:
        c = df.shape[1]
        return (np.empty((0, c, window_size)), np.empty((0,)))
    return np.concatenate(xs), np.concatenate(ys)


def clamp_age(age: float) -> float:
    """Reference age handling: clamp [15, 80], NaN -> 50 (cell 21)."""
    if np.isnan(age):
        return 50.0
    return float(np.clip(age, 15.0, 80.0))


def load_dataset(x: np.ndarray, age_arr: np.ndarray,
                 y: np.ndarray) -> TensorDataset:
    """Reference load_dataset (utils.py:365-384): TensorDataset of
    (data fp32, age fp32, target fp32) — BCEWithLogits wants float targets."""
    data = torch.from_numpy(np.ascontiguousarray(x)).float()
    target = torch.from_numpy(np.ascontiguousarray(y)).float()
    age = torch.from_numpy(np.ascontiguousarray(age_arr)).float()
    return TensorDataset(data, age, target)


def random_undersample(x, age, y, strategy: float = 0.3, seed: int = 0):
    """imblearn RandomUnderSampler analog: subsample the majority (negative)
    class to n_pos/strategy (cells 24-25)."""
    rng = np.random.default_rng(seed)
    y = np.asarray(y)
    pos = np.flatnonzero(y == 1)
    neg = np.flatnonzero(y == 0)
    n_neg_keep = min(len(neg), int(round(len(pos) / strategy))) if len(pos) \
        else len(neg)
    keep = np.concatenate([pos, rng.choice(neg, n_neg_keep, replace=False)])
    rng.shuffle(keep)
    return x[keep], age[keep], y[keep]


def random_oversample(x, age, y, seed: int = 0):
    """imblearn RandomOverSampler analog: duplicate minority samples until
    classes are balanced."""
    rng = np.random.default_rng(seed)
    y = np.asarray(y)
    pos = np.flatnonzero(y == 1)
    neg = np.flatnonzero(y == 0)
    if len(pos) == 0 or len(neg) == 0:
        return x, age, y
    minority, majority = (pos, neg) if len(pos) < len(neg) else (neg, pos)
    extra = rng.choice(minority, len(majority) - len(minority), replace=True)
    keep = np.concatenate([majority, minority, extra])
    rng.shuffle(keep)
    return x[keep], age[keep], y[keep]


def make_synthetic_labeled_windows(n_windows: int, n_channels: int = 10,
                                   window: int = 120, pos_frac: float = 0.2,
                                   seed: int = 0):
    """Synthetic labeled windows (no network, no PhysioNet): positives get a
    drifting-vitals signature so training has signal to find."""
    rng = np.random.default_rng(seed)
    x = rng.normal(size=(n_windows, n_channels, window)).astype(np.float32)
    y = (rng.random(n_windows) < pos_frac).astype(np.float32)
    drift = np.linspace(0, 1.5, window, dtype=np.float32)
    x[y == 1, 0, :] += drift   # HR ramps up before arrest
    x[y == 1, 4, :] -= drift * 0.5  # SpO2 sags
    age = rng.uniform(20, 80, n_windows).astype(np.float32)
    return x, age, y
