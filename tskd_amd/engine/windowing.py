"""CPU-truth implementations of the streaming preprocess semantics.

These mirror the reference's Spark pipeline (reference processStream.py:105-218)
and the offline pandas ETL (explore_torch.ipynb cell 2), and serve as the
oracle for the HIP preprocess kernels (csrc/preprocess_kernels.hip) and the
CPU execution mode of :class:`tskd_amd.engine.StreamEngine`.

Semantics (event-time, all at speed=1):
  - raw samples carry timestamps; 5-s *buckets*: bucket g = [5g, 5(g+1)).
  - the 180 s / 5 s sliding-window average STARTING at grid g is
    mean(raw samples with t in [5g, 5g + 180)) = sum/count over buckets
    [g, g+36)  (Spark window(180s, 5s) keyed by window START).
  - gap handling: ffill (carry forward), then bfill, then fillna(0)
    (processStream.py:114-123).
  - a model window = 120 consecutive processed grid points
    (predictStream.py 600 s / 60 s window -> (10, 120) tensor).
"""

from __future__ import annotations

from typing import Tuple

import numpy as np

BUCKET_S = 5.0
WIN_BUCKETS = 36  # 180 s / 5 s
MODEL_WIN = 120   # 600 s / 5 s


def bucketize(ts: np.ndarray, vals: np.ndarray, n_buckets: int,
              bucket_s: float = BUCKET_S) -> Tuple[np.ndarray, np.ndarray]:
    """Sum/count raw samples into 5-s buckets. NaN samples are missing."""
    bsum = np.zeros(n_buckets)
    bcnt = np.zeros(n_buckets)
    ok = ~np.isnan(vals)
    b = np.floor(ts[ok] / bucket_s).astype(np.int64)
    keep = (b >= 0) & (b < n_buckets)
    np.add.at(bsum, b[keep], vals[ok][keep])
    np.add.at(bcnt, b[keep], 1.0)
    return bsum, bcnt


def window_averages(bsum: np.ndarray, bcnt: np.ndarray,
                    win_buckets: int = WIN_BUCKETS) -> np.ndarray:
    """Sliding raw-sample mean by window START grid; NaN where count==0.

    Output length = len(buckets) - win_buckets + 1 (complete windows only).
    """
    n = len(bsum) - win_buckets + 1
    if n <= 0:
        return np.empty(0)
    cs = np.concatenate([[0.0], np.cumsum(bsum)])
    cc = np.concatenate([[0.0], np.cumsum(bcnt)])
    s = cs[win_buckets:] - cs[:n]
    c = cc[win_buckets:] - cc[:n]
    with np.errstate(invalid="ignore"):
        out = np.where(c > 0, s / np.maximum(c, 1e-30), np.nan)
    return out


def fill_series(vals: np.ndarray, carry: float = np.nan) -> Tuple[np.ndarray, float]:
    """ffill (with cross-batch carry) -> bfill -> fillna(0).

    Returns (filled, new_carry). Mirrors window_fill_kernel exactly.
    """
    out = vals.astype(np.float64).copy()
    c = carry
    for i in range(len(out)):
        if np.isnan(out[i]):
            out[i] = c
        else:
            c = out[i]
    nxt = np.nan
    for i in range(len(out) - 1, -1, -1):
        if np.isnan(out[i]):
            out[i] = nxt
        else:
            nxt = out[i]
    out[np.isnan(out)] = 0.0
    return out, c


def preprocess_series_oracle(ts: np.ndarray, vals: np.ndarray,
                             n_buckets: int) -> np.ndarray:
    """Full per-channel oracle: raw (t, v) events -> processed grid series."""
    bsum, bcnt = bucketize(np.asarray(ts, float), np.asarray(vals, float),
                           n_buckets)
    w = window_averages(bsum, bcnt)
    filled, _ = fill_series(w)
    return filled


def sliding_windows(data: np.ndarray, window_size: int = MODEL_WIN,
                    overlap_pct: float = 0.4,
                    drop_empty: bool = True) -> np.ndarray:
    """Offline training-batch windowing (reference explore_torch.ipynb cell 1
    ``create_batch`` / utils.py:513-538): (T, C) series -> (n, C, window)
    windows with ``overlap_pct`` overlap; windows that are ALL-NaN are
    dropped; remaining NaNs are zero-filled.
    """
    data = np.asarray(data, float)
    if data.ndim == 1:
        data = data[:, None]
    t = len(data)
    if t == window_size:
        wins = data[None, :, :]
    else:
        overlap = int(window_size * overlap_pct)
        step = window_size - overlap
        wins_list = []
        for i in range(0, t - window_size, step):
            w = data[i:i + window_size]
            if drop_empty and np.isnan(w).all():
                continue
            wins_list.append(np.nan_to_num(w, nan=0.0))
        wins = (np.stack(wins_list) if wins_list
                else np.empty((0, window_size, data.shape[1])))
    return np.swapaxes(wins, 1, 2)
