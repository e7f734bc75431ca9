"""WFDB reader — python API over the first-party C++ decoder (_tskd_wfdb).

Replaces wfdb-python for the replay producer (reference sendStream.py:46
wfdb.rdrecord, utils.py:431-437): reads .hea/.dat records (formats 16, 80,
212), scales to physical units, invalid samples -> NaN.
"""

from __future__ import annotations

import datetime as _dt
import os
from dataclasses import dataclass, field
from typing import List, Optional, Sequence

import numpy as np

from tskd_amd import _tskd_wfdb as _C
from tskd_amd.config import GlobalConfig, get_global_config


@dataclass
class Record:
    record_name: str
    fs: float
    n_sig: int
    sig_name: List[str]
    units: List[str]
    gain: List[float]
    p_signal: np.ndarray  # (nsamp, nsig) float64, NaN = invalid
    base_time: str = ""
    base_date: str = ""
    sig_len: int = 0
    n_seg: int = 0  # 0 = single-segment record
    extra: dict = field(default_factory=dict)

    @property
    def base_datetime(self) -> Optional[_dt.datetime]:
        """Combine header base date (dd/mm/yyyy) + time (HH:MM:SS.fff)."""
        if not self.base_date or not self.base_time:
            return None
        try:
            d = _dt.datetime.strptime(self.base_date, "%d/%m/%Y").date()
            tt = self.base_time.split(".")
            t = _dt.datetime.strptime(tt[0], "%H:%M:%S").time()
            us = int(float("0." + tt[1]) * 1e6) if len(tt) > 1 else 0
            return _dt.datetime.combine(d, t).replace(microsecond=us)
        except ValueError:
            return None


def rdrecord(record_path: str,
             channel_names: Optional[Sequence[str]] = None) -> Record:
    d = _C.rdrecord(record_path, list(channel_names or []))
    return Record(
        record_name=d["record_name"], fs=d["fs"], n_sig=d["n_sig"],
        sig_name=list(d["sig_name"]), units=list(d["units"]),
        gain=list(d["gain"]), p_signal=np.asarray(d["p_signal"]),
        base_time=d["base_time"], base_date=d["base_date"],
        sig_len=d["sig_len"], n_seg=d.get("n_seg", 0))


def get_waveform_path(record_name: str,
                      cfg: Optional[GlobalConfig] = None) -> str:
    """reference utils.py:431-433: pXXXXXX-... -> WAVEFPATH/pXX/pXXXXXX/rec."""
    cfg = cfg or get_global_config()
    patient_id = record_name[0:7]
    return os.path.join(cfg.wavef_path, patient_id[0:3], patient_id,
                        record_name)
