"""Prediction store + age table — python API over _tskd_store.

The embedded replacement for the reference's MySQL `predictions` and
`patients_age` tables (reference db/init.sql:24-39), exposing the exact four
queries the reference issues: insert prediction, latest
prediction for a patient, predictions since a timestamp, age lookup with the
65.0 default (predictStream.py:151). Patient ids use the reference string
form ``pXXXXXX`` <-> integer SUBJECT_ID.
"""

from __future__ import annotations

import datetime as _dt
import os
from typing import List, Optional, Tuple

from tskd_amd import _tskd_store as _C


def subject_id(patient: str | int) -> int:
    """'p000194' -> 194 (reference patient-id convention)."""
    if isinstance(patient, int):
        return patient
    return int(patient.lstrip("pP").split("-")[0])


def patient_str(sid: int) -> str:
    return f"p{sid:06d}"


def _us(t) -> int:
    if isinstance(t, (int, float)):
        return int(t * 1e6) if isinstance(t, float) else int(t)
    if isinstance(t, _dt.datetime):
        return int(t.timestamp() * 1e6)
    raise TypeError(type(t))


class PredictionStore:
    """Append-only mmap'd prediction log; multi-process safe."""

    def __init__(self, path: str):
        os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
        self._c = _C.PredictionStore(path)
        self.path = path

    def insert(self, patient: str | int, pred_time, risk_score: float) -> None:
        self._c.insert(subject_id(patient), _us(pred_time), float(risk_score))

    def insert_batch(self, patients, pred_times, risk_scores) -> None:
        sids = [subject_id(p) for p in patients]
        times = [_us(t) for t in pred_times]
        self._c.insert_batch(sids, times, [float(s) for s in risk_scores])

    def count(self) -> int:
        return self._c.count()

    def latest(self, patient: str | int) -> Optional[Tuple[_dt.datetime, float]]:
        r = self._c.latest(subject_id(patient))
        if r is None:
            return None
        t_us, score = r
        return _dt.datetime.fromtimestamp(t_us / 1e6), score

    def since(self, since_time, limit: int = 1 << 20
              ) -> List[Tuple[str, _dt.datetime, float]]:
        rows = self._c.since(_us(since_time), limit)
        return [(patient_str(sid), _dt.datetime.fromtimestamp(t / 1e6), sc)
                for sid, t, sc in rows]

    def today(self) -> List[Tuple[str, _dt.datetime, float]]:
        """plotData.py:198's "today's predictions" query."""
        midnight = _dt.datetime.now().replace(hour=0, minute=0, second=0,
                                              microsecond=0)
        return self.since(midnight)

    def tail(self, k: int = 10) -> List[Tuple[str, _dt.datetime, float]]:
        return [(patient_str(sid), _dt.datetime.fromtimestamp(t / 1e6), sc)
                for sid, t, sc in self._c.tail(k)]


class AgeTable:
    """patients_age: SUBJECT_ID -> age, with the reference's
    (CURRENT_DATE - DOB)/365-day convention (predictStream.py:30) and 65.0
    default for unknown patients (:151)."""

    def __init__(self):
        self._c = _C.AgeTable()

    def set(self, patient: str | int, age: float) -> None:
        self._c.set(subject_id(patient), float(age))

    def set_dob(self, patient: str | int, dob: _dt.date,
                now: Optional[_dt.date] = None) -> None:
        now = now or _dt.date.today()
        self._c.set(subject_id(patient), (now - dob).days / 365.0)

    def get(self, patient: str | int, default: float = 65.0) -> float:
        return self._c.get(subject_id(patient), default)

    def __contains__(self, patient) -> bool:
        return self._c.contains(subject_id(patient))

    def __len__(self) -> int:
        return len(self._c)

    def load_cohort_csv(self, path: str,
                        now: Optional[_dt.date] = None) -> int:
        """Load the reference cohort table format
        (data/patients_age.csv: SUBJECT_ID,dob rows)."""
        now = now or _dt.date.today()
        n = 0
        with open(path) as f:
            header = f.readline()
            cols = [c.strip().lower() for c in header.split(",")]
            sid_i = cols.index("subject_id")
            dob_i = cols.index("dob")
            for line in f:
                parts = line.strip().split(",")
                if len(parts) <= max(sid_i, dob_i):
                    continue
                try:
                    sid = int(parts[sid_i])
                    dob = _dt.datetime.strptime(
                        parts[dob_i].split(" ")[0], "%Y-%m-%d").date()
                except ValueError:
                    continue
                self._c.set(sid, (now - dob).days / 365.0)
                n += 1
        return n

    def save(self, path: str) -> None:
        self._c.save(path)

    def load(self, path: str) -> None:
        self._c.load(path)
