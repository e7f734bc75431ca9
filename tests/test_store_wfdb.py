"""Prediction store / age table / WFDB reader tests."""

import datetime as dt
import multiprocessing as mp
import os

import numpy as np
import pytest

from tskd_amd.store import AgeTable, PredictionStore, patient_str, subject_id


class TestPredictionStore:
    def test_insert_and_latest(self, tmp_path):
        st = PredictionStore(str(tmp_path / "pred.log"))
        t0 = dt.datetime(2026, 9, 13, 12, 0, 0)
        st.insert("p000194", t0, 0.25)
        st.insert("p000194", t0 + dt.timedelta(minutes=1), 0.5)
        st.insert("p044083", t0, 0.75)
        assert st.count() == 3
        t, score = st.latest("p000194")
        assert score == 0.5 and t == t0 + dt.timedelta(minutes=1)
        assert st.latest("p999999") is None

    def test_since_query(self, tmp_path):
        st = PredictionStore(str(tmp_path / "pred.log"))
        t0 = dt.datetime(2026, 9, 13, 0, 0, 0)
        for i in range(10):
            st.insert(194, t0 + dt.timedelta(hours=i), i / 10)
        rows = st.since(t0 + dt.timedelta(hours=5))
        assert len(rows) == 5
        assert rows[0][0] == "p000194"
        assert all(r[1] >= t0 + dt.timedelta(hours=5) for r in rows)

    def test_batch_and_growth(self, tmp_path):
        st = PredictionStore(str(tmp_path / "pred.log"))
        n = 100_000  # > initial 1 MiB capacity (24 B/record)
        t0 = dt.datetime(2026, 1, 1)
        st.insert_batch([194] * n, [t0] * n, np.linspace(0, 1, n).tolist())
        assert st.count() == n
        _, last = st.latest(194)
        assert last == 1.0

    def test_multiprocess_insert(self, tmp_path):
        path = str(tmp_path / "pred.log")
        PredictionStore(path)  # init
        ctx = mp.get_context("spawn")
        ps = [ctx.Process(target=_store_worker, args=(path, b))
              for b in (100, 200)]
        for p in ps:
            p.start()
        for p in ps:
            p.join(30)
            assert p.exitcode == 0
        assert PredictionStore(path).count() == 400

    def test_ids(self):
        assert subject_id("p000194") == 194
        assert subject_id("p044083-2112-05-04-19-50n") == 44083
        assert patient_str(194) == "p000194"


def _store_worker(path, base):
    s = PredictionStore(path)
    for i in range(200):
        s.insert(base + i % 3, dt.datetime(2026, 1, 1), 0.1)


class TestAgeTable:
    def test_default_and_set(self):
        at = AgeTable()
        assert at.get("p000194") == 65.0  # reference default
        at.set("p000194", 47.5)
        assert at.get("p000194") == 47.5
        assert "p000194" in at and "p000001" not in at

    def test_dob_convention(self):
        at = AgeTable()
        now = dt.date(2026, 9, 13)
        at.set_dob(194, dt.date(1976, 9, 13), now=now)
        # (now - dob).days / 365 — the reference's DATEDIFF/365
        assert abs(at.get(194) - (now - dt.date(1976, 9, 13)).days / 365.0) < 1e-3

    def test_save_load(self, tmp_path):
        at = AgeTable()
        at.set(1, 30.0)
        at.set(2, 40.0)
        p = str(tmp_path / "ages.csv")
        at.save(p)
        at2 = AgeTable()
        at2.load(p)
        assert len(at2) == 2 and at2.get(2) == 40.0

    def test_load_cohort_csv(self, tmp_path, reference_dir):
        src = os.path.join(reference_dir, "data", "patients_age.csv")
        if not os.path.exists(src):
            pytest.skip("cohort csv missing")
        at = AgeTable()
        n = at.load_cohort_csv(src, now=dt.date(2112, 5, 23))
        assert n > 10000
        assert at.get(194) != 65.0  # p000194 is in the cohort


class TestWfdb:
    def _ref_record(self, reference_dir):
        p = os.path.join(
            reference_dir, "data/waveform/physionet.org/files/"
            "mimic3wdb-matched/1.0/p00/p000194/p000194-2112-05-23-14-34n")
        if not os.path.exists(p + ".hea"):
            pytest.skip("reference waveform not present")
        return p

    def test_header_fields(self, reference_dir):
        from tskd_amd.io import rdrecord
        rec = rdrecord(self._ref_record(reference_dir))
        assert rec.n_sig == 7
        assert abs(rec.fs - 1 / 60) < 1e-6
        assert rec.sig_len == 1625
        assert rec.sig_name == ["HR", "PULSE", "RESP", "SpO2", "NBPSys",
                                "NBPDias", "NBPMean"]
        assert rec.p_signal.shape == (1625, 7)
        bdt = rec.base_datetime
        assert bdt is not None and bdt.year == 2112 and bdt.hour == 14

    def test_decode_matches_numpy_oracle(self, reference_dir):
        """Independent numpy fmt-16 decode of the multiplexed .dat."""
        from tskd_amd.io import rdrecord
        path = self._ref_record(reference_dir)
        rec = rdrecord(path)
        dat = os.path.join(os.path.dirname(path), "3400942n.dat")
        raw = np.fromfile(dat, dtype="<i2")
        n = 1625 * 7
        adc = raw[:n].reshape(1625, 7).astype(np.float64)
        adc[adc == -32768] = np.nan
        gains = np.array([10.0, 10.0, 10.0, 10.0, 1.0, 1.0, 1.0])
        # header gains: HR/PULSE/RESP/SpO2 = 10, NBP* = 1 (baseline 0)
        hea = open(path + ".hea").read().splitlines()[1:8]
        gains = np.array([float(l.split()[2].split("/")[0].split("(")[0])
                          for l in hea])
        baselines = np.zeros(7)
        want = (adc - baselines) / gains
        np.testing.assert_allclose(rec.p_signal, want, rtol=1e-12,
                                   equal_nan=True)

    def test_channel_selection(self, reference_dir):
        from tskd_amd.io import rdrecord
        rec = rdrecord(self._ref_record(reference_dir),
                       channel_names=["SpO2", "HR"])
        assert rec.sig_name == ["SpO2", "HR"]
        assert rec.p_signal.shape == (1625, 2)

    def test_fmt80_and_212_roundtrip(self, tmp_path):
        """Synthesize tiny fmt-80 and fmt-212 records; decode vs known adc."""
        from tskd_amd.io import rdrecord
        d = str(tmp_path)
        # fmt 80: offset binary, 1 signal
        adc = np.array([-100, -1, 0, 1, 100, 127], dtype=np.int16)
        (adc.astype(np.int16) + 128).astype(np.uint8).tofile(f"{d}/r80.dat")
        open(f"{d}/r80.hea", "w").write(
            "r80 1 125 6\nr80.dat 80 100(0)/mV 8 0 0 0 0 SIG\n")
        rec = rdrecord(f"{d}/r80")
        np.testing.assert_allclose(rec.p_signal[:, 0], adc / 100.0)
        # fmt 212: two 12-bit samples in 3 bytes, 2 signals interleaved
        s = np.array([100, -200, 300, -400, 500, -600], dtype=np.int32)
        b = bytearray()
        for i in range(0, 6, 2):
            a0, a1 = int(s[i]) & 0xFFF, int(s[i + 1]) & 0xFFF
            b += bytes([a0 & 0xFF, ((a0 >> 8) & 0x0F) | (((a1 >> 8) & 0x0F) << 4),
                        a1 & 0xFF])
        open(f"{d}/r212.dat", "wb").write(bytes(b))
        open(f"{d}/r212.hea", "w").write(
            "r212 2 360 3\nr212.dat 212 200(0)/mV 12 0 0 0 0 A\n"
            "r212.dat 212 100(0)/mV 12 0 0 0 0 B\n")
        rec = rdrecord(f"{d}/r212")
        np.testing.assert_allclose(rec.p_signal[:, 0],
                                   np.array([100, 300, 500]) / 200.0)
        np.testing.assert_allclose(rec.p_signal[:, 1],
                                   np.array([-200, -400, -600]) / 100.0)


class TestWfdbRobustness:
    def _mk(self, d, hea, dat=None, name="r"):
        open(f"{d}/{name}.hea", "w").write(hea)
        if dat is not None:
            np.asarray(dat, dtype="<i2").tofile(f"{d}/{name}.dat")
        from tskd_amd.io import rdrecord
        return lambda **kw: rdrecord(f"{d}/{name}", **kw)

    def test_truncated_dat_reads_nan(self, tmp_path):
        # header says 10 samples, file holds 4 -> the tail decodes as NaN
        rd = self._mk(str(tmp_path), "r 1 125 10\nr.dat 16 10(0)/u 16 0 0 0 0 S\n",
                      [10, 20, 30, 40])
        rec = rd()
        assert rec.p_signal.shape == (10, 1)
        assert np.isfinite(rec.p_signal[:4, 0]).all()
        assert np.isnan(rec.p_signal[4:, 0]).all()

    def test_invalid_sentinel_is_nan(self, tmp_path):
        rd = self._mk(str(tmp_path), "r 1 125 3\nr.dat 16 10(0)/u 16 0 0 0 0 S\n",
                      [100, -32768, 200])
        rec = rd()
        assert np.isnan(rec.p_signal[1, 0])
        np.testing.assert_allclose(rec.p_signal[[0, 2], 0], [10.0, 20.0])

    def test_unsupported_format_raises(self, tmp_path):
        rd = self._mk(str(tmp_path), "r 1 125 2\nr.dat 24 10(0)/u 24 0 0 0 0 S\n",
                      [1, 2])
        with pytest.raises(RuntimeError, match="unsupported format"):
            rd()

    def test_multisegment_no_readable_segment(self, tmp_path):
        # all-gap segment list -> no layout/real segment to define signals
        rd = self._mk(str(tmp_path), "r/2 1 125 100\n~ 60\n~ 40\n")
        with pytest.raises(RuntimeError, match="no readable segment"):
            rd()

    def test_missing_files(self, tmp_path):
        from tskd_amd.io import rdrecord
        with pytest.raises(RuntimeError, match="cannot open"):
            rdrecord(str(tmp_path / "nope"))
        rd = self._mk(str(tmp_path), "r 1 125 2\nr.dat 16 10(0)/u 16 0 0 0 0 S\n")
        with pytest.raises(RuntimeError, match="cannot open"):
            rd()  # .hea exists, .dat missing

    def test_comments_and_baseline(self, tmp_path):
        hea = ("# a comment line\n"
               "r 1 125 3 14:00:00 01/02/2100\n"
               "# another\n"
               "r.dat 16 200(100)/mV 16 0 0 0 0 ECG\n")
        rd = self._mk(str(tmp_path), hea, [100, 300, 500])
        rec = rd()
        # physical = (adc - baseline 100) / gain 200
        np.testing.assert_allclose(rec.p_signal[:, 0], [0.0, 1.0, 2.0])
        assert rec.base_datetime.year == 2100

    def test_unknown_channel_selection_skipped(self, tmp_path):
        rd = self._mk(str(tmp_path), "r 1 125 2\nr.dat 16 10(0)/u 16 0 0 0 0 HR\n",
                      [1, 2])
        rec = rd(channel_names=["SpO2", "HR"])  # SpO2 absent
        assert rec.sig_name == ["HR"]
        assert rec.p_signal.shape == (2, 1)


class TestWfdbMultiSegment:
    """Variable-layout multi-segment records (the MIMIC waveform format):
    master header `rec/nseg`, `~ n` gap segments, a `_layout 0` header
    naming the canonical channels, per-segment headers mapping a SUBSET of
    those channels. Gaps and unmirrored segments read as NaN
    (ref record p000194-2112-05-23-14-34.hea has exactly this shape)."""

    def _mk_multiseg(self, d):
        # layout: channels II, V  — master: gap 4, seg a (II only, 3),
        # seg b (V,II reordered, 2), missing seg m (2) -> 11 samples total
        open(f"{d}/ml_layout.hea", "w").write(
            "ml_layout 2 125 0 10:00:00\n"
            "~ 0 10/mV 16 0 0 0 0 II\n"
            "~ 0 20/mV 16 0 0 0 0 V\n")
        open(f"{d}/r.hea", "w").write(
            "r/5 2 125 11 10:00:00 01/02/2100\n"
            "ml_layout 0\n~ 4\n"
            "sa 3\nsb 2\nsm 2\n")
        open(f"{d}/sa.hea", "w").write(
            "sa 1 125 3\nsa.dat 16 10/mV 16 0 0 0 0 II\n")
        np.asarray([10, 20, 30], dtype="<i2").tofile(f"{d}/sa.dat")
        open(f"{d}/sb.hea", "w").write(
            "sb 2 125 2\n"
            "sb.dat 16 20/mV 16 0 0 0 0 V\n"
            "sb.dat 16 10/mV 16 0 0 0 0 II\n")
        # interleaved (V, II) x 2 samples
        np.asarray([40, 50, 80, 90], dtype="<i2").tofile(f"{d}/sb.dat")
        open(f"{d}/sm.hea", "w").write(
            "sm 1 125 2\nsm.dat 16 10/mV 16 0 0 0 0 II\n")
        # sm.dat deliberately absent (unmirrored segment)
        from tskd_amd.io import rdrecord
        return lambda **kw: rdrecord(f"{d}/r", **kw)

    def test_stitched_layout(self, tmp_path):
        rec = self._mk_multiseg(str(tmp_path))()
        assert rec.n_seg == 5
        assert rec.sig_name == ["II", "V"]
        assert rec.p_signal.shape == (11, 2)
        ii, v = rec.p_signal[:, 0], rec.p_signal[:, 1]
        assert np.isnan(ii[:4]).all() and np.isnan(v[:4]).all()  # gap
        np.testing.assert_allclose(ii[4:7], [1.0, 2.0, 3.0])     # sa /10
        assert np.isnan(v[4:7]).all()        # V absent from segment sa
        np.testing.assert_allclose(v[7:9], [2.0, 4.0])           # sb /20
        np.testing.assert_allclose(ii[7:9], [5.0, 9.0])          # sb /10
        assert np.isnan(ii[9:]).all()        # sm.dat missing -> NaN span
        assert rec.base_datetime.year == 2100

    def test_channel_selection_on_layout(self, tmp_path):
        rec = self._mk_multiseg(str(tmp_path))(channel_names=["V"])
        assert rec.sig_name == ["V"]
        np.testing.assert_allclose(rec.p_signal[7:9, 0], [2.0, 4.0])

    def test_reference_record(self, reference_dir):
        import os
        path = os.path.join(
            reference_dir, "data/waveform/physionet.org/files/"
            "mimic3wdb-matched/1.0/p00/p000194/p000194-2112-05-23-14-34")
        if not os.path.exists(path + ".hea"):
            pytest.skip("reference multi-segment record missing")
        from tskd_amd.io import rdrecord
        rec = rdrecord(path)
        assert rec.n_seg == 7 and rec.fs == 125.0
        assert rec.sig_name == ["II"] and rec.sig_len == 12192771
        a = rec.p_signal[:, 0]
        # exactly the three mirrored segments (0002/0003/0004) are finite
        assert int(np.isfinite(a).sum()) == 384 + 1449229 + 255000
        assert np.isnan(a[:15060]).all()          # leading gap segment
        s0 = 15060 + 10467887                      # 0001.dat not mirrored
        # fmt-80 decode of segment 0002: initval 17, gain 11
        np.testing.assert_allclose(a[s0], 17 / 11)
