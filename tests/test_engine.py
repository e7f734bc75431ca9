"""Streaming window engine: oracle properties + StreamEngine (CPU & GPU).

The preprocess semantics under test (reference processStream.py:105-218):
5-s buckets, 180 s/5 s sliding raw-sample mean keyed by window start,
ffill -> bfill -> fillna(0), then 120-point model windows
(predictStream.py:248-259).
"""

import numpy as np
import pytest
import torch

from tskd_amd.engine import StreamEngine, preprocess_series_oracle
from tskd_amd.engine.windowing import (WIN_BUCKETS, bucketize, fill_series,
                                       window_averages)


def brute_force_window_avgs(ts, vals, n_buckets):
    """Independent O(n^2) implementation: mean of raw samples with
    t in [5g, 5g+180) for each complete window start g."""
    ts, vals = np.asarray(ts, float), np.asarray(vals, float)
    ok = ~np.isnan(vals)
    nw = n_buckets - WIN_BUCKETS + 1
    out = np.full(max(nw, 0), np.nan)
    for g in range(max(nw, 0)):
        m = ok & (ts >= 5 * g) & (ts < 5 * g + 180) & (ts >= 0) & (ts < n_buckets * 5)
        if m.any():
            out[g] = vals[m].mean()
    return out


class TestOracle:
    def test_window_avg_matches_brute_force(self):
        rng = np.random.default_rng(0)
        n_buckets = 100
        ts = rng.uniform(0, n_buckets * 5, 400)
        vals = rng.normal(size=400)
        vals[rng.random(400) < 0.2] = np.nan
        bsum, bcnt = bucketize(ts, vals, n_buckets)
        got = window_averages(bsum, bcnt)
        want = brute_force_window_avgs(ts, vals, n_buckets)
        np.testing.assert_allclose(got, want, rtol=1e-12, equal_nan=True)

    def test_fill_series_semantics(self):
        v = np.array([np.nan, np.nan, 3.0, np.nan, 5.0, np.nan])
        filled, carry = fill_series(v)
        # leading NaNs bfilled from 3.0; interior ffilled; trailing ffilled
        np.testing.assert_allclose(filled, [3, 3, 3, 3, 5, 5])
        assert carry == 5.0

    def test_fill_series_carry_across_batches(self):
        filled1, carry = fill_series(np.array([1.0, np.nan]))
        assert carry == 1.0
        filled2, carry2 = fill_series(np.array([np.nan, 2.0]), carry)
        np.testing.assert_allclose(filled2, [1.0, 2.0])
        assert carry2 == 2.0

    def test_fill_all_nan_is_zero(self):
        filled, carry = fill_series(np.array([np.nan, np.nan]))
        np.testing.assert_allclose(filled, [0.0, 0.0])
        assert np.isnan(carry)

    def test_sparse_numerics_rate(self):
        # fs = 1/60 Hz (the MIMIC numerics rate): one sample per 12 buckets;
        # every complete window still averages the 3 raw samples inside it.
        ts = np.arange(0, 1200, 60.0)
        vals = np.arange(len(ts), dtype=float)
        out = preprocess_series_oracle(ts, vals, 240)
        want = brute_force_window_avgs(ts, vals, 240)
        # no gaps after fills: every window contains >= 2 samples
        assert not np.isnan(out).any()
        np.testing.assert_allclose(out, want)


def _mk_engine(S=3, C=4, fs=25.0, G=512, device="cpu"):
    return StreamEngine(S, C, ring_grid=G, fs=fs, device=device)


def _dense_raw(S, CIN, T, seed=0, nan_frac=0.0):
    rng = np.random.default_rng(seed)
    raw = rng.normal(size=(S, CIN, T))
    if nan_frac:
        raw[rng.random(raw.shape) < nan_frac] = np.nan
    return torch.from_numpy(raw).float()


class TestStreamEngineCPU:
    def test_dense_matches_oracle(self):
        S, C, fs = 2, 3, 25.0
        eng = _mk_engine(S, C, fs)
        T = int(fs * 60 * 20)  # 20 minutes
        raw = _dense_raw(S, C, T, nan_frac=0.3)
        eng.ingest_dense(raw)
        assert eng.nproc == eng.head - WIN_BUCKETS + 1
        n_buckets = eng.head
        for s in range(S):
            for c in range(C):
                ts = np.arange(T) / fs
                want = preprocess_series_oracle(ts, raw[s, c].numpy(), n_buckets)
                got = eng.proc[s, c, :eng.nproc].numpy()
                np.testing.assert_allclose(got, want, rtol=1e-5, atol=1e-6)

    def test_incremental_equals_oneshot(self):
        S, C, fs = 2, 2, 25.0
        T = int(fs * 60 * 10)
        raw = _dense_raw(S, C, T, seed=3, nan_frac=0.4)
        e1 = _mk_engine(S, C, fs)
        e1.ingest_dense(raw)
        e2 = _mk_engine(S, C, fs)
        chunk = int(fs * 60)  # 60-s triggers
        for i in range(0, T, chunk):
            e2.ingest_dense(raw[:, :, i:i + chunk])
        assert e1.head == e2.head and e1.nproc == e2.nproc
        np.testing.assert_allclose(
            e1.proc[:, :, :e1.nproc].numpy(), e2.proc[:, :, :e2.nproc].numpy(),
            rtol=1e-5, atol=1e-6)

    def test_events_match_dense(self):
        # The same samples pushed as events must produce the same grid.
        S, C, fs = 1, 2, 5.0
        T = int(fs * 60 * 8)
        raw = _dense_raw(S, C, T, seed=5)
        e1 = _mk_engine(S, C, fs)
        e1.ingest_dense(raw)
        e2 = _mk_engine(S, C, fs)
        ts, si, ci, vv = [], [], [], []
        for c in range(C):
            for t in range(T):
                si.append(0); ci.append(c); ts.append(t / fs)
                vv.append(float(raw[0, c, t]))
        e2.ingest_events(torch.tensor(si), torch.tensor(ci),
                         torch.tensor(ts), torch.tensor(vv),
                         advance_to=T / fs)
        assert e2.head == e1.head and e2.nproc == e1.nproc
        np.testing.assert_allclose(e1.proc[:, :, :e1.nproc].numpy(),
                                   e2.proc[:, :, :e2.nproc].numpy(),
                                   rtol=1e-4, atol=1e-5)

    def test_missing_channel_zero(self):
        S, C, fs = 1, 3, 25.0
        eng = _mk_engine(S, C, fs)
        raw = _dense_raw(S, 2, int(fs * 60 * 15))
        eng.ingest_dense(raw, chan_map=[0, 2])  # channel 1 never fed
        w = eng.windows(batch=1)
        assert w.shape == (1, 1, 3, 120)
        assert (w[0, 0, 1] == 0).all()
        assert (w[0, 0, 0] != 0).any()

    def test_ready_timing(self):
        # First full model window needs 180 s (first processed point) +
        # 595 s more of grid — the reference's ~10-minutes-to-first-prediction.
        fs = 25.0
        eng = _mk_engine(1, 1, fs)
        minute = int(fs * 60)
        mins = 0
        while not eng.ready:
            eng.ingest_dense(_dense_raw(1, 1, minute, seed=mins))
            mins += 1
            assert mins < 20
        assert 12 <= mins <= 14  # 775 s of event time

    def test_window_batch_strides(self):
        S, fs = 1, 25.0
        eng = _mk_engine(S, 1, fs)
        eng.ingest_dense(_dense_raw(S, 1, int(fs * 60 * 30), seed=7))
        w = eng.windows(batch=4, stride=12)
        assert w.shape == (1, 4, 1, 120)
        # window b ends at nproc - (3-b)*12: consecutive windows overlap by 108
        np.testing.assert_allclose(w[0, 0, 0, 12:].numpy(),
                                   w[0, 1, 0, :-12].numpy(), rtol=1e-6)

    def test_ring_wrap_events(self):
        # Push events past one full ring turn; engine must not mix stale data.
        fs = 5.0
        eng = StreamEngine(1, 1, ring_grid=256, fs=fs)
        total_buckets = 600  # > 2 ring turns of 256
        ts = np.arange(0, total_buckets * 5, 1.0 / fs)
        vals = np.sin(ts / 100.0)
        chunk_buckets = 50
        samples_per_bucket = int(5 * fs)
        cs = chunk_buckets * samples_per_bucket
        for i in range(0, len(ts), cs):
            eng.ingest_events(
                torch.zeros(min(cs, len(ts) - i), dtype=torch.long),
                torch.zeros(min(cs, len(ts) - i), dtype=torch.long),
                torch.tensor(ts[i:i + cs]),
                torch.tensor(vals[i:i + cs], dtype=torch.float32),
                advance_to=min((i + cs) / fs, total_buckets * 5))
        want = preprocess_series_oracle(ts, vals, total_buckets)
        got = np.empty(eng.model_win)
        for i, g in enumerate(range(eng.nproc - eng.model_win, eng.nproc)):
            got[i] = eng.proc[0, 0, g % eng.G].item()
        np.testing.assert_allclose(got, want[-eng.model_win:], rtol=1e-4,
                                   atol=1e-5)


@pytest.mark.gpu
class TestStreamEngineGPU:
    def test_gpu_matches_cpu_dense(self):
        S, C, fs = 4, 10, 125.0
        T = int(fs * 60 * 16)
        raw = _dense_raw(S, 8, T, seed=11, nan_frac=0.1)
        cpu = StreamEngine(S, C, ring_grid=1024, fs=fs, device="cpu")
        gpu = StreamEngine(S, C, ring_grid=1024, fs=fs, device="cuda")
        cm = list(range(8))
        cpu.ingest_dense(raw, chan_map=cm)
        gpu.ingest_dense(raw.cuda(), chan_map=cm)
        torch.cuda.synchronize()
        assert gpu.nproc == cpu.nproc
        np.testing.assert_allclose(
            gpu.proc[:, :, :gpu.nproc].cpu().numpy(),
            cpu.proc[:, :, :cpu.nproc].numpy(), rtol=1e-4, atol=1e-5)
        wc = cpu.windows(batch=3, stride=12)
        wg = gpu.windows(batch=3, stride=12)
        np.testing.assert_allclose(wg.cpu().numpy(), wc.numpy(), rtol=1e-4,
                                   atol=1e-5)

    def test_gpu_bf16_dense_with_nans_matches_cpu(self):
        """bf16 dense ingest takes the masked-boundary kernel (mode 5) —
        its NaN-skip/count path must match the CPU oracle fed the same
        bf16-rounded samples (10% NaNs: dropped samples, like wfdb
        invalids)."""
        S, C, fs = 4, 10, 125.0
        T = int(fs * 60 * 8)
        raw = _dense_raw(S, 8, T, seed=17, nan_frac=0.1).to(torch.bfloat16)
        cpu = StreamEngine(S, C, ring_grid=1024, fs=fs, device="cpu")
        gpu = StreamEngine(S, C, ring_grid=1024, fs=fs, device="cuda")
        cm = list(range(8))
        cpu.ingest_dense(raw.float(), chan_map=cm)  # same rounded values
        gpu.ingest_dense(raw.cuda(), chan_map=cm)
        torch.cuda.synchronize()
        assert gpu.nproc == cpu.nproc
        np.testing.assert_allclose(
            gpu.bcnt.cpu().numpy(), cpu.bcnt.numpy())  # NaN counts EXACT
        np.testing.assert_allclose(
            gpu.proc[:, :, :gpu.nproc].cpu().numpy(),
            cpu.proc[:, :, :cpu.nproc].numpy(), rtol=2e-3, atol=1e-3)

    def test_gpu_bf16_windows_and_pipeline(self):
        # Full fused pipeline: raw -> preprocess -> windows -> MyCNN5 engine.
        from tskd_amd.models import build_model
        from tskd_amd.ops import MyCNNEngine
        S, fs = 8, 125.0
        eng = StreamEngine(S, 10, ring_grid=2048, fs=fs, device="cuda")
        raw = _dense_raw(S, 8, int(fs * 60 * 16), seed=13).cuda()
        eng.ingest_dense(raw, chan_map=list(range(8)))
        w = eng.windows(batch=2, stride=12, dtype=torch.bfloat16)
        assert w.shape == (S, 2, 10, 120)
        model = build_model("MyCNN5").eval()
        me = MyCNNEngine(model, device="cuda")
        probs = me.forward(w, apply_sigmoid=True)
        torch.cuda.synchronize()
        assert probs.shape == (S, 2)
        assert torch.isfinite(probs).all()

    def test_gpu_events_match_cpu(self):
        fs = 5.0
        rng = np.random.default_rng(17)
        n = 4000
        ts = rng.uniform(0, 1000, n)
        si = rng.integers(0, 2, n)
        ci = rng.integers(0, 3, n)
        vv = rng.normal(size=n).astype(np.float32)
        args = (torch.tensor(si, dtype=torch.long), torch.tensor(ci, dtype=torch.long),
                torch.tensor(ts), torch.tensor(vv))
        cpu = StreamEngine(2, 3, ring_grid=512, fs=fs, device="cpu")
        gpu = StreamEngine(2, 3, ring_grid=512, fs=fs, device="cuda")
        cpu.ingest_events(*args, advance_to=1000)
        gpu.ingest_events(*args, advance_to=1000)
        torch.cuda.synchronize()
        np.testing.assert_allclose(
            gpu.proc[:, :, :gpu.nproc].cpu().numpy(),
            cpu.proc[:, :, :cpu.nproc].numpy(), rtol=1e-3, atol=1e-4)


@pytest.mark.gpu
class TestTriggerGraph:
    def test_graphed_trigger_matches_eager(self):
        """Whole-trigger hipGraph (ingest+fill+gather+model) vs an eager twin
        engine fed the same raw chunks."""
        from tskd_amd.engine.stream_engine import TriggerGraph
        from tskd_amd.models import build_model
        from tskd_amd.ops import GraphedForward, MyCNNEngine
        fs = 125.0
        S = 8
        torch.manual_seed(5)
        model = build_model("MyCNN5").eval()
        me = MyCNNEngine(model, device="cuda")
        chunks = [torch.randn(S, 8, int(fs * 60), device="cuda",
                              dtype=torch.bfloat16) for _ in range(20)]
        cm = list(range(8))

        # eager twin
        e1 = StreamEngine(S, 10, ring_grid=512, fs=fs, device="cuda")
        eager_probs = []
        for ch in chunks:
            e1.ingest_dense(ch, chan_map=cm)
            w = e1.windows(batch=1, stride=12, dtype=torch.bfloat16)
            eager_probs.append(me.forward(
                w, torch.full((S, 1), 65.0, device="cuda"),
                apply_sigmoid=True).clone())

        # graphed engine: warm 14 triggers eagerly, capture on the 15th
        e2 = StreamEngine(S, 10, ring_grid=512, fs=fs, device="cuda")
        raw_buf = chunks[0].clone()
        idx = 0
        while e2.nproc == 0 or e2.nproc < e2.head - e2.win_buckets + 1:
            raw_buf.copy_(chunks[idx])
            e2.ingest_dense(raw_buf, chan_map=cm)
            idx += 1
        torch.cuda.synchronize()
        gf = GraphedForward(me, s=S, n=1, dtype=torch.bfloat16, timelast=True)
        raw_buf.copy_(chunks[idx])  # consumed by TriggerGraph's warm run
        tg = TriggerGraph(e2, raw_buf, cm, gf, stride=12)
        idx += 1
        while idx < len(chunks):
            raw_buf.copy_(chunks[idx])
            out = tg.replay()
            torch.cuda.synchronize()
            torch.testing.assert_close(out, eager_probs[idx], rtol=2e-3,
                                       atol=2e-3)
            idx += 1
        assert e2.head == e1.head and e2.nproc == e1.nproc

    def test_trigger_graph_survives_ring_wrap(self):
        """Replay the captured trigger past several full ring turns: the
        device-side mod-G indexing must keep matching the eager twin."""
        from tskd_amd.engine.stream_engine import TriggerGraph
        from tskd_amd.models import build_model
        from tskd_amd.ops import GraphedForward, MyCNNEngine
        fs, S, G = 25.0, 4, 192  # 12 buckets/trigger -> wraps every 16
        torch.manual_seed(6)
        me = MyCNNEngine(build_model("MyCNN5").eval(), device="cuda")
        gen = torch.Generator(device="cuda").manual_seed(7)
        n_triggers = 60  # ~3.75 ring turns
        chunks = [torch.randn(S, 8, int(fs * 60), device="cuda",
                              generator=gen, dtype=torch.float32)
                  .to(torch.bfloat16) for _ in range(n_triggers)]
        cm = list(range(8))
        e1 = StreamEngine(S, 10, ring_grid=G, fs=fs, device="cuda")
        eager = []
        for ch in chunks:
            e1.ingest_dense(ch, chan_map=cm)
            w = e1.windows(batch=1, stride=12, dtype=torch.bfloat16)
            eager.append(me.forward(w, None, apply_sigmoid=True).clone())
        e2 = StreamEngine(S, 10, ring_grid=G, fs=fs, device="cuda")
        raw_buf = chunks[0].clone()
        idx = 0
        while e2.nproc == 0 or e2.nproc < e2.head - e2.win_buckets + 1:
            raw_buf.copy_(chunks[idx]); e2.ingest_dense(raw_buf, chan_map=cm)
            idx += 1
        torch.cuda.synchronize()
        gf = GraphedForward(me, s=S, n=1, dtype=torch.bfloat16, timelast=True)
        gf.age.zero_()
        raw_buf.copy_(chunks[idx])
        tg = TriggerGraph(e2, raw_buf, cm, gf, stride=12)
        idx += 1
        while idx < n_triggers:
            raw_buf.copy_(chunks[idx])
            out = tg.replay()
            torch.cuda.synchronize()
            torch.testing.assert_close(out, eager[idx], rtol=2e-3, atol=2e-3)
            idx += 1


@pytest.mark.gpu
class TestTriggerGraphOverlap:
    def test_overlap_matches_eager_with_refresh(self):
        """Two-stream overlapped TriggerGraph, raw refreshed per trigger
        (raw_refresh=True serializes the refresh against ingest): outputs
        must match the eager twin exactly per trigger."""
        from tskd_amd.engine.stream_engine import TriggerGraph
        from tskd_amd.models import build_model
        from tskd_amd.ops import GraphedForward, MyCNNEngine
        fs, S = 125.0, 8
        torch.manual_seed(11)
        me = MyCNNEngine(build_model("MyCNN5").eval(), device="cuda")
        chunks = [torch.randn(S, 8, int(fs * 60), device="cuda",
                              dtype=torch.bfloat16) for _ in range(20)]
        cm = list(range(8))
        e1 = StreamEngine(S, 10, ring_grid=512, fs=fs, device="cuda")
        eager = []
        for ch in chunks:
            e1.ingest_dense(ch, chan_map=cm)
            w = e1.windows(batch=1, stride=12, dtype=torch.bfloat16)
            eager.append(me.forward(
                w, torch.full((S, 1), 65.0, device="cuda"),
                apply_sigmoid=True).clone())
        e2 = StreamEngine(S, 10, ring_grid=512, fs=fs, device="cuda")
        raw_buf = chunks[0].clone()
        idx = 0
        while e2.nproc == 0 or e2.nproc < e2.head - e2.win_buckets + 1:
            raw_buf.copy_(chunks[idx])
            e2.ingest_dense(raw_buf, chan_map=cm)
            idx += 1
        torch.cuda.synchronize()
        gf = GraphedForward(me, s=S, n=1, dtype=torch.bfloat16, timelast=True)
        raw_buf.copy_(chunks[idx])  # consumed by the warm trigger
        tg = TriggerGraph(e2, raw_buf, cm, gf, stride=12, overlap=True)
        idx += 1
        while idx < len(chunks):
            raw_buf.copy_(chunks[idx])
            out = tg.replay(raw_refresh=True)
            torch.cuda.synchronize()
            torch.testing.assert_close(out, eager[idx], rtol=2e-3,
                                       atol=2e-3)
            idx += 1
        assert e2.head == e1.head and e2.nproc == e1.nproc

    def test_overlap_pipelined_constant_raw_ring_wrap(self):
        """Fully-pipelined replays (no refresh, no per-trigger sync —
        ingest(T+1) genuinely overlaps model(T)) across several ring turns
        with constant raw: every trigger's output must equal the eager
        twin's — any ingest/fill race would corrupt it."""
        from tskd_amd.engine.stream_engine import TriggerGraph
        from tskd_amd.models import build_model
        from tskd_amd.ops import GraphedForward, MyCNNEngine
        fs, S, G = 25.0, 4, 192
        torch.manual_seed(12)
        me = MyCNNEngine(build_model("MyCNN5").eval(), device="cuda")
        chunk = torch.randn(S, 8, int(fs * 60), device="cuda",
                            dtype=torch.bfloat16)
        cm = list(range(8))
        n_triggers = 56  # ~3.5 ring turns at 12 buckets/trigger, G=192
        e1 = StreamEngine(S, 10, ring_grid=G, fs=fs, device="cuda")
        eager = []
        for _ in range(n_triggers):
            e1.ingest_dense(chunk, chan_map=cm)
            w = e1.windows(batch=1, stride=12, dtype=torch.bfloat16)
            eager.append(me.forward(w, None, apply_sigmoid=True).clone())
        e2 = StreamEngine(S, 10, ring_grid=G, fs=fs, device="cuda")
        idx = 0
        while e2.nproc == 0 or e2.nproc < e2.head - e2.win_buckets + 1:
            e2.ingest_dense(chunk, chan_map=cm)
            idx += 1
        torch.cuda.synchronize()
        gf = GraphedForward(me, s=S, n=1, dtype=torch.bfloat16, timelast=True)
        gf.age.zero_()
        tg = TriggerGraph(e2, chunk, cm, gf, stride=12, overlap=True)
        idx += 1
        outs = []
        while idx < n_triggers:
            outs.append((idx, tg.replay().clone()))  # clone on cur: ordered
            idx += 1
        torch.cuda.synchronize()
        for i, out in outs:
            torch.testing.assert_close(out, eager[i], rtol=2e-3, atol=2e-3)
        assert e2.head == e1.head and e2.nproc == e1.nproc


class TestZNormalize:
    def test_cpu_znorm(self):
        eng = _mk_engine(2, 3, 25.0)
        eng.ingest_dense(_dense_raw(2, 3, int(25 * 60 * 20), seed=9))
        w = eng.windows(batch=2, stride=12, znormalize=True)
        m = w.mean(dim=-1)
        sd = w.std(dim=-1, unbiased=False)
        assert m.abs().max() < 1e-4
        np.testing.assert_allclose(sd.numpy(), 1.0, atol=1e-3)

    @pytest.mark.gpu
    def test_gpu_matches_cpu_znorm(self):
        S, C, fs = 2, 4, 125.0
        raw = _dense_raw(S, C, int(fs * 60 * 16), seed=10)
        cpu = StreamEngine(S, C, ring_grid=1024, fs=fs, device="cpu")
        gpu = StreamEngine(S, C, ring_grid=1024, fs=fs, device="cuda")
        cpu.ingest_dense(raw)
        gpu.ingest_dense(raw.cuda())
        wc = cpu.windows(batch=2, stride=12, znormalize=True)
        wg = gpu.windows(batch=2, stride=12, znormalize=True)
        torch.cuda.synchronize()
        np.testing.assert_allclose(wg.cpu().numpy(), wc.numpy(), rtol=1e-3,
                                   atol=1e-4)


@pytest.mark.gpu
class TestHeavyEventLoad:  # noqa: E302
    def test_half_million_events_match_cpu(self):
        """Bulk atomically-ingested events (the sparse path under load)."""
        fs = 5.0
        rng = np.random.default_rng(23)
        n = 500_000
        S, C = 64, 10
        ts = rng.uniform(0, 2000, n)
        si = rng.integers(0, S, n)
        ci = rng.integers(0, C, n)
        vv = rng.normal(80, 10, n).astype(np.float32)
        args = (torch.tensor(si, dtype=torch.long),
                torch.tensor(ci, dtype=torch.long),
                torch.tensor(ts), torch.tensor(vv))
        gpu = StreamEngine(S, C, ring_grid=1024, fs=fs, device="cuda")
        gpu.ingest_events(*args, advance_to=2000)
        torch.cuda.synchronize()
        # spot-check 8 (stream, channel) pairs against the numpy oracle
        for s_ in (0, 17, 40, 63):
            for c_ in (0, 7):
                m = (si == s_) & (ci == c_)
                want = preprocess_series_oracle(ts[m], vv[m], gpu.head)
                got = gpu.proc[s_, c_, :gpu.nproc].cpu().numpy()
                np.testing.assert_allclose(got, want[:gpu.nproc], rtol=2e-4,
                                           atol=2e-4)

    def test_chunked_catchup_gpu_matches_oracle(self):
        """A backlog far wider than the ring replays correctly in chunks;
        the retained tail matches the oracle."""
        fs = 5.0
        G = 256
        eng = StreamEngine(1, 1, ring_grid=G, fs=fs, device="cuda")
        total_buckets = 900  # ~4 ring turns in ONE call
        ts = np.arange(0, total_buckets * 5, 1.0 / fs)
        vals = np.cos(ts / 77.0) * 50 + 80
        eng.ingest_events_chunked(
            torch.zeros(len(ts), dtype=torch.long),
            torch.zeros(len(ts), dtype=torch.long),
            torch.tensor(ts), torch.tensor(vals, dtype=torch.float32),
            advance_to=total_buckets * 5)
        torch.cuda.synchronize()
        want = preprocess_series_oracle(ts, vals, total_buckets)
        got = np.empty(eng.model_win)
        for i, g in enumerate(range(eng.nproc - eng.model_win, eng.nproc)):
            got[i] = eng.proc[0, 0, g % eng.G].item()
        np.testing.assert_allclose(got, want[-eng.model_win:], rtol=1e-4,
                                   atol=1e-4)
