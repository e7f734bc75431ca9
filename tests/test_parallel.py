"""Distributed-path tests: gloo world_size=2 (CPU), sharding, supervisor.

The multi-GPU RCCL path shares these exact call sites (backend string is the
only difference); the driver's round-end scaling bench exercises RCCL.
"""

import json
import multiprocessing as mp
import os
import subprocess
import sys
import time

import numpy as np
import pytest
import torch

from tskd_amd.parallel import shard_for_key, shard_streams
from tskd_amd.parallel.supervisor import (Supervisor, load_offsets,
                                          restore_consumer, save_offsets)

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


class TestSharding:
    def test_deterministic_and_balanced(self):
        keys = [f"p{i:06d}" for i in range(2000)]
        ranks = [shard_for_key(k, 8) for k in keys]
        assert ranks == [shard_for_key(k, 8) for k in keys]  # stable
        counts = np.bincount(ranks, minlength=8)
        assert counts.min() > 150  # roughly balanced
        assert sum(counts) == 2000

    def test_shard_streams_partition(self):
        keys = [f"p{i:06d}" for i in range(100)]
        shards = [shard_streams(keys, r, 4) for r in range(4)]
        flat = sorted(k for s in shards for k in s)
        assert flat == sorted(keys)  # disjoint + complete

    def test_world_one(self):
        assert shard_for_key("p000194", 1) == 0


def _dp_worker(rank, world, port, tmpdir):
    os.environ.update({
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
        "RANK": str(rank), "WORLD_SIZE": str(world),
    })
    import torch.distributed as dist

    from tskd_amd.models import build_model
    from tskd_amd.parallel import DPServing, all_gather_predictions
    dist.init_process_group("gloo")
    torch.manual_seed(0)  # same model weights on every rank
    model = build_model("MyCNN5").eval()
    dp = DPServing(streams_per_rank=2, model=model, device="cpu", fs=25.0)
    g = torch.Generator().manual_seed(100 + rank)  # different data per rank
    raw = torch.randn(2, 8, int(25 * 60 * 16), generator=g)
    gathered = dp.step(raw, chan_map=list(range(8)))
    assert gathered.shape == (world, 2)
    np.save(os.path.join(tmpdir, f"gathered_{rank}.npy"), gathered.numpy())
    # direct collective check too
    t = torch.full((3,), float(rank))
    out = all_gather_predictions(t)
    assert out.shape == (world, 3) and out[1, 0] == 1.0
    dist.destroy_process_group()


class TestGlooDP:
    @pytest.mark.parametrize("world,port", [(2, 29531), (4, 29533)])
    def test_dp_serving(self, tmp_path, world, port):
        ctx = mp.get_context("spawn")
        ps = [ctx.Process(target=_dp_worker,
                          args=(r, world, port, str(tmp_path)))
              for r in range(world)]
        for p in ps:
            p.start()
        for p in ps:
            p.join(120)
            assert p.exitcode == 0
        g0 = np.load(tmp_path / "gathered_0.npy")
        g1 = np.load(tmp_path / "gathered_1.npy")
        # every rank sees the identical full prediction set
        np.testing.assert_allclose(g0, g1)
        assert np.isfinite(g0).all() and (g0 >= 0).all() and (g0 <= 1).all()
        # ranks had different data => different predictions
        assert not np.allclose(g0[0], g0[1])


class TestSupervisor:
    def test_restarts_until_success(self, tmp_path):
        marker = tmp_path / "attempts"
        script = (
            "import os,sys,pathlib\n"
            f"p = pathlib.Path({str(marker)!r})\n"
            "n = int(p.read_text()) if p.exists() else 0\n"
            "p.write_text(str(n+1))\n"
            "sys.exit(0 if n >= 2 else 1)\n"
        )
        sup = Supervisor([sys.executable, "-c", script], max_restarts=5,
                         backoff_s=0.01)
        assert sup.run() == 0
        assert sup.restarts == 2

    def test_gives_up(self, tmp_path):
        sup = Supervisor([sys.executable, "-c", "import sys; sys.exit(3)"],
                         max_restarts=2, backoff_s=0.01)
        assert sup.run() == 3

    def test_offsets_roundtrip(self, tmp_path):
        path = str(tmp_path / "off.json")
        save_offsets(path, {"HR/0": 128, "call-stream/2": 4096})
        assert load_offsets(path) == {"HR/0": 128, "call-stream/2": 4096}

    def test_restart_resumes_no_loss(self, tmp_path):
        """Kill-and-respawn drill: messages produced while the consumer is
        down are NOT lost (vs the reference's startingOffsets=latest)."""
        from tskd_amd.bus import Bus, Consumer, Producer
        bus = Bus(str(tmp_path / "bus"))
        bus.create_topic("t")
        prod = Producer(bus)
        offsets = str(tmp_path / "off.json")

        # first consumer session: read 3, persist offsets, "crash"
        c1 = Consumer(bus, starting="earliest")
        c1.subscribe(["t"])
        for i in range(3):
            prod.produce("t", "k", f"v{i}")
        assert len(c1.poll(timeout_ms=500)) == 3
        save_offsets(offsets, c1.positions())
        del c1

        # messages arrive while the stage is down
        for i in range(3, 6):
            prod.produce("t", "k", f"v{i}")

        # respawned consumer resumes from persisted offsets
        c2 = Consumer(bus, starting="latest")  # latest would lose v3..v5
        c2.subscribe(["t"])
        restored = restore_consumer(c2, offsets)
        assert restored == 1
        msgs = c2.poll(timeout_ms=500)
        assert [m.value for m in msgs] == [b"v3", b"v4", b"v5"]


class TestCliOffsetFlow:
    def test_processstream_offsets_file(self, tmp_path):
        """Run the processstream CLI twice with --offsets-file; the second
        run must resume, not reprocess (subprocess smoke of the real CLI)."""
        bus_dir = str(tmp_path / "bus")
        off = str(tmp_path / "off.json")
        from tskd_amd.bus import Bus, Producer
        bus = Bus(bus_dir)
        bus.create_topic("HR")
        p = Producer(bus)
        for i in range(100):
            p.produce("HR", "p000194", json.dumps([0, float(i)]),
                      ts_us=int(i * 60e6))
        env = dict(os.environ, PYTHONPATH=REPO)
        cmd = [sys.executable, "-m", "tskd_amd.cli.processstream",
               "--bus-dir", bus_dir, "--starting", "earliest",
               "--offsets-file", off, "--max-triggers", "1",
               "--device", "cpu", "--speed", "1e9"]
        r = subprocess.run(cmd, env=env, capture_output=True, text=True,
                           timeout=120)
        assert r.returncode == 0, r.stderr[-2000:]
        pos1 = load_offsets(off)
        assert any(v > 0 for v in pos1.values())
        r2 = subprocess.run(cmd, env=env, capture_output=True, text=True,
                            timeout=120)
        assert r2.returncode == 0, r2.stderr[-2000:]
        assert load_offsets(off) == pos1  # nothing new to consume


def _fused_dp_worker(rank, world, port, tmpdir, wavef):
    os.environ.update({
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
        "RANK": str(rank), "WORLD_SIZE": str(world),
    })
    import torch.distributed as dist

    from tskd_amd.bus import Bus
    from tskd_amd.cli.serve import FusedServer
    from tskd_amd.cli.sendstream import send_record_data
    from tskd_amd.config import GlobalConfig
    from tskd_amd.models import build_model
    from tskd_amd.store import PredictionStore
    dist.init_process_group("gloo")
    cfg = GlobalConfig()
    cfg.wavef_path = wavef
    cfg.channel_names = ["HR", "RESP", "PULSE", "SpO2"]
    torch.manual_seed(0)
    model = build_model("MyCNN5").eval()
    bus = Bus(os.path.join(tmpdir, f"bus_{rank}"))
    store = PredictionStore(os.path.join(tmpdir, "pred.log"))  # SHARED
    srv = FusedServer(bus, cfg, store, model=model, device="cpu",
                      max_streams=8, ring_grid=1024, starting="earliest",
                      rank=rank, world=world)
    # every rank sees every patient's messages; sharding filters them
    for rec in ("p000194-test", "p000007-test", "p000021-test",
                "p000042-test"):
        send_record_data(bus, rec, None, 1e6, 1.0, cfg)
    n = srv.trigger()
    dist.barrier()
    with open(os.path.join(tmpdir, f"rank{rank}.n"), "w") as f:
        f.write(str(n))
    dist.destroy_process_group()


class TestFusedServeDP:
    def test_sharded_serving_world2(self, tmp_path):
        """Two fused-server ranks over the same 4 patients: disjoint shards,
        one shared store, complete coverage."""
        import numpy as np

        from tskd_amd.parallel import shard_for_key
        from tskd_amd.store import PredictionStore
        sys.path.insert(0, REPO)
        from tests.test_pipeline import _write_wfdb_record
        wavef = str(tmp_path / "wavef")
        pids = ["p000194", "p000007", "p000021", "p000042"]
        for pid in pids:
            _write_wfdb_record(wavef, f"{pid}-test",
                               ["HR", "RESP", "PULSE", "SpO2"], 1 / 60, 40)
        ctx = mp.get_context("spawn")
        ps = [ctx.Process(target=_fused_dp_worker,
                          args=(r, 2, 29537, str(tmp_path), wavef))
              for r in range(2)]
        for p in ps:
            p.start()
        for p in ps:
            p.join(180)
            assert p.exitcode == 0
        counts = [int(open(tmp_path / f"rank{r}.n").read()) for r in range(2)]
        expect = [len([p_ for p_ in pids if shard_for_key(p_, 2) == r])
                  for r in range(2)]
        assert counts == expect and sum(counts) == 4
        st = PredictionStore(str(tmp_path / "pred.log"))
        assert st.count() == 4  # both ranks persisted into ONE store
        got = {r[0] for r in st.tail(10)}
        assert got == set(pids)


class TestRetentionFlow:
    def test_processstream_trim_consumed(self, tmp_path):
        """--trim-consumed reclaims raw-topic storage behind the consumer."""
        bus_dir = str(tmp_path / "bus")
        from tskd_amd.bus import Bus, Producer
        bus = Bus(bus_dir)
        bus.create_topic("HR")
        p = Producer(bus)
        blob = 600  # messages ~ 40 B each
        for i in range(20000):
            p.produce("HR", "p000194", json.dumps([0, float(i % blob)]),
                      ts_us=int(i * 60e6))
        end_before = bus.end_offset("HR", 0)
        env = dict(os.environ, PYTHONPATH=REPO)
        cmd = [sys.executable, "-m", "tskd_amd.cli.processstream",
               "--bus-dir", bus_dir, "--starting", "earliest",
               "--trim-consumed", "--max-triggers", "5",
               "--device", "cpu", "--speed", "1e9", "--max-streams", "4"]
        r = subprocess.run(cmd, env=env, capture_output=True, text=True,
                           timeout=180)
        assert r.returncode == 0, r.stderr[-2000:]
        assert bus.trim_offset("HR", 0) == end_before  # fully consumed+trimmed
        assert bus.end_offset("HR", 0) == end_before   # offsets stable
