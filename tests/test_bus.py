"""Bus tests: keyed topics, offsets, replay, cross-process delivery."""

import json
import multiprocessing as mp
import os
import time

import pytest

from tskd_amd.bus import Bus, Consumer, Producer


@pytest.fixture
def bus(tmp_path):
    return Bus(str(tmp_path / "bus"))


class TestBasic:
    def test_produce_consume(self, bus):
        bus.create_topic("HR")
        c = Consumer(bus, starting="earliest")
        c.subscribe(["HR"])
        p = Producer(bus)
        p.produce_sample("HR", "p000194", 0, 71.5)
        p.produce_sample("HR", "p000194", 0, 72.0)
        msgs = c.poll(timeout_ms=1000)
        assert len(msgs) == 2
        assert msgs[0].key == b"p000194"
        ch, val = json.loads(msgs[0].value)
        assert ch == 0 and val == 71.5
        assert msgs[0].seq == 0 and msgs[1].seq == 1

    def test_starting_latest_skips_history(self, bus):
        bus.create_topic("t")
        p = Producer(bus)
        p.produce("t", "k", "old")
        c = Consumer(bus, starting="latest")
        c.subscribe(["t"])
        assert c.poll(timeout_ms=10) == []
        p.produce("t", "k", "new")
        msgs = c.poll(timeout_ms=1000)
        assert [m.value for m in msgs] == [b"new"]

    def test_replay_from_offset(self, bus):
        bus.create_topic("t")
        p = Producer(bus)
        for i in range(5):
            p.produce("t", "k", f"v{i}")
        c = Consumer(bus, starting="earliest")
        c.subscribe(["t"])
        msgs = c.poll()
        assert len(msgs) == 5
        # replay from message 2's byte offset
        c2 = Consumer(bus)
        c2.seek("t", 0, msgs[2].offset)
        replay = c2.poll(timeout_ms=500)
        assert [m.value for m in replay] == [b"v2", b"v3", b"v4"]

    def test_keyed_partitions(self, bus):
        bus.create_topic("multi", nparts=4)
        p = Producer(bus)
        keys = [f"p{i:06d}" for i in range(20)]
        for k in keys:
            for j in range(3):
                p.produce("multi", k, f"{k}:{j}")
        c = Consumer(bus, starting="earliest")
        c.subscribe(["multi"])
        msgs = c.poll(max_msgs=100)
        assert len(msgs) == 60
        # same key always lands in the same partition, order preserved
        by_key = {}
        for m in msgs:
            by_key.setdefault(m.key, []).append((m.partition, m.value))
        for k, lst in by_key.items():
            parts = {p_ for p_, _ in lst}
            assert len(parts) == 1
            assert [v.decode().split(":")[1] for _, v in lst] == ["0", "1", "2"]
        assert len({lst[0][0] for lst in by_key.values()}) > 1  # spread out

    def test_grow_beyond_initial_capacity(self, bus):
        bus.create_topic("big")
        p = Producer(bus)
        blob = "x" * 10000
        for i in range(300):  # ~3 MB > 1 MB initial capacity
            p.produce("big", f"k{i}", blob)
        c = Consumer(bus, starting="earliest")
        c.subscribe(["big"])
        n = 0
        while True:
            msgs = c.poll(max_msgs=128, timeout_ms=100)
            if not msgs:
                break
            n += len(msgs)
        assert n == 300

    def test_producer_callback(self, bus):
        bus.create_topic("cb")
        p = Producer(bus)
        acked = []
        p.produce("cb", "k", "v", callback=lambda err, info: acked.append((err, info)))
        assert acked == [(None, ("cb", "k"))]


def _producer_proc(bus_dir, topic, n, tag):
    b = Bus(bus_dir)
    p = Producer(b)
    for i in range(n):
        p.produce(topic, f"key{tag}", f"{tag}:{i}")


class TestCrossProcess:
    def test_two_producer_processes(self, tmp_path):
        bus_dir = str(tmp_path / "xbus")
        b = Bus(bus_dir)
        b.create_topic("xp")
        c = Consumer(b, starting="earliest")
        c.subscribe(["xp"])
        ctx = mp.get_context("spawn")
        ps = [ctx.Process(target=_producer_proc, args=(bus_dir, "xp", 50, t))
              for t in ("A", "B")]
        for p in ps:
            p.start()
        for p in ps:
            p.join(30)
            assert p.exitcode == 0
        got = []
        for _ in range(50):
            msgs = c.poll(max_msgs=256, timeout_ms=200)
            got.extend(msgs)
            if len(got) >= 100:
                break
        assert len(got) == 100
        # per-key FIFO across processes
        for tag in ("A", "B"):
            vals = [m.value.decode() for m in got if m.key.decode() == f"key{tag}"]
            assert vals == [f"{tag}:{i}" for i in range(50)]


class TestNativeParse:
    def test_poll_samples(self, bus):
        bus.create_topic("HR")
        c = Consumer(bus, starting="earliest")
        c.subscribe(["HR"])
        p = Producer(bus)
        p.produce_sample("HR", "p000194", 0, 71.5, ts_us=5_000_000)
        p.produce("HR", "p000194", "not json")  # skipped, offset advances
        p.produce_sample("HR", "p000194", 3, -2.25, ts_us=6_000_000)
        keys, topics, chans, vals, ts = c.poll_samples(timeout_ms=500)
        assert keys == ["p000194", "p000194"]
        assert topics == ["HR", "HR"]
        assert chans.tolist() == [0, 3]
        assert vals.tolist() == [71.5, -2.25]
        assert ts.tolist() == [5.0, 6.0]
        # the malformed message was consumed too (no re-delivery)
        assert c.poll(timeout_ms=10) == []


class TestRetention:
    def test_trim_reclaims_and_preserves_offsets(self, bus, tmp_path):
        bus.create_topic("ret")
        p = Producer(bus)
        blob = "y" * 8000
        for i in range(200):  # ~1.6 MB
            p.produce("ret", f"k{i}", blob)
        c = Consumer(bus, starting="earliest")
        c.subscribe(["ret"])
        msgs = c.poll(max_msgs=200, timeout_ms=500)
        assert len(msgs) == 200
        import os
        part_file = None
        for root, _, files in os.walk(bus.dir):
            for f in files:
                if f == "p0.log" and "ret" in root:
                    part_file = os.path.join(root, f)
        blocks_before = os.stat(part_file).st_blocks
        cut = msgs[150].offset
        applied = bus.trim_topic("ret", 0, cut)
        assert 0 < applied <= cut
        blocks_after = os.stat(part_file).st_blocks
        assert blocks_after < blocks_before  # storage actually reclaimed
        # new earliest consumer starts at the trim point, offsets unchanged
        c2 = Consumer(bus, starting="earliest")
        c2.subscribe(["ret"])
        survivors = c2.poll(max_msgs=200, timeout_ms=500)
        assert survivors
        assert survivors[0].offset >= applied
        assert survivors[-1].offset == msgs[-1].offset  # absolute offsets
        keys = [m.key.decode() for m in survivors]
        assert keys == [f"k{i}" for i in range(200 - len(keys), 200)]

    def test_consumer_mid_gap_skips_forward(self, bus):
        bus.create_topic("ret2")
        p = Producer(bus)
        for i in range(50):
            p.produce("ret2", "k", "x" * 4000)
        c = Consumer(bus, starting="earliest")
        c.subscribe(["ret2"])
        first = c.poll(max_msgs=5, timeout_ms=200)  # position now at msg 5
        side = Consumer(bus, starting="earliest")
        side.subscribe(["ret2"])
        all_msgs = side.poll(max_msgs=100, timeout_ms=200)
        bus.trim_topic("ret2", 0, all_msgs[25].offset)  # a record boundary
        rest = c.poll(max_msgs=100, timeout_ms=200)
        # consumer skipped the retention gap and continued
        assert rest
        assert rest[0].offset >= bus.trim_offset("ret2")
        assert first[-1].next_offset < rest[0].offset


class TestConcurrentStress:
    def test_threaded_producers_single_consumer(self, bus):
        """4 producer threads x 500 keyed messages against one consumer —
        at-least-once, per-key ordering preserved (the bus's Kafka-style
        contract under the GIL-free pybind produce path)."""
        import threading
        bus.create_topic("t")
        NT, NM = 4, 500
        errs = []

        def run(tid):
            try:
                p = Producer(bus)
                for i in range(NM):
                    p.produce("t", f"k{tid}", f"{tid}:{i}")
                p.flush()
            except Exception as e:  # pragma: no cover
                errs.append(e)

        ts = [threading.Thread(target=run, args=(t,)) for t in range(NT)]
        for t in ts:
            t.start()
        for t in ts:
            t.join()
        assert not errs
        c = Consumer(bus, starting="earliest")
        c.subscribe(["t"])
        seen = {t: [] for t in range(NT)}
        while True:
            msgs = c.poll(max_msgs=4096, timeout_ms=100)
            if not msgs:
                break
            for m in msgs:
                tid, i = m.value.decode().split(":")
                seen[int(tid)].append(int(i))
        for t in range(NT):
            assert seen[t] == list(range(NM)), f"thread {t} order broken"

    def test_produce_consume_trim_interleaved(self, bus):
        """Producer + consumer + periodic retention trims running together:
        the consumer sees every message exactly once in order (trims only
        ever remove what it has already consumed)."""
        bus.create_topic("t")
        p = Producer(bus)
        c = Consumer(bus, starting="earliest")
        c.subscribe(["t"])
        got = []
        last_off = 0
        for i in range(300):
            p.produce("t", "k", str(i))
            if i % 7 == 0:
                for m in c.poll(max_msgs=64, timeout_ms=10):
                    got.append(int(m.value))
                    last_off = m.offset + 1
            if i % 50 == 49 and last_off:
                bus.trim_topic("t", 0, last_off)
        for m in c.poll(max_msgs=1024, timeout_ms=100):
            got.append(int(m.value))
        assert got == list(range(300))


class TestConcurrentGrowAndPoll:
    def test_grow_remap_under_gil_free_poll(self, bus):
        """Regression (round 2): poll releases the GIL, so a producer
        thread can grow+remap the partition mmap WHILE the consumer is
        mid-scan — without PartMap's in-process op lock this segfaulted
        (observed once on a GPU box). 8 KB values force several 2x file
        growths past the 1 MiB initial mapping during a live poll loop."""
        import threading
        bus.create_topic("big")
        p = Producer(bus)
        c = Consumer(bus, starting="earliest")
        c.subscribe(["big"])
        N = 1200
        payload = "x" * 8192

        def producer():
            for i in range(N):
                p.produce("big", f"k{i % 7}", f"[{i % 10}, {float(i)}]"
                          + payload[: 8000], ts_us=i + 1)

        th = threading.Thread(target=producer)
        th.start()
        got = 0
        deadline = time.time() + 60
        while got < N and time.time() < deadline:
            sa, ca, va, ta, nk = c.poll_samples_sid(max_msgs=4096,
                                                    timeout_ms=50)
            got += len(sa)
        th.join()
        assert got == N  # every message survived concurrent growth


class TestTcpRelay:
    """Node-edge TCP transport over the mmap bus (bus/relay.py)."""

    def test_remote_produce_and_subscribe(self, bus):
        import json
        from tskd_amd.bus.relay import (RelayConsumer, RelayProducer,
                                        RelayServer)
        srv = RelayServer(bus, port=0)
        try:
            rp = RelayProducer("127.0.0.1", srv.port)
            for i in range(20):
                rp.produce("HR", "p000194", json.dumps([0, float(i)]),
                           ts_us=i)
            # remote messages land in the LOCAL log (durability point)
            import time
            lc = Consumer(bus, starting="earliest")
            lc.subscribe(["HR"])
            got = []
            for _ in range(50):
                got += lc.poll(max_msgs=64, timeout_ms=100)
                if len(got) == 20:
                    break
            assert len(got) == 20
            assert got[0].key == b"p000194"
            # a remote subscriber tails them back out over TCP
            rc = RelayConsumer("127.0.0.1", srv.port, ["HR"],
                               starting="earliest")
            frames = []
            for _ in range(50):
                frames += rc.poll(max_msgs=64, timeout_s=0.5)
                if len(frames) == 20:
                    break
            assert [json.loads(f["value"])[1] for f in frames] == \
                [float(i) for i in range(20)]
            assert frames[0]["topic"] == "HR" and frames[0]["ts_us"] == 0
            rc.close()
            rp.close()
        finally:
            srv.close()

    def test_relay_feeds_pipeline_consumer(self, bus):
        """A remote producer's samples are indistinguishable from local
        ones: the native poll_samples fast path parses them."""
        import json
        from tskd_amd.bus.relay import RelayProducer, RelayServer
        srv = RelayServer(bus, port=0)
        try:
            rp = RelayProducer("127.0.0.1", srv.port)
            for i in range(10):
                rp.produce("SpO2", "p044083", json.dumps([4, 97.0 + i]),
                           ts_us=i * 1000)
            c = Consumer(bus, starting="earliest")
            c.subscribe(["SpO2"])
            keys, chans, vals = [], [], []
            for _ in range(50):
                k, t, ch, v, ts = c.poll_samples(max_msgs=64, timeout_ms=100)
                keys += list(k)
                chans += list(ch)
                vals += list(v)
                if len(keys) == 10:
                    break
            assert len(keys) == 10
            assert chans == [4] * 10
            assert vals[-1] == 106.0
            rp.close()
        finally:
            srv.close()

    def test_relay_unicode_and_large_values(self, bus):
        """Framing robustness: multi-frame-sized values and non-ASCII keys
        round-trip exactly."""
        from tskd_amd.bus.relay import (RelayConsumer, RelayProducer,
                                        RelayServer)
        srv = RelayServer(bus, port=0)
        try:
            rp = RelayProducer("127.0.0.1", srv.port)
            big = "x" * 300000  # > one socket buffer
            rp.produce("t", "péti€nt-1", big)
            rp.produce("t", "k", "sm\u00e5ll")
            rc = RelayConsumer("127.0.0.1", srv.port, ["t"],
                               starting="earliest")
            frames = []
            for _ in range(50):
                frames += rc.poll(max_msgs=16, timeout_s=0.5)
                if len(frames) == 2:
                    break
            assert frames[0]["key"] == "péti€nt-1"
            assert frames[0]["value"] == big
            assert frames[1]["value"] == "sm\u00e5ll"
            rc.close()
            rp.close()
        finally:
            srv.close()

    def test_sendstream_over_relay(self, bus, tmp_path):
        """The replay producer streams a WFDB record to a REMOTE node's bus
        via --relay host:port (cross-host ingest, end to end)."""
        from tskd_amd.bus.relay import RelayServer
        from tskd_amd.cli import sendstream
        from tests.test_pipeline import _write_wfdb_record
        from tskd_amd.config import GlobalConfig
        cfg = GlobalConfig()
        cfg.wavef_path = str(tmp_path / "wavef")
        cfg.channel_names = ["HR", "RESP"]
        cfg.patient_records = ["p000194-test"]
        _write_wfdb_record(cfg.wavef_path, "p000194-test", cfg.channel_names,
                           1 / 60, 8)
        srv = RelayServer(bus, port=0)
        try:
            import unittest.mock as mock
            with mock.patch("tskd_amd.cli.sendstream.get_global_config",
                            return_value=cfg):
                sendstream.main(["--relay", f"127.0.0.1:{srv.port}",
                                 "--speed", "1e6"])
            c = Consumer(bus, starting="earliest")
            c.subscribe(["HR", "RESP"])
            got = []
            for _ in range(50):
                got += c.poll(max_msgs=64, timeout_ms=100)
                if len(got) == 16:
                    break
            assert len(got) == 16  # 8 samples x 2 channels
            assert {m.topic for m in got} == {"HR", "RESP"}
        finally:
            srv.close()
