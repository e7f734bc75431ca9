"""Keyed topic bus — python API over the C++ broker-less bus (_tskd_bus).

Drop-in functional analog of the reference's confluent-kafka usage
(reference sendStream.py:41-72, utils.py:409-428): keyed topics, acks=all,
retries=5, startingOffsets latest/earliest, replay via seek. Topics are
mmap'd partition logs under a shared directory (tmpfs by default), so the
pipeline stages remain separate OS processes like the reference's
docker-compose services — without a broker.
"""

from __future__ import annotations

import json
import time
from typing import List, Optional, Sequence

from tskd_amd import _tskd_bus as _C
from tskd_amd.config import GlobalConfig, get_global_config

Message = _C.Message


def default_bus_dir(cfg: Optional[GlobalConfig] = None) -> str:
    cfg = cfg or get_global_config()
    d = cfg.bus_dir
    if not d.startswith("/"):
        import os
        base = "/dev/shm" if os.path.isdir("/dev/shm") else "/tmp"
        d = f"{base}/{d}"
    return d


class Bus:
    def __init__(self, directory: Optional[str] = None, default_parts: int = 1):
        self._c = _C.Bus(directory or default_bus_dir(), default_parts)

    @property
    def dir(self) -> str:
        return self._c.dir

    def create_topic(self, topic: str, nparts: int = 1) -> None:
        self._c.create_topic(topic, nparts)

    def list_topics(self) -> List[str]:
        return self._c.list_topics()

    def end_offset(self, topic: str, partition: int = 0) -> int:
        return self._c.end_offset(topic, partition)

    def partition_for(self, topic: str, key: str) -> int:
        return self._c.partition_for(topic, key)

    def trim_offset(self, topic: str, partition: int = 0) -> int:
        return self._c.trim_offset(topic, partition)

    def trim_topic(self, topic: str, partition: int = 0,
                   before_offset: int = 0) -> int:
        """Retention: reclaim storage for messages before ``before_offset``
        (hole punch; offsets stay absolute — reads below the trim point are
        a consumer-visible gap, like Kafka retention). Returns the applied
        page-aligned trim offset."""
        return self._c.trim_topic(topic, partition, before_offset)


class Producer:
    """acks=all, retries=5 producer (reference utils.py:417-422)."""

    def __init__(self, bus: Bus, retries: int = 5):
        self._c = _C.Producer(bus._c, retries)

    def produce(self, topic: str, key: bytes | str, value: bytes | str,
                partition: int = -1, ts_us: int = -1,
                callback=None) -> None:
        k = key.decode() if isinstance(key, bytes) else key
        v = value.decode() if isinstance(value, bytes) else value
        try:
            self._c.produce(topic, k, v, partition, ts_us)
        except Exception as e:
            if callback:
                callback(e, None)  # reference `acked` error path (utils.py:409)
                return
            raise
        if callback:
            callback(None, (topic, k))

    def produce_sample(self, topic: str, patient_id: str, channel_index: int,
                       value: float, ts_us: int = -1) -> None:
        """The reference wire format: key=patientid,
        value=JSON [channel_index, float] (sendStream.py:59-64)."""
        self.produce(topic, patient_id,
                     json.dumps([channel_index, value]), ts_us=ts_us)

    def flush(self, topic: str = "") -> None:
        self._c.flush(topic)

    @property
    def produced(self) -> int:
        return self._c.produced


class Consumer:
    """startingOffsets latest|earliest; poll; seek for replay-from-offset."""

    def __init__(self, bus: Bus, starting: str = "latest"):
        assert starting in ("latest", "earliest")
        self._c = _C.Consumer(bus._c, starting)

    def subscribe(self, topics: Sequence[str]) -> None:
        self._c.subscribe(list(topics))

    def poll(self, max_msgs: int = 256, timeout_ms: int = 0) -> List[Message]:
        return self._c.poll(max_msgs, timeout_ms)

    def poll_samples(self, max_msgs: int = 4096, timeout_ms: int = 0):
        """Poll + parse the reference "[chan, value]" wire format natively
        (C++): returns (keys, topics, chan int32[], value float32[],
        ts float64[] seconds). Unparseable messages are skipped (their
        offsets still advance)."""
        return self._c.poll_samples(max_msgs, timeout_ms)

    def poll_samples_sid(self, max_msgs: int = 4096, timeout_ms: int = 0,
                         rank: int = 0, world: int = 1,
                         max_streams: int = 0):
        """The fully-native serving ingest edge: poll + wire parse + key->
        dense-stream-id mapping + DP shard filter (FNV-1a % world == rank,
        same hash as parallel.shard_for_key) in one C++ pass. Returns
        (sid int32[], chan int32[], value float32[], ts float64[] seconds,
        new_keys [(key, sid), ...] in sid order)."""
        return self._c.poll_samples_sid(max_msgs, timeout_ms, rank, world,
                                        max_streams)

    def seek(self, topic: str, partition: int, offset: int) -> None:
        self._c.seek(topic, partition, offset)

    def positions(self):
        return self._c.positions()


def wait_for_topic(bus: Bus, topic: str, timeout_s: float = 10.0) -> bool:
    t0 = time.time()
    while time.time() - t0 < timeout_s:
        if topic in bus.list_topics():
            return True
        time.sleep(0.05)
    return False
