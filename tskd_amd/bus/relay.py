"""TCP edge relay for the bus — cross-host ingest/egress.

The bus itself is a broker-less mmap log (single host, multi-process —
the reference's docker-compose topology). This relay is the node-edge
transport ("keep a bus only at the node edge"): a remote
producer ships samples to the node over TCP, and a remote consumer tails
topics from it; inside the node everything stays on the mmap fast path.

Framing: 4-byte big-endian length + msgpack map per message.
Ops:   {"op": "produce", "topic", "key", "value", "ts_us"}
       {"op": "subscribe", "topics": [...], "starting": "earliest"|"latest"}
Server -> subscriber frames: {"topic", "key", "value", "ts_us", "offset",
"partition"}. At-least-once: the local log is the durability point; a
reconnecting subscriber replays from its own offsets like any consumer.
"""

from __future__ import annotations

import logging
import socket
import struct
import threading
from typing import Iterator, List, Optional

import msgpack

from tskd_amd.bus import Bus, Consumer, Producer

log = logging.getLogger("bus.relay")
_LEN = struct.Struct(">I")


def _send_frame(sock: socket.socket, obj: dict) -> None:
    data = msgpack.packb(obj, use_bin_type=True)
    sock.sendall(_LEN.pack(len(data)) + data)


def _recv_frame(sock: socket.socket) -> Optional[dict]:
    hdr = b""
    while len(hdr) < 4:
        chunk = sock.recv(4 - len(hdr))
        if not chunk:
            return None
        hdr += chunk
    (n,) = _LEN.unpack(hdr)
    body = b""
    while len(body) < n:
        chunk = sock.recv(min(65536, n - len(body)))
        if not chunk:
            return None
        body += chunk
    return msgpack.unpackb(body, raw=False)


class RelayServer:
    """Serves a local bus over TCP: remote produces append to the local
    log; remote subscribes tail it. One thread per connection."""

    def __init__(self, bus: Bus, host: str = "127.0.0.1", port: int = 0):
        self.bus = bus
        self._srv = socket.create_server((host, port))
        self.port = self._srv.getsockname()[1]
        self._stop = threading.Event()
        self._threads: List[threading.Thread] = []
        self._accept_thread = threading.Thread(target=self._accept,
                                               daemon=True)
        self._accept_thread.start()

    def _accept(self) -> None:
        self._srv.settimeout(0.2)
        while not self._stop.is_set():
            try:
                conn, addr = self._srv.accept()
            except socket.timeout:
                continue
            except OSError:
                break
            t = threading.Thread(target=self._serve_conn, args=(conn,),
                                 daemon=True)
            t.start()
            self._threads.append(t)

    def _serve_conn(self, conn: socket.socket) -> None:
        prod = None
        try:
            while not self._stop.is_set():
                conn.settimeout(0.2)
                try:
                    frame = _recv_frame(conn)
                except socket.timeout:
                    continue
                if frame is None:
                    return
                op = frame.get("op")
                if op == "produce":
                    if prod is None:
                        prod = Producer(self.bus)
                        self._known_topics = set()
                    topic = frame["topic"]
                    if topic not in self._known_topics:
                        self.bus.create_topic(topic)
                        self._known_topics.add(topic)
                    prod.produce(topic, frame["key"], frame["value"],
                                 ts_us=int(frame.get("ts_us", -1)))
                    prod.flush()
                elif op == "subscribe":
                    self._stream_to(conn, frame["topics"],
                                    frame.get("starting", "latest"))
                    return
        except (ConnectionError, OSError):
            pass
        finally:
            conn.close()

    def _stream_to(self, conn: socket.socket, topics, starting: str) -> None:
        for t in topics:
            self.bus.create_topic(t)
        cons = Consumer(self.bus, starting=starting)
        cons.subscribe(list(topics))
        conn.settimeout(None)
        while not self._stop.is_set():
            msgs = cons.poll(max_msgs=1024, timeout_ms=200)
            for m in msgs:
                _send_frame(conn, {
                    "topic": m.topic,
                    "key": m.key.decode(),
                    "value": m.value.decode(),
                    "ts_us": m.ts_us,
                    "offset": m.offset,
                    "partition": m.partition,
                })

    def close(self) -> None:
        self._stop.set()
        try:
            self._srv.close()
        except OSError:
            pass


class RelayProducer:
    """Remote producer: ships keyed messages to a RelayServer's bus."""

    def __init__(self, host: str, port: int):
        self._sock = socket.create_connection((host, port), timeout=10)

    def produce(self, topic: str, key: str, value: str,
                ts_us: int = -1) -> None:
        _send_frame(self._sock, {"op": "produce", "topic": topic,
                                 "key": key, "value": value, "ts_us": ts_us})

    def flush(self) -> None:
        """sendall() is already synchronous; durability is the server's
        local log (drop-in for Producer.flush in the replay CLIs)."""

    def close(self) -> None:
        self._sock.close()


class RelayConsumer:
    """Remote subscriber: tails topics from a RelayServer's bus."""

    def __init__(self, host: str, port: int, topics,
                 starting: str = "latest"):
        self._sock = socket.create_connection((host, port), timeout=10)
        _send_frame(self._sock, {"op": "subscribe", "topics": list(topics),
                                 "starting": starting})

    def poll(self, max_msgs: int = 1024,
             timeout_s: float = 1.0) -> List[dict]:
        out = []
        self._sock.settimeout(timeout_s)
        try:
            while len(out) < max_msgs:
                frame = _recv_frame(self._sock)
                if frame is None:
                    break
                out.append(frame)
                self._sock.settimeout(0.01)  # drain whatever is buffered
        except socket.timeout:
            pass
        return out

    def __iter__(self) -> Iterator[dict]:
        while True:
            for m in self.poll():
                yield m

    def close(self) -> None:
        self._sock.close()
