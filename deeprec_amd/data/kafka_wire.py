"""Kafka binary wire protocol — client + in-process broker.

Capability parity with the reference's KafkaDataset transport
(kernels/data/kafka_dataset_op.cc over librdkafka): this module speaks
the ACTUAL Kafka protocol (framing, Metadata v0, Fetch v0, Produce v0,
MessageSet v0 with CRC), so `KafkaDataset(servers="host:port")` consumes
from any broker that serves these baseline API versions. MiniKafkaBroker
is an in-process broker speaking the same wire format — the offline test
peer and a single-host log server.

Wire reference: the public Kafka protocol spec (v0 APIs):
  request  = int32 size | int16 api_key | int16 api_version
             | int32 correlation_id | string client_id | body
  string   = int16 len | bytes      bytes = int32 len | data (-1 = null)
  MessageSet = repeat( int64 offset | int32 msg_size |
                       int32 crc32(body) | int8 magic(0) | int8 attrs
                       | bytes key | bytes value )
"""
from __future__ import annotations

import socket
import socketserver
import struct
import threading
import zlib
from typing import Dict, List, Tuple

API_PRODUCE = 0
API_FETCH = 1
API_METADATA = 3


# ---------------------------------------------------------------------
# primitives
# ---------------------------------------------------------------------

def _s16(v):
    return struct.pack(">h", v)


def _s32(v):
    return struct.pack(">i", v)


def _s64(v):
    return struct.pack(">q", v)


def _string(s: str) -> bytes:
    b = s.encode()
    return struct.pack(">h", len(b)) + b


def _bytes(b) -> bytes:
    if b is None:
        return struct.pack(">i", -1)
    return struct.pack(">i", len(b)) + b


class _Reader:
    def __init__(self, data: bytes):
        self.d = data
        self.o = 0

    def i8(self):
        v = struct.unpack_from(">b", self.d, self.o)[0]
        self.o += 1
        return v

    def i16(self):
        v = struct.unpack_from(">h", self.d, self.o)[0]
        self.o += 2
        return v

    def i32(self):
        v = struct.unpack_from(">i", self.d, self.o)[0]
        self.o += 4
        return v

    def i64(self):
        v = struct.unpack_from(">q", self.d, self.o)[0]
        self.o += 8
        return v

    def string(self) -> str:
        n = self.i16()
        if n < 0:
            return ""
        v = self.d[self.o:self.o + n].decode()
        self.o += n
        return v

    def bytes_(self):
        n = self.i32()
        if n < 0:
            return None
        v = self.d[self.o:self.o + n]
        self.o += n
        return v

    def remaining(self) -> int:
        return len(self.d) - self.o


def encode_message(value: bytes, key: bytes = None) -> bytes:
    body = struct.pack(">bb", 0, 0) + _bytes(key) + _bytes(value)
    crc = zlib.crc32(body) & 0xFFFFFFFF
    return struct.pack(">I", crc) + body


def encode_message_set(msgs: List[Tuple[int, bytes]]) -> bytes:
    out = []
    for offset, value in msgs:
        m = encode_message(value)
        out.append(_s64(offset) + _s32(len(m)) + m)
    return b"".join(out)


def decode_message_set(data: bytes) -> List[Tuple[int, bytes]]:
    """-> [(offset, value)] — tolerates a truncated tail (protocol
    allows partial final messages in a fetch response)."""
    out = []
    o = 0
    while o + 12 <= len(data):
        offset, size = struct.unpack_from(">qi", data, o)
        o += 12
        if o + size > len(data):
            break
        m = data[o:o + size]
        o += size
        crc = struct.unpack_from(">I", m, 0)[0]
        body = m[4:]
        if (zlib.crc32(body) & 0xFFFFFFFF) != crc:
            raise IOError("kafka message CRC mismatch")
        r = _Reader(body)
        r.i8()  # magic
        r.i8()  # attrs
        r.bytes_()  # key
        value = r.bytes_()
        out.append((offset, value or b""))
    return out


# ---------------------------------------------------------------------
# client
# ---------------------------------------------------------------------

class KafkaWireClient:
    """Blocking single-connection client for the v0 baseline APIs."""

    def __init__(self, host: str, port: int, client_id="deeprec",
                 timeout=10.0):
        self.sock = socket.create_connection((host, port), timeout=timeout)
        self.sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        self.client_id = client_id
        self._corr = 0
        self._lock = threading.Lock()

    def _call(self, api_key: int, body: bytes) -> _Reader:
        with self._lock:
            self._corr += 1
            corr = self._corr
            req = (struct.pack(">hhi", api_key, 0, corr)
                   + _string(self.client_id) + body)
            self.sock.sendall(_s32(len(req)) + req)
            hdr = self._read_exact(4)
            (size,) = struct.unpack(">i", hdr)
            payload = self._read_exact(size)
        r = _Reader(payload)
        got_corr = r.i32()
        if got_corr != corr:
            raise IOError(f"kafka correlation mismatch {got_corr}!={corr}")
        return r

    def _read_exact(self, n: int) -> bytes:
        out = b""
        while len(out) < n:
            d = self.sock.recv(n - len(out))
            if not d:
                raise ConnectionError("kafka peer closed")
            out += d
        return out

    def metadata(self, topics: List[str]) -> dict:
        body = _s32(len(topics)) + b"".join(_string(t) for t in topics)
        r = self._call(API_METADATA, body)
        brokers = []
        for _ in range(r.i32()):
            node = r.i32()
            host = r.string()
            port = r.i32()
            brokers.append((node, host, port))
        topics_meta = {}
        for _ in range(r.i32()):
            r.i16()  # topic error
            name = r.string()
            parts = []
            for _ in range(r.i32()):
                r.i16()  # partition error
                pid = r.i32()
                r.i32()  # leader
                for _ in range(r.i32()):
                    r.i32()  # replicas
                for _ in range(r.i32()):
                    r.i32()  # isr
                parts.append(pid)
            topics_meta[name] = parts
        return {"brokers": brokers, "topics": topics_meta}

    def fetch(self, topic: str, partition: int, offset: int,
              max_bytes: int = 1 << 20) -> Tuple[List[Tuple[int, bytes]],
                                                 int]:
        """-> ([(offset, value)], high_watermark)."""
        body = (_s32(-1) + _s32(100) + _s32(0)  # replica, max_wait, min
                + _s32(1) + _string(topic)
                + _s32(1) + _s32(partition) + _s64(offset)
                + _s32(max_bytes))
        r = self._call(API_FETCH, body)
        msgs: List[Tuple[int, bytes]] = []
        hw = -1
        for _ in range(r.i32()):
            r.string()  # topic name
            for _ in range(r.i32()):
                r.i32()  # partition
                err = r.i16()
                hw = r.i64()
                ms_size = r.i32()
                data = r.d[r.o:r.o + ms_size]
                r.o += ms_size
                if err == 0:
                    msgs.extend(decode_message_set(data))
        return msgs, hw

    def produce(self, topic: str, partition: int,
                values: List[bytes]) -> int:
        ms = encode_message_set([(0, v) for v in values])
        body = (_s16(1) + _s32(5000)  # acks=1, timeout
                + _s32(1) + _string(topic)
                + _s32(1) + _s32(partition) + _s32(len(ms)) + ms)
        r = self._call(API_PRODUCE, body)
        base = -1
        for _ in range(r.i32()):
            r.string()
            for _ in range(r.i32()):
                r.i32()
                err = r.i16()
                base = r.i64()
                if err:
                    raise IOError(f"kafka produce error {err}")
        return base

    def close(self):
        try:
            self.sock.close()
        except OSError:
            pass


# ---------------------------------------------------------------------
# in-process broker
# ---------------------------------------------------------------------

class _BrokerHandler(socketserver.BaseRequestHandler):
    def handle(self):
        while True:
            try:
                hdr = self._read_exact(4)
            except (ConnectionError, OSError):
                return
            (size,) = struct.unpack(">i", hdr)
            payload = self._read_exact(size)
            r = _Reader(payload)
            api = r.i16()
            r.i16()  # version (v0 assumed)
            corr = r.i32()
            r.string()  # client id
            if api == API_METADATA:
                resp = self._metadata(r)
            elif api == API_FETCH:
                resp = self._fetch(r)
            elif api == API_PRODUCE:
                resp = self._produce(r)
            else:
                return
            out = _s32(corr) + resp
            try:
                self.request.sendall(_s32(len(out)) + out)
            except OSError:
                return

    def _read_exact(self, n):
        out = b""
        while len(out) < n:
            d = self.request.recv(n - len(out))
            if not d:
                raise ConnectionError
            out += d
        return out

    @property
    def logs(self) -> Dict[Tuple[str, int], List[bytes]]:
        return self.server.logs  # type: ignore

    def _metadata(self, r):
        n = r.i32()
        names = [r.string() for _ in range(n)]
        if not names:
            names = sorted({t for t, _ in self.logs})
        host, port = self.server.server_address  # type: ignore
        out = [_s32(1), _s32(0), _string(host), _s32(port)]
        out.append(_s32(len(names)))
        for name in names:
            parts = sorted(p for t, p in self.logs if t == name) or [0]
            out.append(_s16(0) + _string(name) + _s32(len(parts)))
            for p in parts:
                out.append(_s16(0) + _s32(p) + _s32(0)
                           + _s32(1) + _s32(0) + _s32(1) + _s32(0))
        return b"".join(out)

    def _fetch(self, r):
        r.i32()  # replica
        r.i32()  # max wait
        r.i32()  # min bytes
        out = []
        ntop = r.i32()
        out.append(_s32(ntop))
        for _ in range(ntop):
            topic = r.string()
            nparts = r.i32()
            out.append(_string(topic) + _s32(nparts))
            for _ in range(nparts):
                pid = r.i32()
                off = r.i64()
                maxb = r.i32()
                with self.server.lock:  # type: ignore
                    log = list(self.logs.get((topic, pid), []))
                hw = len(log)
                msgs, total = [], 0
                for i in range(off, hw):
                    total += len(log[i]) + 30
                    if msgs and total > maxb:
                        break
                    msgs.append((i, log[i]))
                ms = encode_message_set(msgs)
                out.append(_s32(pid) + _s16(0) + _s64(hw)
                           + _s32(len(ms)) + ms)
        return b"".join(out)

    def _produce(self, r):
        r.i16()  # acks
        r.i32()  # timeout
        out = []
        ntop = r.i32()
        out.append(_s32(ntop))
        for _ in range(ntop):
            topic = r.string()
            nparts = r.i32()
            out.append(_string(topic) + _s32(nparts))
            for _ in range(nparts):
                pid = r.i32()
                ms_size = r.i32()
                ms = r.d[r.o:r.o + ms_size]
                r.o += ms_size
                values = [v for _, v in decode_message_set(ms)]
                with self.server.lock:  # type: ignore
                    log = self.logs.setdefault((topic, pid), [])
                    base = len(log)
                    log.extend(values)
                out.append(_s32(pid) + _s16(0) + _s64(base))
        return b"".join(out)


class MiniKafkaBroker:
    """In-process broker speaking the v0 wire protocol."""

    def __init__(self, host="127.0.0.1", port: int = 0):
        self.server = socketserver.ThreadingTCPServer(
            (host, port), _BrokerHandler, bind_and_activate=True)
        self.server.daemon_threads = True
        self.server.logs = {}
        self.server.lock = threading.Lock()
        self.host = host
        self.port = self.server.server_address[1]
        self._thread = threading.Thread(target=self.server.serve_forever,
                                        daemon=True)
        self._thread.start()

    def seed(self, topic: str, partition: int, values: List[bytes]):
        with self.server.lock:
            self.server.logs.setdefault((topic, partition),
                                        []).extend(values)

    def close(self):
        self.server.shutdown()
        self.server.server_close()


class KafkaWireConsumer:
    """The transport KafkaDataset uses for servers='host:port'."""

    def __init__(self, servers: str):
        host, port = servers.rsplit(":", 1)
        self.client = KafkaWireClient(host, int(port))

    def read_from(self, topic: str, partition: int, offset: int,
                  max_messages: int) -> List[str]:
        msgs, _hw = self.client.fetch(topic, partition, offset)
        return [v.decode() for _, v in msgs[:max_messages]]
