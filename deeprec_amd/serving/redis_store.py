"""Redis-protocol (RESP2) feature store.

Capability parity with the reference's Redis-backed sparse-weight store
(serving/processor/storage/redis_feature_store.cc: embedding rows live
in an external Redis so many serving replicas share one weight store,
updated online). Implemented as:

- RedisFeatureStore — a FeatureStore speaking RESP2 over a socket
  (MGET/MSET/pipelined SET) to ANY Redis-compatible server; rows are
  raw fp32 bytes keyed "emb:{table}:{id}".
- MiniRedisServer — an in-process RESP2 subset server (GET/SET/MGET/
  MSET/DEL/PING/FLUSHDB) so the wire protocol is testable in this
  offline environment; also usable as a tiny single-host weight server.
"""
from __future__ import annotations

import socket
import socketserver
import struct
import threading
from typing import Dict, Optional

import torch

from deeprec_amd.serving.feature_store import FeatureStore


# ---------------------------------------------------------------------
# RESP2 encoding
# ---------------------------------------------------------------------

def _enc_array(parts) -> bytes:
    out = [b"*%d\r\n" % len(parts)]
    for p in parts:
        if isinstance(p, str):
            p = p.encode()
        out.append(b"$%d\r\n" % len(p))
        out.append(p)
        out.append(b"\r\n")
    return b"".join(out)


class _RespReader:
    def __init__(self, sock: socket.socket):
        self.sock = sock
        self.buf = b""

    def _read_more(self):
        d = self.sock.recv(1 << 16)
        if not d:
            raise ConnectionError("peer closed")
        self.buf += d

    def read_line(self) -> bytes:
        while b"\r\n" not in self.buf:
            self._read_more()
        line, self.buf = self.buf.split(b"\r\n", 1)
        return line

    def read_exact(self, n: int) -> bytes:
        while len(self.buf) < n + 2:
            self._read_more()
        out, self.buf = self.buf[:n], self.buf[n + 2:]  # strip \r\n
        return out

    def read_reply(self):
        line = self.read_line()
        t, rest = line[:1], line[1:]
        if t == b"+":
            return rest.decode()
        if t == b"-":
            raise RuntimeError(f"redis error: {rest.decode()}")
        if t == b":":
            return int(rest)
        if t == b"$":
            n = int(rest)
            if n == -1:
                return None
            return self.read_exact(n)
        if t == b"*":
            n = int(rest)
            return [self.read_reply() for _ in range(n)]
        raise RuntimeError(f"bad RESP type byte {t!r}")


class RedisClient:
    """Minimal pipelining RESP2 client."""

    def __init__(self, host: str = "127.0.0.1", port: int = 6379,
                 timeout: float = 10.0):
        self.sock = socket.create_connection((host, port), timeout=timeout)
        self.sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        self.reader = _RespReader(self.sock)
        self._lock = threading.Lock()

    def execute(self, *parts):
        with self._lock:
            self.sock.sendall(_enc_array(parts))
            return self.reader.read_reply()

    def pipeline(self, commands):
        with self._lock:
            self.sock.sendall(b"".join(_enc_array(c) for c in commands))
            return [self.reader.read_reply() for _ in commands]

    def close(self):
        try:
            self.sock.close()
        except OSError:
            pass


class RedisFeatureStore(FeatureStore):
    """Embedding rows in Redis: key 'emb:{table}:{id}' -> raw fp32 row."""

    def __init__(self, host="127.0.0.1", port=6379, batch: int = 1024):
        self.client = RedisClient(host, port)
        self.batch = batch
        assert self.client.execute("PING") == "PONG"
        self._tables: Dict[str, int] = {}

    @staticmethod
    def _key(table: str, k: int) -> str:
        return f"emb:{table}:{k}"

    def put(self, table: str, keys: torch.Tensor, values: torch.Tensor):
        ks = keys.cpu().tolist()
        vs = values.detach().cpu().float()
        self._tables[table] = vs.shape[1]
        for i in range(0, len(ks), self.batch):
            cmd = ["MSET"]
            for k, row in zip(ks[i:i + self.batch],
                              vs[i:i + self.batch]):
                cmd.append(self._key(table, k))
                cmd.append(struct.pack(f"<{row.numel()}f",
                                       *row.tolist()))
            self.client.execute(*cmd)

    def get(self, table: str, keys: torch.Tensor, dim: int,
            default: float = 0.0) -> torch.Tensor:
        ks = keys.cpu().tolist()
        out = torch.full((len(ks), dim), float(default))
        for i in range(0, len(ks), self.batch):
            chunk = ks[i:i + self.batch]
            reply = self.client.execute(
                "MGET", *[self._key(table, k) for k in chunk])
            for j, raw in enumerate(reply):
                if raw is not None:
                    out[i + j] = torch.tensor(
                        struct.unpack(f"<{dim}f", raw))
        return out

    def tables(self):
        return list(self._tables)


# ---------------------------------------------------------------------
# in-process RESP2 server (tests / single-host weight serving)
# ---------------------------------------------------------------------

class _MiniRedisHandler(socketserver.BaseRequestHandler):
    def handle(self):
        reader = _RespReader(self.request)
        store: Dict[bytes, bytes] = self.server.kv  # type: ignore
        lock = self.server.kv_lock  # type: ignore
        while True:
            try:
                parts = reader.read_reply()
            except (ConnectionError, OSError):
                return
            if not isinstance(parts, list) or not parts:
                return
            cmd = parts[0].upper()
            try:
                if cmd == b"PING":
                    self.request.sendall(b"+PONG\r\n")
                elif cmd == b"SET":
                    with lock:
                        store[parts[1]] = parts[2]
                    self.request.sendall(b"+OK\r\n")
                elif cmd == b"MSET":
                    with lock:
                        for i in range(1, len(parts), 2):
                            store[parts[i]] = parts[i + 1]
                    self.request.sendall(b"+OK\r\n")
                elif cmd == b"GET":
                    with lock:
                        v = store.get(parts[1])
                    self._bulk(v)
                elif cmd == b"MGET":
                    with lock:
                        vs = [store.get(k) for k in parts[1:]]
                    self.request.sendall(b"*%d\r\n" % len(vs))
                    for v in vs:
                        self._bulk(v)
                elif cmd == b"DEL":
                    n = 0
                    with lock:
                        for k in parts[1:]:
                            n += 1 if store.pop(k, None) is not None else 0
                    self.request.sendall(b":%d\r\n" % n)
                elif cmd == b"FLUSHDB":
                    with lock:
                        store.clear()
                    self.request.sendall(b"+OK\r\n")
                elif cmd == b"DBSIZE":
                    with lock:
                        self.request.sendall(b":%d\r\n" % len(store))
                else:
                    self.request.sendall(b"-ERR unknown command\r\n")
            except (BrokenPipeError, OSError):
                return

    def _bulk(self, v: Optional[bytes]):
        if v is None:
            self.request.sendall(b"$-1\r\n")
        else:
            self.request.sendall(b"$%d\r\n%s\r\n" % (len(v), v))


class MiniRedisServer:
    def __init__(self, host="127.0.0.1", port: int = 0):
        self.server = socketserver.ThreadingTCPServer(
            (host, port), _MiniRedisHandler, bind_and_activate=True)
        self.server.daemon_threads = True
        self.server.kv = {}
        self.server.kv_lock = threading.Lock()
        self.port = self.server.server_address[1]
        self._thread = threading.Thread(
            target=self.server.serve_forever, daemon=True)
        self._thread.start()

    def close(self):
        self.server.shutdown()
        self.server.server_close()
