"""Parameter-server data plane: pull/push embedding shards over TCP.

Capability parity with the reference's async-PS training mode (the
functional contract of star_server_lib.cc:60-63 pull/push semantics and
the grpc PS path) — NOT a seastar port: on MI355X the
single-node fast path is RCCL over xGMI (sharded_collection.py); this
plane is the MULTI-NODE capability: PS processes host EV shards, worker
processes pull rows per batch and push gradients asynchronously (the
owner applies them with its own optimizer — lock-free async-PS
semantics: no worker barrier, bounded staleness).

Protocol (length-framed binary, one request per frame):
  PULL  table, keys[int64]           -> rows fp32 [n, dim]
  PUSH  table, keys, grads           -> ack (owner applies async)
  SAVE  dir, step                    -> ack (failover checkpoint)
  STAT                               -> {"tables": {...}, "applied": n}
Routing: key % len(ps_addrs) (the reference's mod partitioner,
python/ops/embedding_ops.py:96-365).
"""
from __future__ import annotations

import pickle
import socket
import socketserver
import struct
import threading
from typing import Dict, List, Optional, Tuple

import torch


def _send_frame(sock, obj):
    payload = pickle.dumps(obj, protocol=4)
    sock.sendall(struct.pack(">i", len(payload)) + payload)


def _recv_exact(sock, n):
    out = b""
    while len(out) < n:
        d = sock.recv(n - len(out))
        if not d:
            raise ConnectionError("ps peer closed")
        out += d
    return out


def _recv_frame(sock):
    (n,) = struct.unpack(">i", _recv_exact(sock, 4))
    return pickle.loads(_recv_exact(sock, n))


class _PsHandler(socketserver.BaseRequestHandler):
    def handle(self):
        srv = self.server
        while True:
            try:
                req = _recv_frame(self.request)
            except (ConnectionError, OSError, EOFError):
                return
            op = req["op"]
            try:
                if op == "PULL":
                    rows = srv.ps.pull(req["table"], req["keys"],
                                       req.get("train", True))
                    _send_frame(self.request, {"ok": True, "rows": rows})
                elif op == "PUSH":
                    srv.ps.push(req["table"], req["keys"], req["grads"])
                    _send_frame(self.request, {"ok": True})
                elif op == "SAVE":
                    path = srv.ps.save(req["dir"], req["step"])
                    _send_frame(self.request, {"ok": True, "path": path})
                elif op == "INCRSAVE":
                    path = srv.ps.incremental_save(req["dir"], req["step"])
                    _send_frame(self.request, {"ok": True, "path": path})
                elif op == "STAT":
                    _send_frame(self.request, {"ok": True,
                                               "stat": srv.ps.stat()})
                else:
                    _send_frame(self.request,
                                {"ok": False, "err": f"bad op {op}"})
            except Exception as e:  # noqa: BLE001
                try:
                    _send_frame(self.request,
                                {"ok": False, "err": repr(e)})
                except OSError:
                    return


class PsServer:
    """Hosts this PS task's EV shards and applies pushed gradients with
    its own optimizer (async apply: workers never wait on each other)."""

    def __init__(self, tables: Dict[str, int], ps_index: int = 0,
                 optimizer: str = "adagrad", lr: float = 0.1,
                 host: str = "127.0.0.1", port: int = 0,
                 checkpoint_dir: Optional[str] = None, device="cpu",
                 initializer=0.5):
        from deeprec_amd.checkpoint.saver import Saver, latest_checkpoint
        from deeprec_amd.embedding import (EmbeddingVariable,
                                           EmbeddingVariableOption)
        from deeprec_amd.embedding.options import InitializerOption
        from deeprec_amd.optimizers import make_optimizer
        self.ps_index = ps_index
        self.evs: Dict[str, EmbeddingVariable] = {}
        for name, dim in tables.items():
            opt = EmbeddingVariableOption(
                init_option=InitializerOption(initializer=initializer))
            self.evs[name] = EmbeddingVariable(
                f"{name}/ps{ps_index}", dim, ev_option=opt, device=device)
        self.opt = make_optimizer(optimizer,
                                  embedding_variables=list(
                                      self.evs.values()),
                                  learning_rate=lr)
        self.saver = Saver(embedding_variables=list(self.evs.values()),
                           rank=ps_index)
        self._applied = 0
        self._lock = threading.Lock()
        if checkpoint_dir:
            ck = latest_checkpoint(checkpoint_dir)
            if ck:
                self.saver.restore(ck)
        class _Srv(socketserver.ThreadingTCPServer):
            allow_reuse_address = True  # failover rebinds the same port
            daemon_threads = True

        self.server = _Srv((host, port), _PsHandler,
                           bind_and_activate=True)
        self.server.ps = self
        self.port = self.server.server_address[1]
        self._thread = threading.Thread(target=self.server.serve_forever,
                                        daemon=True)
        self._thread.start()

    # ---- request handlers (any handler thread) ----
    def pull(self, table: str, keys: torch.Tensor,
             train: bool) -> torch.Tensor:
        ev = self.evs[table]
        with self._lock:
            if train:
                uniq, inverse = torch.unique(keys, return_inverse=True)
                slots = ev.lookup_or_create(uniq)
                rows = ev.storage.gather(uniq, slots)
                return rows[inverse].cpu()
            return ev.gather(keys).cpu()

    def push(self, table: str, keys: torch.Tensor, grads: torch.Tensor):
        ev = self.evs[table]
        with self._lock:
            uniq, inverse = torch.unique(keys, return_inverse=True)
            g = torch.zeros(uniq.numel(), grads.shape[1])
            g.index_add_(0, inverse, grads.float())
            slots = ev.storage.lookup(uniq)
            ev.accumulate_grad(slots, uniq, g)
            # the PS owns its shard's optimizer state but NOT the
            # training step counter (matters when a PS shares a process
            # with a worker, e.g. tests)
            self.opt.step(increment_global_step=False)
            self._applied += 1

    def save(self, directory: str, step: int) -> str:
        with self._lock:
            return self.saver.save(directory, step)

    def incremental_save(self, directory: str, step: int) -> str:
        """Delta checkpoint of keys touched since the last save
        (reference failover contract: restore = last full + ordered
        incremental replay, Incremental-Checkpoint.md)."""
        with self._lock:
            return self.saver.incremental_save(directory, step)

    def stat(self):
        return {"tables": {n: ev.size() for n, ev in self.evs.items()},
                "applied": self._applied, "ps_index": self.ps_index}

    def close(self):
        self.server.shutdown()
        self.server.server_close()


class PsClient:
    """Worker-side connections to every PS task (one socket per PS,
    per client)."""

    def __init__(self, addrs: List[Tuple[str, int]]):
        self.addrs = addrs
        self._socks = [None] * len(addrs)
        self._locks = [threading.Lock() for _ in addrs]

    def _sock(self, i):
        if self._socks[i] is None:
            self._socks[i] = socket.create_connection(self.addrs[i],
                                                      timeout=30)
        return self._socks[i]

    def call(self, ps: int, req: dict, retries: int = 1):
        for attempt in range(retries + 1):
            try:
                with self._locks[ps]:
                    s = self._sock(ps)
                    _send_frame(s, req)
                    resp = _recv_frame(s)
                if not resp.get("ok"):
                    raise RuntimeError(resp.get("err"))
                return resp
            except (ConnectionError, OSError):
                # failover path: drop the socket and retry (a restarted
                # PS restores from its checkpoint and resumes serving)
                self._socks[ps] = None
                if attempt == retries:
                    raise
        return None

    @property
    def world(self):
        return len(self.addrs)

    def close(self):
        for s in self._socks:
            if s is not None:
                try:
                    s.close()
                except OSError:
                    pass


class _PsLookup(torch.autograd.Function):
    @staticmethod
    def forward(ctx, anchor, psemb, keys, rows):
        ctx.psemb = psemb
        ctx.save_for_backward(keys)
        return rows

    @staticmethod
    def backward(ctx, grad_rows):
        (keys,) = ctx.saved_tensors
        ctx.psemb._push(keys, grad_rows)
        return torch.zeros(()), None, None, None


class PsShardedEmbedding:
    """Worker-side EV facade: pull rows per batch, push grads on
    backward (async — the push returns before the apply lands)."""

    def __init__(self, client: PsClient, table: str, dim: int,
                 async_push: bool = True):
        self.client = client
        self.table = table
        self.dim = dim
        self.async_push = async_push
        self._anchor = torch.zeros((), requires_grad=True)
        self._push_threads: List[threading.Thread] = []

    def _route(self, keys: torch.Tensor):
        owner = keys % self.client.world
        return [torch.nonzero(owner == p).squeeze(1)
                for p in range(self.client.world)]

    def lookup(self, keys: torch.Tensor, train: bool = True):
        flat = keys.reshape(-1).cpu()
        rows = torch.empty(flat.numel(), self.dim)
        for p, idx in enumerate(self._route(flat)):
            if idx.numel() == 0:
                continue
            resp = self.client.call(
                p, {"op": "PULL", "table": self.table,
                    "keys": flat[idx], "train": train}, retries=2)
            rows[idx] = resp["rows"]
        rows = rows.reshape(*keys.shape, self.dim)
        if not train:
            return rows
        rows.requires_grad_(False)
        return _PsLookup.apply(self._anchor, self, flat, rows)

    def _push(self, keys: torch.Tensor, grad_rows: torch.Tensor):
        g = grad_rows.reshape(-1, self.dim).detach()

        def do_push():
            for p, idx in enumerate(self._route(keys)):
                if idx.numel() == 0:
                    continue
                self.client.call(p, {"op": "PUSH", "table": self.table,
                                     "keys": keys[idx],
                                     "grads": g[idx]}, retries=2)

        if self.async_push:
            t = threading.Thread(target=do_push, daemon=True)
            t.start()
            self._push_threads.append(t)
        else:
            do_push()

    def flush(self):
        """Join outstanding async pushes (checkpoint/eval barriers)."""
        for t in self._push_threads:
            t.join()
        self._push_threads.clear()
