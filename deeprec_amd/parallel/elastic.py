"""Elastic training coordination.

Capability parity with the reference's elastic gRPC server
(contrib/elastic_grpc_server/elastic_grpc_server_lib.cc:294 — an
UpdateServerDef RPC lets the cluster grow/shrink at runtime). The
MI355X-native analog is restart-based, built on two pieces this framework
already has:

- repartition-safe EV checkpoints (keys bucketed mod 1000, restore
  filters to owned keys — any world size restores from any other,
  tests/test_checkpoint.py);
- a generation protocol over torch.distributed.TCPStore: a coordinator
  announces (generation, world_size); workers poll between steps, and on
  a generation bump they checkpoint, tear down the process group, and
  re-exec under the new world size (the launcher owns process lifecycle,
  as torchrun does).

ElasticController is the store-side state machine; ElasticAgent is the
per-worker poll handle.
"""
from __future__ import annotations

import datetime
from typing import Optional

import torch.distributed as dist

_GEN_KEY = "deeprec/elastic/gen"
_WORLD_KEY = "deeprec/elastic/world"


class ElasticController:
    """Runs next to (or inside) rank 0; owns the TCPStore server."""

    def __init__(self, host: str = "127.0.0.1", port: int = 29699,
                 store: Optional[object] = None):
        self.store = store or dist.TCPStore(
            host, port, is_master=True,
            timeout=datetime.timedelta(seconds=30))
        self.store.set(_GEN_KEY, "0")
        self.store.set(_WORLD_KEY, "0")

    def propose_resize(self, new_world_size: int) -> int:
        """Announce a new cluster size; returns the new generation."""
        gen = int(self.store.get(_GEN_KEY)) + 1
        self.store.set(_WORLD_KEY, str(new_world_size))
        self.store.set(_GEN_KEY, str(gen))
        return gen

    def current(self):
        return (int(self.store.get(_GEN_KEY)),
                int(self.store.get(_WORLD_KEY)))


class ElasticAgent:
    """Worker-side handle: poll for scale events between steps."""

    def __init__(self, host: str = "127.0.0.1", port: int = 29699,
                 store: Optional[object] = None):
        self.store = store or dist.TCPStore(
            host, port, is_master=False,
            timeout=datetime.timedelta(seconds=30))
        self.generation = int(self.store.get(_GEN_KEY))

    def check_resize(self) -> Optional[int]:
        """New world size if a scale event happened since last check,
        else None. Consumes the event."""
        gen = int(self.store.get(_GEN_KEY))
        if gen == self.generation:
            return None
        self.generation = gen
        return int(self.store.get(_WORLD_KEY))


def elastic_step_hook(agent: ElasticAgent, saver, ckpt_dir: str,
                      global_step: int) -> bool:
    """Call between steps: on a scale event, write a full checkpoint and
    return True (caller should exit so the launcher can restart it under
    the new world size; restore is repartition-safe)."""
    new_world = agent.check_resize()
    if new_world is None:
        return False
    saver.save(ckpt_dir, global_step=global_step)
    return True
