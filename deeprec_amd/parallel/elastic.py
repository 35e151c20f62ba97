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


def _export_full(base):
    keys, values, freqs, versions = base.export()
    names = list(base.storage.slabs)
    rows = base.storage.export_slabs(names) if names else []
    return {"keys": keys.cpu(), "values": values.cpu().float(),
            "freqs": freqs.cpu(), "versions": versions.cpu(),
            "slabs": {n: r.cpu() for n, r in zip(names, rows)},
            "slab_init": {n: float(base.storage._slab_init.get(n, 0.0))
                          for n in names}}


def _import_owned(ev, shard, world, rank, collection):
    base = getattr(ev, "local", ev)
    keys = shard["keys"]
    if collection:
        from deeprec_amd.embedding.collection import KEY_BITS
        raw = keys & ((1 << KEY_BITS) - 1)
        mask = (raw % world) == rank
    else:
        mask = (keys % world) == rank
    slab_rows = {n: r[mask] for n, r in shard["slabs"].items()}
    for n, r in slab_rows.items():
        base.storage.get_slab(n, r.shape[1], shard["slab_init"][n])
    base.storage.import_(keys[mask].to(base.device),
                         shard["values"][mask].to(base.device),
                         shard["freqs"][mask], shard["versions"][mask],
                         slab_rows=slab_rows or None)


def live_resize(new_world: int, sharded_evs, reinit_fn) -> bool:
    """IN-PROCESS cluster shrink (no restart): every rank exports its
    shard state (values + optimizer slabs) and all-gathers it in the OLD
    process group; the group is torn down; surviving ranks re-initialize
    via `reinit_fn(rank, new_world)` and rebuild their shards under the
    new routing. Departing ranks return False and should exit.

    This is the live counterpart of the reference's UpdateServerDef
    resize (elastic_grpc_server_lib.cc:294); growth still goes through
    the restart path (new processes cannot be conjured in-process —
    exactly torchrun's model)."""
    import torch.distributed as dist

    from deeprec_amd.parallel import comm
    old_world = comm.world_size()
    rank = comm.rank()
    assert 0 < new_world <= old_world, \
        "live_resize handles shrink; growth is restart-based"
    payload = [_export_full(getattr(ev, "local", ev))
               for ev in sharded_evs]
    gathered = [None] * old_world
    dist.all_gather_object(gathered, payload)
    dist.destroy_process_group()
    if rank >= new_world:
        return False  # departed; shard state lives on in the survivors
    reinit_fn(rank, new_world)
    for i, ev in enumerate(sharded_evs):
        collection = hasattr(ev, "export_tables")
        ev.reshard(new_world, rank)
        for rank_payload in gathered:
            _import_owned(ev, rank_payload[i], new_world, rank,
                          collection)
    return True


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
