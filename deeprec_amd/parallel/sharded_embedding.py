"""Model-parallel (embedding-parallel) EmbeddingVariable.

Capability parity with the reference's SOK sharded embedding (SURVEY.md
§2.3 / §3.3): the table is sharded across ranks by key % world_size;
lookup = two-phase all-to-all (per-peer key counts, then key payloads),
local lookup on the owner, all-to-all of embedding rows back; backward
reverses the exchange and the owner applies the sparse update.

MI355X design note: xGMI is point-to-point (7 links/GPU), so the all-to-all
drives all links concurrently — RCCL all_to_all_single is the native fit,
unlike ring collectives which serialize on one link.
"""
from __future__ import annotations

from typing import Optional

import torch

from deeprec_amd.embedding.options import EmbeddingVariableOption
from deeprec_amd.embedding.ragged import RaggedIds
from deeprec_amd.embedding.variable import EmbeddingVariable
from deeprec_amd.ops import functional as F
from deeprec_amd.parallel import comm


class ShardedEmbeddingVariable:
    """Holds this rank's shard; routes lookups by key % world_size."""

    def __init__(self, name: str, embedding_dim: int,
                 ev_option: Optional[EmbeddingVariableOption] = None,
                 device=None, value_dtype=torch.float32):
        self.name = name
        self.dim = embedding_dim
        self.world = comm.world_size()
        self.rank = comm.rank()
        self.local_ev = EmbeddingVariable(
            f"{name}/part_{self.rank}", embedding_dim, value_dtype,
            ev_option, device)

    # delegation so optimizers/savers treat this like an EV
    @property
    def local(self):
        """The rank-local EV (Saver and hooks expect `.local` on every
        sharded wrapper, like ShardedEmbeddingCollection)."""
        return self.local_ev

    @property
    def device(self):
        return self.local_ev.device

    @property
    def storage(self):
        return self.local_ev.storage

    @property
    def trainable(self):
        return self.local_ev.trainable

    @property
    def _anchor(self):
        return self.local_ev._anchor

    @property
    def _pending_grads(self):
        return self.local_ev._pending_grads

    def consume_grads(self):
        return self.local_ev.consume_grads()

    def export(self, include_filtered=False):
        return self.local_ev.export(include_filtered)

    def restore(self, keys, values, freqs=None, versions=None):
        mine = (keys % self.world) == self.rank
        self.local_ev.restore(keys[mine], values[mine],
                              None if freqs is None else freqs[mine],
                              None if versions is None else versions[mine])

    def shrink(self, step=None):
        return self.local_ev.shrink(step)

    def size(self):
        return self.local_ev.size()

    def get_slab(self, name, width=None, init_value=0.0, dtype=torch.float32):
        return self.local_ev.get_slab(name, width, init_value, dtype)

    def reshard(self, world: int, rank: int):
        """Live-resize support: swap in a FRESH local shard under the
        new (world, rank) routing; the caller re-imports owned rows
        (parallel/elastic.live_resize)."""
        old = self.local_ev
        self.world = world
        self.rank = rank
        self.local_ev = EmbeddingVariable(
            f"{self.name}/part_{rank}", self.dim, old.value_dtype,
            old.ev_option, old.device, trainable=old.trainable)


class _ShardedPooledLookup(torch.autograd.Function):
    @staticmethod
    def forward(ctx, anchor, sev: ShardedEmbeddingVariable, uniq, counts,
                inverse, offsets, row_ids, combiner, weights, out_dtype):
        w = sev.world
        dev = uniq.device
        # route by owner; stable sort keeps determinism
        owner = (uniq % w).to(torch.int64)
        order = torch.argsort(owner, stable=True)
        send_keys = uniq[order]
        send_counts_payload = counts[order]
        send_splits = torch.bincount(owner, minlength=w)
        recv_splits = comm.exchange_counts(send_splits)
        in_sp = send_splits.tolist()
        out_sp = recv_splits.tolist()
        # payload exchange: keys + occurrence counts (for admission filters)
        recv_keys = comm.all_to_all_single(send_keys, in_sp, out_sp)
        recv_counts = comm.all_to_all_single(send_counts_payload, in_sp, out_sp)
        # owner-side lookup (keys from different peers may repeat)
        uniq2, inv2 = torch.unique(recv_keys, return_inverse=True)
        counts2 = torch.zeros(uniq2.numel(), dtype=recv_counts.dtype,
                              device=dev)
        counts2.index_add_(0, inv2, recv_counts)
        slots2 = sev.local_ev.lookup_or_create(uniq2, counts2)
        emb2 = sev.local_ev.storage.gather(uniq2, slots2)   # [m2, D] fp32
        emb_out = emb2[inv2]                                 # [n_recv, D]
        # embeddings ride back to the requesting ranks
        emb_back = comm.all_to_all_single(emb_out.contiguous(), out_sp, in_sp)
        # un-permute to uniq order
        emb_uniq = torch.empty_like(emb_back)
        emb_uniq[order] = emb_back
        out = F.pooled_forward(emb_uniq, inverse, offsets, row_ids, combiner,
                               weights, out_dtype)
        ctx.sev = sev
        ctx.combiner = combiner
        ctx.weights = weights
        ctx.splits = (in_sp, out_sp)
        ctx.save_for_backward(order, inv2, slots2, uniq2, inverse, offsets,
                              row_ids)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        sev = ctx.sev
        in_sp, out_sp = ctx.splits
        order, inv2, slots2, uniq2, inverse, offsets, row_ids = \
            ctx.saved_tensors
        m = order.numel()
        grad_unique = F.pooled_backward(grad_out, inverse, offsets, row_ids,
                                        m, ctx.combiner, ctx.weights)
        # to send order, then to owners
        grad_send = grad_unique[order]
        grad_recv = comm.all_to_all_single(grad_send.contiguous(),
                                           in_sp, out_sp)
        # owner: reduce duplicate keys across peers
        grad2 = torch.zeros(uniq2.numel(), grad_recv.shape[1],
                            device=grad_recv.device, dtype=grad_recv.dtype)
        grad2.index_add_(0, inv2, grad_recv)
        sev.local_ev.accumulate_grad(slots2, uniq2, grad2)
        return (torch.zeros_like(sev.local_ev._anchor),) + (None,) * 9


def _sharded_lookup_infer(sev: ShardedEmbeddingVariable, uniq, inverse,
                          offsets, row_ids, combiner, weights, out_dtype):
    """Inference path: no insert, no freq/version bump, no autograd —
    mirrors ShardedEmbeddingCollection._lookup_cat's train=False branch."""
    w = sev.world
    owner = (uniq % w).to(torch.int64)
    order = torch.argsort(owner, stable=True)
    send_keys = uniq[order]
    send_splits = torch.bincount(owner, minlength=w)
    recv_splits = comm.exchange_counts(send_splits)
    in_sp, out_sp = send_splits.tolist(), recv_splits.tolist()
    recv_keys = comm.all_to_all_single(send_keys, in_sp, out_sp)
    uniq2, inv2 = torch.unique(recv_keys, return_inverse=True)
    slots2 = sev.local_ev.storage.lookup(uniq2)   # read-only probe
    emb2 = sev.local_ev.storage.gather(uniq2, slots2)
    emb_out = emb2[inv2]
    emb_back = comm.all_to_all_single(emb_out.contiguous(), out_sp, in_sp)
    emb_uniq = torch.empty_like(emb_back)
    emb_uniq[order] = emb_back
    return F.pooled_forward(emb_uniq, inverse, offsets, row_ids, combiner,
                            weights, out_dtype)


def sharded_embedding_lookup_sparse(sev: ShardedEmbeddingVariable,
                                    sp_ids: RaggedIds, combiner="mean",
                                    out_dtype=None, train=True):
    uniq, inverse, counts = torch.unique(
        sp_ids.values, return_inverse=True, return_counts=True)
    row_ids = sp_ids.row_ids()
    if not (train and sev.trainable):
        return _sharded_lookup_infer(
            sev, uniq, inverse.to(torch.int32), sp_ids.offsets, row_ids,
            combiner, sp_ids.weights, out_dtype)
    return _ShardedPooledLookup.apply(
        sev._anchor, sev, uniq, counts, inverse.to(torch.int32),
        sp_ids.offsets, row_ids, combiner, sp_ids.weights, out_dtype)
