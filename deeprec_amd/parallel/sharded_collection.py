"""ShardedEmbeddingCollection — model-parallel multi-table embeddings.

The multi-GPU fast path: all N tables share one composite-key space, so a
training step does ONE two-phase all-to-all for keys (counts then payload,
reference SOK protocol SURVEY.md §3.3), one local probe+gather on the
owner, ONE all-to-all back for embedding rows, then the same fused group
pooling as the single-GPU collection. Backward reverses the exchange; the
owner applies the fused sparse update.

xGMI note: one big all-to-all per step (vs 26 per-table exchanges) is the
shape that drives all 7 point-to-point links concurrently.
"""
from __future__ import annotations

from typing import Optional, Sequence

import torch

from deeprec_amd.embedding.collection import KEY_BITS, EmbeddingCollection
from deeprec_amd.embedding.options import EmbeddingVariableOption
from deeprec_amd.embedding.ragged import RaggedIds
from deeprec_amd.embedding.variable import get_global_step
from deeprec_amd.parallel import comm


class ShardedEmbeddingCollection:
    """Per-rank shard of an EmbeddingCollection; keys routed by
    raw_id % world_size (composite tags are multiples of 2^48, so routing
    is table-independent for power-of-two worlds and deterministic for
    all)."""

    def __init__(self, name: str, table_names: Sequence[str],
                 embedding_dim: int,
                 ev_option: Optional[EmbeddingVariableOption] = None,
                 combiners=None, device=None, value_dtype=torch.float32,
                 generator=None, trainable: bool = True,
                 comm_dtype: Optional[torch.dtype] = None,
                 node_size: Optional[int] = None):
        self.world = comm.world_size()
        self.rank = comm.rank()
        # comm_dtype=torch.bfloat16 halves the xGMI bytes of the row-return
        # all-to-all; lossless end-to-end for bf16 models (the pooled rows
        # are cast to bf16 for the MLP anyway). Gradients stay fp32.
        self.comm_dtype = comm_dtype
        # node_size (= GPUs per node, e.g. LOCAL_WORLD_SIZE) switches the
        # exchanges to the two-hop node-aware all-to-all when the job
        # spans nodes (parallel/hierarchical.py)
        self.node_size = node_size
        self.local = EmbeddingCollection(
            f"{name}/part_{self.rank}", table_names, embedding_dim,
            ev_option, combiners, device, value_dtype, generator, trainable)
        self.name = name
        self.dim = embedding_dim
        self.n_tables = self.local.n_tables
        # static-shape (padded) exchange mode — see static_mode()
        self._pad_cap = None
        self._observed_max_split = 0

    # ---- optimizer-facing delegation ----
    @property
    def device(self):
        return self.local.device

    @property
    def storage(self):
        return self.local.storage

    @property
    def trainable(self):
        return self.local.trainable

    @property
    def _anchor(self):
        return self.local._anchor

    @property
    def _pending_grads(self):
        return self.local._pending_grads

    def consume_grads(self):
        return self.local.consume_grads()

    def get_slab(self, *a, **kw):
        return self.local.get_slab(*a, **kw)

    def size(self):
        return self.local.size()

    def shrink(self, step=None):
        return self.local.shrink(step)

    def export_tables(self, include_filtered=False):
        return self.local.export_tables(include_filtered)

    def reshard(self, world: int, rank: int):
        """Live-resize support: fresh local collection under the new
        (world, rank) routing (parallel/elastic.live_resize re-imports)."""
        old = self.local
        self.world = world
        self.rank = rank
        self.local = EmbeddingCollection(
            f"{self.name}/part_{rank}", list(old.table_names), self.dim,
            old.ev_option, list(old.combiners), old.device,
            trainable=old.trainable)

    def restore_table(self, table, keys, values, freqs=None, versions=None):
        mask = (keys % self.world) == self.rank
        self.local.restore_table(
            table, keys[mask], values[mask],
            None if freqs is None else freqs[mask],
            None if versions is None else versions[mask])

    # ---- lookups ----
    def lookup_matrix(self, ids: torch.Tensor, out_dtype=None,
                      train: bool = True) -> torch.Tensor:
        batch, n = ids.shape
        assert n == self.n_tables
        coll = self.local
        cache = coll._matrix_cache.get(batch)
        if cache is None:
            dev = coll.device
            nb = n * batch
            cache = {
                "offsets": torch.arange(nb + 1, dtype=torch.int32,
                                        device=dev),
                "row_ids": torch.arange(nb, dtype=torch.int32, device=dev),
                "row_coeff": torch.ones(nb, device=dev),
                "tags": (torch.arange(n, dtype=torch.int64, device=dev)
                         << KEY_BITS).repeat_interleave(batch),
            }
            coll._matrix_cache[batch] = cache
        values_cat = ids.t().reshape(-1) + cache["tags"]
        return self._lookup_cat(values_cat, cache["offsets"],
                                cache["row_ids"], cache["row_coeff"], None,
                                batch, out_dtype, train)

    def lookup(self, sp_list: Sequence[RaggedIds], out_dtype=None,
               train: bool = True) -> torch.Tensor:
        coll = self.local
        batch, values_cat, offsets_cat, row_ids_cat, weights_cat = \
            coll._concat_inputs(sp_list)
        row_coeff = coll._row_coeffs(offsets_cat, row_ids_cat, weights_cat,
                                     batch)
        return self._lookup_cat(values_cat, offsets_cat, row_ids_cat,
                                row_coeff, weights_cat, batch, out_dtype,
                                train)

    def _lookup_cat(self, values_cat, offsets_cat, row_ids_cat, row_coeff,
                    weights_cat, batch, out_dtype, train):
        coll = self.local
        train = train and coll.trainable
        if train and self._pad_cap is not None:
            return _padded_sharded_lookup(
                self, values_cat, offsets_cat, row_ids_cat, row_coeff,
                weights_cat, batch, out_dtype)
        uniq, inverse, counts = torch.unique(
            values_cat, return_inverse=True, return_counts=True)
        inverse = inverse.to(torch.int32)
        if not train:
            emb = _exchange_lookup(self, uniq, counts, train=False)[0]
            return coll._forward(uniq, None, inverse, offsets_cat,
                                 weights_cat, batch, out_dtype,
                                 emb_override=emb)
        order, bounds, chunk_u, chunk_k0 = coll._prep_backward(inverse,
                                                               counts)
        return _ShardedCollectionLookup.apply(
            coll._anchor, self, uniq, counts, inverse, offsets_cat,
            row_ids_cat, order, bounds, chunk_u, chunk_k0, row_coeff,
            weights_cat, batch, out_dtype)

    # ---- static-shape (padded) exchange — hipGraph-capturable ----
    def static_mode(self, pad_cap: int):
        """Switch the training exchange to the fixed-shape padded
        all-to-all: every peer slot padded to pad_cap rows, pad keys =
        PAD_KEY, no count exchange on the wire. All tensor shapes become
        data-independent, which is what hipGraph capture (and RCCL
        capture) requires. Works eagerly too (gloo/CPU included) — the
        wire protocol is identical; only the local dedup engine differs.
        pad_cap must exceed the per-peer unique-key count of every step
        (observed_max_split() after eager warmup + slack is the intended
        sizing) and must be IDENTICAL on every rank — it defines the
        wire shape, so agree on it with an all-reduce max before calling
        (bench.py does). Overflow raises via the engine error flag on
        GPU and ValueError on CPU."""
        self._pad_cap = int(pad_cap)

    def observed_max_split(self) -> int:
        """Largest per-peer split seen by the eager exchange (cap sizing
        aid for static_mode)."""
        return self._observed_max_split

    def enable_graph_mode(self, expected_entries: int, expected_slots: int,
                          pad_cap: int):
        """Prepare the full distributed step for hipGraph capture:
        pre-size the local table (requester-side dedup inserts slot-less
        entries for EVERY key this rank sees, so size for the whole id
        space, not the owned share), device-resident epoch/step, and the
        padded exchange."""
        self.local.storage.enable_graph_mode(expected_entries,
                                             expected_slots)
        self.local.graph_mode = True
        self.static_mode(pad_cap)


PAD_KEY = (1 << 63) - 1  # == ext.PAD_KEY (engine wire-padding sentinel)


def _route_pad_cpu(uniq, counts, world, cap, key_bits):
    """CPU/gloo reference of the k_route_pad kernel: scatter unique keys
    into per-peer blocks of `cap` rows, pad with PAD_KEY. Returns
    (send_keys [w*cap], send_cnt [w*cap] i32, route_pos [m] i64)."""
    m = uniq.numel()
    raw = uniq & ((1 << key_bits) - 1)
    owner = (raw % world).to(torch.int64)
    send_splits = torch.bincount(owner, minlength=world)
    if m and int(send_splits.max()) > cap:
        raise ValueError(
            f"padded exchange overflow: peer split {int(send_splits.max())}"
            f" exceeds pad cap {cap}")
    order = torch.argsort(owner, stable=True)
    group_start = torch.zeros(world, dtype=torch.int64)
    group_start[1:] = send_splits.cumsum(0)[:-1]
    sorted_owner = owner[order]
    rank_in_group = torch.arange(m, dtype=torch.int64) \
        - group_start[sorted_owner]
    dest = sorted_owner * cap + rank_in_group
    send_keys = torch.full((world * cap,), PAD_KEY, dtype=torch.int64)
    send_cnt = torch.zeros(world * cap, dtype=torch.int32)
    send_keys[dest] = uniq[order]
    send_cnt[dest] = counts[order].to(torch.int32)
    route_pos = torch.empty(m, dtype=torch.int64)
    route_pos[order] = dest
    return send_keys, send_cnt, route_pos


def _padded_sharded_lookup(sev, values_cat, offsets_cat, row_ids_cat,
                           row_coeff, weights_cat, batch, out_dtype):
    """Fixed-shape training lookup: requester dedup -> padded all-to-all
    of (keys, counts) -> owner dedup+admission -> padded all-to-all of
    rows back -> fused pooling. Every tensor shape is independent of the
    data, so the WHOLE sequence (RCCL collectives included) records into
    a hipGraph and replays. Eager execution uses the identical wire
    protocol (gloo/CPU runs it in tests)."""
    coll = sev.local
    st = coll.storage
    w, cap = sev.world, sev._pad_cap
    dev = coll.device
    is_gpu = dev.type == "cuda"
    splits = [cap] * w
    if is_gpu:
        uniq_buf, inverse, counts, m_dev = st.dedup_only_capture(values_cat)
        send_keys, send_cnt, route_pos = st.ext.route_pad(
            uniq_buf, counts, m_dev, w, cap, KEY_BITS, st.error_flag)
        # CSR prep consumes pass C's rank before anything else overwrites
        order, bounds, _, _ = coll._prep_backward(inverse, counts)
        m_req = values_cat.numel()
    else:
        uniq, inv_t, cnt_t = torch.unique(
            values_cat, return_inverse=True, return_counts=True)
        inverse = inv_t.to(torch.int32)
        st._last_rank = None
        order, bounds, _, _ = coll._prep_backward(inverse, cnt_t)
        send_keys, send_cnt, route_pos = _route_pad_cpu(
            uniq, cnt_t, w, cap, KEY_BITS)
        m_dev = None
        m_req = uniq.numel()
    recv_keys = comm.all_to_all_single(send_keys, splits, splits)
    recv_cnt = comm.all_to_all_single(send_cnt, splits, splits)
    if is_gpu:
        uniq2, inv2, slots2 = st.dedup_lookup_capture_owner(recv_keys,
                                                            recv_cnt)
        slots_elem = st.ext.slots_gather_pad(inv2, slots2)
        wire = sev.comm_dtype or torch.float32
        reply = st.ext.ev_gather(
            st.values, st.default_values, recv_keys, slots_elem,
            st._no_permission_value(), st._use_no_permission(), wire)
        own_valid = None
        emb_rows = comm.all_to_all_single(reply, splits, splits).float()
        inverse_final = st.ext.compose_i32(inverse, route_pos)
        keys_for_kernel = recv_keys
    else:
        uniq2, inv2 = torch.unique(recv_keys, return_inverse=True)
        cnt2 = torch.zeros(uniq2.numel(), dtype=torch.int64)
        cnt2.index_add_(0, inv2, recv_cnt.long())
        own_valid = uniq2 != PAD_KEY
        slots_v = st.lookup_or_create(uniq2[own_valid], cnt2[own_valid],
                                      get_global_step(), train=True)
        emb_v = st.gather(uniq2[own_valid], slots_v)
        reply = torch.zeros(uniq2.numel(), coll.dim)
        reply[own_valid] = emb_v
        reply = reply[inv2]
        slots2 = slots_v
        if sev.comm_dtype is not None:
            lp = reply.to(sev.comm_dtype).contiguous()
            emb_rows = comm.all_to_all_single(
                lp.view(torch.int16), splits, splits) \
                .view(sev.comm_dtype).float()
        else:
            emb_rows = comm.all_to_all_single(reply.contiguous(), splits,
                                              splits)
        inverse_final = route_pos[inverse.long()].to(torch.int32)
        keys_for_kernel = uniq2
    if is_gpu:
        inverse_final = inverse_final.to(torch.int32)
    return _PaddedShardedLookup.apply(
        coll._anchor, sev, emb_rows, keys_for_kernel, inverse_final,
        offsets_cat, row_ids_cat, order, bounds, row_coeff, weights_cat,
        batch, out_dtype, m_req, m_dev, route_pos, inv2, slots2, uniq2,
        own_valid)


class _PaddedShardedLookup(torch.autograd.Function):
    @staticmethod
    def forward(ctx, anchor, sev, emb_rows, keys_for_kernel, inverse_final,
                offsets_cat, row_ids_cat, order, bounds, row_coeff,
                weights_cat, batch, out_dtype, m_req, m_dev, route_pos,
                inv2, slots2, uniq2, own_valid):
        out = sev.local._forward(keys_for_kernel, None, inverse_final,
                                 offsets_cat, weights_cat, batch, out_dtype,
                                 emb_override=emb_rows)
        ctx.sev = sev
        ctx.batch = batch
        ctx.weights_cat = weights_cat
        ctx.m_req = m_req
        ctx.m_dev = m_dev
        ctx.own_valid = own_valid
        ctx.save_for_backward(order, bounds, row_ids_cat, row_coeff,
                              route_pos, inv2, slots2, uniq2)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        sev = ctx.sev
        coll = sev.local
        st = coll.storage
        w, cap = sev.world, sev._pad_cap
        splits = [cap] * w
        order, bounds, row_ids_cat, row_coeff, route_pos, inv2, slots2, \
            uniq2 = ctx.saved_tensors
        grad_unique = coll._backward(grad_out, order, bounds, None, None,
                                     row_ids_cat, ctx.weights_cat,
                                     row_coeff, ctx.m_req, ctx.batch)
        if coll.device.type == "cuda":
            grad_send = st.ext.rows_to_padded(grad_unique, route_pos,
                                              ctx.m_dev, w * cap)
            grad_recv = comm.all_to_all_single(grad_send, splits, splits)
            grad2 = st.ext.rows_segsum_pad(grad_recv, inv2, w * cap)
            coll.accumulate_grad(slots2, uniq2, grad2)
        else:
            grad_send = torch.zeros(w * cap, grad_unique.shape[1])
            grad_send[route_pos] = grad_unique
            grad_recv = comm.all_to_all_single(grad_send, splits, splits)
            grad2 = torch.zeros(uniq2.numel(), grad_recv.shape[1])
            grad2.index_add_(0, inv2, grad_recv)
            valid = ctx.own_valid
            coll.accumulate_grad(slots2, uniq2[valid], grad2[valid])
        return (torch.zeros_like(coll._anchor),) + (None,) * 19


def _a2a(sev: ShardedEmbeddingCollection, tensor, in_sp, out_sp):
    if sev.node_size and sev.world > sev.node_size:
        from deeprec_amd.parallel.hierarchical import hierarchical_all_to_all
        out, oc = hierarchical_all_to_all(tensor, in_sp, sev.node_size)
        assert oc == list(out_sp)
        return out
    return comm.all_to_all_single(tensor, in_sp, out_sp)


def _exchange_lookup(sev: ShardedEmbeddingCollection, uniq, counts, train):
    """Route unique composite keys to owners, lookup there, return
    ([m, D] fp32 embedding rows in uniq order, owner-side context)."""
    coll = sev.local
    w = sev.world
    raw = uniq & ((1 << KEY_BITS) - 1)
    owner = (raw % w).to(torch.int64)
    order_o = torch.argsort(owner, stable=True)
    send_keys = uniq[order_o]
    send_counts = counts[order_o]
    send_splits = torch.bincount(owner, minlength=w)
    recv_splits = comm.exchange_counts(send_splits)
    in_sp, out_sp = send_splits.tolist(), recv_splits.tolist()
    # cap-sizing aid for static_mode: track the largest split observed
    sev._observed_max_split = max(sev._observed_max_split,
                                  max(in_sp), max(out_sp))
    recv_keys = _a2a(sev, send_keys, in_sp, out_sp)
    recv_counts = _a2a(sev, send_counts, in_sp, out_sp)
    uniq2, inv2 = torch.unique(recv_keys, return_inverse=True)
    counts2 = torch.zeros(uniq2.numel(), dtype=recv_counts.dtype,
                          device=uniq.device)
    counts2.index_add_(0, inv2, recv_counts)
    slots2 = coll.storage.lookup_or_create(uniq2, counts2, get_global_step(),
                                           train=train)
    emb2 = coll.storage.gather(uniq2, slots2)        # [m2, D] fp32
    emb_out = emb2[inv2]                              # [n_recv, D]
    if sev.comm_dtype is not None:
        # reduced-precision row exchange. RCCL ships bf16 natively
        # (ncclBfloat16) but has NO int16 dtype; gloo is the opposite —
        # the wire dtype is backend-dependent, payload bits identical.
        lp = emb_out.to(sev.comm_dtype).contiguous()
        if torch.distributed.get_backend() == "gloo":
            emb_back = _a2a(sev, lp.view(torch.int16), out_sp,
                            in_sp).view(sev.comm_dtype).float()
        else:
            emb_back = _a2a(sev, lp, out_sp, in_sp).float()
    else:
        emb_back = _a2a(sev, emb_out.contiguous(), out_sp, in_sp)
    emb = torch.empty_like(emb_back)
    emb[order_o] = emb_back                           # uniq order
    ctx = (order_o, inv2, slots2, uniq2, in_sp, out_sp)
    return emb, ctx


class _ShardedCollectionLookup(torch.autograd.Function):
    @staticmethod
    def forward(ctx, anchor, sev, uniq, counts, inverse, offsets_cat,
                row_ids_cat, order, bounds, chunk_u, chunk_k0, row_coeff,
                weights_cat, batch, out_dtype):
        emb, ex_ctx = _exchange_lookup(sev, uniq, counts, train=True)
        out = sev.local._forward(uniq, None, inverse, offsets_cat,
                                 weights_cat, batch, out_dtype,
                                 emb_override=emb)
        ctx.sev = sev
        ctx.batch = batch
        ctx.weights_cat = weights_cat
        ctx.chunks = (chunk_u, chunk_k0)
        ctx.ex = ex_ctx
        ctx.save_for_backward(uniq, order, bounds, row_ids_cat, row_coeff)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        sev = ctx.sev
        coll = sev.local
        uniq, order, bounds, row_ids_cat, row_coeff = ctx.saved_tensors
        chunk_u, chunk_k0 = ctx.chunks
        order_o, inv2, slots2, uniq2, in_sp, out_sp = ctx.ex
        grad_unique = coll._backward(grad_out, order, bounds, chunk_u,
                                     chunk_k0, row_ids_cat, ctx.weights_cat,
                                     row_coeff, uniq.numel(), ctx.batch)
        grad_send = grad_unique[order_o]
        grad_recv = _a2a(sev, grad_send.contiguous(), in_sp, out_sp)
        grad2 = torch.zeros(uniq2.numel(), grad_recv.shape[1],
                            device=grad_recv.device, dtype=grad_recv.dtype)
        grad2.index_add_(0, inv2.long(), grad_recv)
        coll.accumulate_grad(slots2, uniq2, grad2)
        return (torch.zeros_like(coll._anchor),) + (None,) * 14
