"""Hierarchical (node-aware) all-to-all for multi-node embedding sharding.

A flat all-to-all over N nodes x G GPUs sends W-1 = N*G-1 messages per
rank, most crossing the slow inter-node fabric as many small packets. The
hierarchical exchange does it in two hops (the classic NUMA-aware
algorithm, and the multi-node analog of the reference's SOK two-phase
protocol):

  1. intra-node all-to-all over xGMI: rows for destination rank d are
     handed to the LOCAL peer with index d % G (that peer's cross-node
     group contains d);
  2. inter-node all-to-all within each cross-node group (ranks sharing a
     local index): the bundle for node n lands directly on rank n*G +
     local — the final owner. Two hops, one large inter-node message per
     peer node instead of G small ones.

Row order within each (src, dst) pair is preserved end-to-end, and the
composition is tested equal to the flat exchange (gloo, world=4 as
2 nodes x 2). On one node (n_nodes == 1) this degrades to the flat path.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist

from deeprec_amd.parallel import comm

_GROUPS = {}


def _groups(node_size: int):
    """(intra-node group, inter-node group) for this rank, cached."""
    key = node_size
    if key in _GROUPS:
        return _GROUPS[key]
    w, r = comm.world_size(), comm.rank()
    assert w % node_size == 0, "world size must be a multiple of node_size"
    n_nodes = w // node_size
    intra = inter = None
    for node in range(n_nodes):
        ranks = list(range(node * node_size, (node + 1) * node_size))
        g = dist.new_group(ranks=ranks)
        if r in ranks:
            intra = g
    for local in range(node_size):
        ranks = list(range(local, w, node_size))
        g = dist.new_group(ranks=ranks)
        if r in ranks:
            inter = g
    _GROUPS[key] = (intra, inter)
    return intra, inter


def _a2a_group(inp: torch.Tensor, in_splits: List[int],
               out_splits: List[int], group) -> torch.Tensor:
    """Variable all-to-all within a process subgroup (gloo-safe)."""
    out = torch.empty((sum(out_splits),) + tuple(inp.shape[1:]),
                      dtype=inp.dtype, device=inp.device)
    if dist.get_backend() != "gloo":
        dist.all_to_all_single(out, inp.contiguous(),
                               output_split_sizes=list(out_splits),
                               input_split_sizes=list(in_splits),
                               group=group)
        return out
    ranks = dist.get_process_group_ranks(group)
    me = ranks.index(comm.rank())
    in_off = [0]
    for s in in_splits:
        in_off.append(in_off[-1] + s)
    out_off = [0]
    for s in out_splits:
        out_off.append(out_off[-1] + s)
    reqs = []
    inp = inp.contiguous()
    for i, peer in enumerate(ranks):
        if i == me:
            out[out_off[i]:out_off[i + 1]] = inp[in_off[i]:in_off[i + 1]]
            continue
        if in_splits[i]:
            reqs.append(dist.isend(inp[in_off[i]:in_off[i + 1]], peer))
        if out_splits[i]:
            reqs.append(dist.irecv(out[out_off[i]:out_off[i + 1]], peer))
    for q in reqs:
        q.wait()
    return out


def hierarchical_all_to_all(inp: torch.Tensor, splits_per_rank: List[int],
                            node_size: Optional[int] = None):
    """All-to-all of dim-0 rows where splits_per_rank[d] rows go to rank d.

    Returns (output rows grouped by source rank, out_splits list). With
    node_size None or world<=node_size, falls back to the flat exchange.
    """
    w = comm.world_size()
    if node_size is None or w <= node_size or w % node_size != 0:
        counts = torch.tensor(splits_per_rank, dtype=torch.int64)
        out_counts = comm.exchange_counts(counts)
        out = comm.all_to_all_single(inp, list(splits_per_rank),
                                     out_counts.tolist())
        return out, out_counts.tolist()

    n_nodes = w // node_size
    intra, inter = _groups(node_size)

    # hop 1 (intra-node): local peer g aggregates this node's stripe for
    # node-group g... rows for destination rank d = node(d)*G + local(d)
    # go to local peer local(d)'s inter-group only if that group contains
    # node(d) — every inter-group spans all nodes, so route rows for d to
    # the local peer with index local(d); it forwards them to node(d).
    h1_in = [0] * node_size
    offs = [0]
    for d in range(w):
        offs.append(offs[-1] + splits_per_rank[d])
    # group rows by local(d), keeping (node, src-order) order inside
    seg_by_local = [[] for _ in range(node_size)]
    for d in range(w):
        seg_by_local[d % node_size].append(d)
    idx = torch.arange(inp.shape[0])
    perm_parts = []
    h1_meta = []  # per local peer: list of (dest node, n_rows)
    for local in range(node_size):
        meta = []
        for d in seg_by_local[local]:
            perm_parts.append(idx[offs[d]:offs[d + 1]])
            meta.append((d // node_size, splits_per_rank[d]))
            h1_in[local] += splits_per_rank[d]
        h1_meta.append(meta)
    perm = torch.cat(perm_parts) if perm_parts else idx[:0]
    sendbuf = inp[perm]

    # the full [w, w] splits matrix (one all_gather) provides every
    # hop's exact counts: sender s holds all_splits[s][d] rows for rank d.
    # nccl/RCCL wants CUDA tensors in collectives; gloo wants CPU.
    cdev = inp.device if dist.get_backend() != "gloo" else "cpu"
    all_splits = [torch.zeros(w, dtype=torch.int64, device=cdev)
                  for _ in range(w)]
    dist.all_gather(all_splits,
                    torch.tensor(splits_per_rank, dtype=torch.int64,
                                 device=cdev))
    all_splits = [t.cpu() for t in all_splits]
    me = comm.rank()
    my_node, my_local = me // node_size, me % node_size
    h1_out = [sum(int(all_splits[my_node * node_size + a][d])
                  for d in seg_by_local[my_local])
              for a in range(node_size)]
    stage1 = _a2a_group(sendbuf, h1_in, h1_out, intra)

    # reorder stage1 (grouped by intra source, then by its dest-node
    # sequence) into destination-node-major order for hop 2
    pos = 0
    chunks_by_node = [[] for _ in range(n_nodes)]
    h2_in = [0] * n_nodes
    for a in range(node_size):
        src = my_node * node_size + a
        for d in seg_by_local[my_local]:
            c = int(all_splits[src][d])
            chunks_by_node[d // node_size].append(stage1[pos:pos + c])
            h2_in[d // node_size] += c
            pos += c
    sendbuf2 = torch.cat([torch.cat(cs) if cs else stage1[:0]
                          for cs in chunks_by_node])
    # hop 2 receives, from each node s, every row that node sends to ME
    # (the intermediate on node s with local index my_local routed them)
    h2_out = [sum(int(all_splits[nd * node_size + a][me])
                  for a in range(node_size)) for nd in range(n_nodes)]
    stage2 = _a2a_group(sendbuf2, h2_in, h2_out, inter)

    # routing invariant: rows for rank d went to local peer d%G (hop 1)
    # whose inter-group member on node d//G IS d — two hops deliver.
    # stage2 is node-major then intra-source-major = ascending source
    # rank, the same grouping the flat exchange returns.
    out_counts = [int(all_splits[src][me]) for src in range(w)]
    return stage2, out_counts
