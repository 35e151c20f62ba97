"""EV variants beyond the core table.

Capability parity with the reference's secondary embedding systems:
- MultiHashVariable (reference: get_multihash_variable,
  variable_scope.py:2317; docs Multi-Hash-Variable.md): Q-R complementary
  partition compositional embedding — two (or more) small tables indexed
  by complementary key partitions (k // m, k % m), combined by
  add/mul/concat; memory O(sqrt(N)) instead of O(N).
- DynamicEmbeddingVariable (reference:
  get_dynamic_dimension_embedding_variable, variable_scope.py:1642;
  blocknum-indexed gather embedding_ops.py:86 _gather_fae): per-key
  frequency-adaptive dimension — rows are block_num blocks of block_dim;
  low-frequency keys expose only their first blocks (tail zeroed).
- adaptive_embedding_lookup_sparse (reference: embedding_ops.py:668):
  hot ids above a frequency threshold live in the dynamic EV; cold ids
  fall back to a fixed hash-bucket table.
"""
from __future__ import annotations

from typing import Optional

import torch

from deeprec_amd.embedding.options import EmbeddingVariableOption
from deeprec_amd.embedding.ragged import RaggedIds
from deeprec_amd.embedding.variable import EmbeddingVariable
from deeprec_amd.embedding.lookup import embedding_lookup


class MultiHashVariable:
    """Compositional embedding over complementary partitions."""

    def __init__(self, name: str, dims, num_of_partitions: int = 2,
                 complementary_strategy: str = "Q-R", operation: str = "add",
                 embedding_dim: int = None, device=None,
                 ev_option: Optional[EmbeddingVariableOption] = None):
        assert complementary_strategy == "Q-R"
        assert num_of_partitions == 2
        assert operation in ("add", "mul", "concat")
        self.name = name
        self.operation = operation
        # dims: [rows_q, rows_r] partition sizes (quotient modulus = rows_r)
        self.modulus = int(dims[1]) if hasattr(dims, "__len__") else int(dims)
        self.embedding_dim = embedding_dim or 16
        self.q = EmbeddingVariable(f"{name}/Q", self.embedding_dim,
                                   ev_option=ev_option, device=device)
        self.r = EmbeddingVariable(f"{name}/R", self.embedding_dim,
                                   ev_option=ev_option, device=device)

    @property
    def dim(self):
        return (self.embedding_dim * 2 if self.operation == "concat"
                else self.embedding_dim)

    def embedding_variables(self):
        return [self.q, self.r]

    def lookup(self, ids: torch.Tensor, train: bool = True) -> torch.Tensor:
        q_ids = ids // self.modulus
        r_ids = ids % self.modulus
        eq = embedding_lookup(self.q, q_ids, train=train)
        er = embedding_lookup(self.r, r_ids, train=train)
        if self.operation == "add":
            return eq + er
        if self.operation == "mul":
            return eq * er
        return torch.cat([eq, er], dim=-1)


def get_multihash_variable(name, dims, num_of_partitions=2,
                           complementary_strategy="Q-R", operation="add",
                           embedding_dim=None, device=None, ev_option=None):
    return MultiHashVariable(name, dims, num_of_partitions,
                             complementary_strategy, operation,
                             embedding_dim, device, ev_option)


class DynamicEmbeddingVariable(EmbeddingVariable):
    """Frequency-adaptive dimension: row = block_num blocks of block_dim;
    a key at frequency f exposes blocks up to the largest threshold it has
    crossed (thresholds double per block by default)."""

    def __init__(self, name: str, embedding_block_dim: int,
                 embedding_block_num: int, block_thresholds=None,
                 ev_option=None, device=None, **kw):
        super().__init__(name, embedding_block_dim * embedding_block_num,
                         ev_option=ev_option, device=device, **kw)
        self.block_dim = embedding_block_dim
        self.block_num = embedding_block_num
        if block_thresholds is None:
            block_thresholds = [2 ** i for i in range(embedding_block_num)]
        self.block_thresholds = torch.tensor(
            sorted(block_thresholds), dtype=torch.int64, device=self.device)

    def blocks_for(self, freqs: torch.Tensor) -> torch.Tensor:
        """number of active blocks per key (>=1)."""
        return torch.searchsorted(
            self.block_thresholds, freqs, right=True).clamp(min=1)

    def lookup(self, ids: torch.Tensor, train: bool = True) -> torch.Tensor:
        """[n, block_num*block_dim] with inactive tail blocks zeroed."""
        out = embedding_lookup(self, ids, train=train)
        freqs = self.get_frequency(ids.reshape(-1))
        nblocks = self.blocks_for(freqs)            # [n]
        block_idx = torch.arange(self.block_num, device=out.device)
        mask = (block_idx.unsqueeze(0) < nblocks.unsqueeze(1))  # [n, Bn]
        mask = mask.repeat_interleave(self.block_dim, dim=1).to(out.dtype)
        return out * mask.reshape(*out.shape)


def get_dynamic_dimension_embedding_variable(
        name, embedding_block_dim, embedding_block_num, ev_option=None,
        device=None, **kw):
    return DynamicEmbeddingVariable(name, embedding_block_dim,
                                    embedding_block_num,
                                    ev_option=ev_option, device=device, **kw)


def adaptive_embedding_lookup_sparse(
        ev: EmbeddingVariable, static_weight: torch.Tensor,
        sp_ids: RaggedIds, threshold: int, combiner: str = "mean",
        train: bool = True) -> torch.Tensor:
    """Hot ids (frequency >= threshold after this batch) read/train the EV;
    cold ids use row (id % bucket) of the fixed `static_weight` table.
    (reference: adaptive_embedding_lookup_sparse, embedding_ops.py:668)"""
    values = sp_ids.values
    uniq, inverse = torch.unique(values, return_inverse=True)
    freqs = ev.get_frequency(uniq) if train else ev.get_frequency(uniq)
    # count this batch too (the lookup below will record it)
    counts = torch.bincount(inverse, minlength=uniq.numel())
    hot_u = (freqs + counts.to(freqs.dtype)) >= threshold
    hot = hot_u[inverse]

    out_dim = static_weight.shape[1]
    bucket = static_weight.shape[0]
    emb_nnz = torch.empty(values.numel(), out_dim,
                          device=static_weight.device,
                          dtype=static_weight.dtype)
    if bool(hot.any()):
        hot_vals = values[hot]
        emb_hot = embedding_lookup(ev, hot_vals, train=train)
        emb_nnz[hot] = emb_hot.to(emb_nnz.dtype)
    if bool((~hot).any()):
        cold_vals = values[~hot]
        emb_nnz[~hot] = static_weight[cold_vals % bucket]
    from deeprec_amd.ops import functional as F
    row_ids = sp_ids.row_ids()
    n = values.numel()
    ar = torch.arange(n, dtype=torch.int32, device=values.device)
    return F.pooled_forward(emb_nnz, ar, sp_ids.offsets, row_ids, combiner,
                            sp_ids.weights)
