"""Synthetic Criteo-shaped data generation.

There is no dataset network access in this environment; benchmarks run on
synthetic data of the reference's shape (13 continuous + 26 categorical
Criteo columns, per-column cardinalities from the reference's DLRM
HASH_BUCKET_SIZES table, modelzoo/dlrm/train.py:39-66) with a Zipf-like id
distribution so hash-table behavior (hot keys, long tail) is realistic.
"""
from __future__ import annotations

from typing import Optional

import torch

from deeprec_amd.embedding.ragged import RaggedIds

# Criteo categorical cardinalities (C1..C26), mirroring the reference's
# bucket sizes so table-size distribution matches the benchmark config.
CRITEO_CARDINALITIES = [
    2500, 2000, 5000000, 1500000, 1000, 100, 20000, 4000, 20, 100000,
    10000, 5000000, 40000, 100, 100, 3000000, 50, 10000, 4000, 20,
    4000000, 100, 100, 250000, 400, 100000,
]
NUM_DENSE = 13
NUM_SPARSE = 26


class CriteoSyntheticDataset:
    """Iterable of (dense [B,13] float32, sparse list[26] RaggedIds,
    labels [B] float32). Deterministic per (seed, rank)."""

    def __init__(self, batch_size: int, device="cpu", seed: int = 1234,
                 rank: int = 0, zipf_alpha: float = 1.05,
                 multi_hot: Optional[int] = None,
                 cardinalities=None, num_batches: Optional[int] = None,
                 matrix_format: bool = False):
        self.batch_size = batch_size
        self.device = torch.device(device)
        self.gen = torch.Generator(device="cpu")
        self.gen.manual_seed(seed + 1000003 * rank)
        self.cardinalities = list(cardinalities or CRITEO_CARDINALITIES)
        self.zipf_alpha = zipf_alpha
        self.multi_hot = multi_hot  # ids per sample per feature (None=1)
        self.num_batches = num_batches
        # matrix_format: sparse ids as one [batch, 26] int64 tensor (the
        # 1-hot Criteo layout) feeding EmbeddingCollection.lookup_matrix
        self.matrix_format = matrix_format
        # precompute per-feature Zipf CDF over a capped support for sampling
        self._cdfs = []
        for card in self.cardinalities:
            support = min(card, 1 << 20)
            ranks = torch.arange(1, support + 1, dtype=torch.float64)
            p = ranks.pow(-zipf_alpha)
            self._cdfs.append((torch.cumsum(p, 0) / p.sum()).float())

    def _sample_ids(self, fidx: int, n: int) -> torch.Tensor:
        cdf = self._cdfs[fidx]
        u = torch.rand(n, generator=self.gen)
        idx = torch.searchsorted(cdf, u).clamp(max=cdf.numel() - 1)
        card = self.cardinalities[fidx]
        if card > cdf.numel():
            # spread the capped support over the full cardinality
            idx = (idx.to(torch.int64) * (card // cdf.numel())
                   + (idx.to(torch.int64) % max(1, card // cdf.numel())))
        return idx.to(torch.int64)

    def next_batch(self):
        b = self.batch_size
        dense = torch.randn(b, NUM_DENSE, generator=self.gen)
        if self.matrix_format:
            ids = torch.stack([self._sample_ids(f, b)
                               for f in range(NUM_SPARSE)], dim=1)
            labels = (torch.rand(b, generator=self.gen) < 0.3).float()
            if self.device.type != "cpu":
                dense = dense.to(self.device, non_blocking=True)
                ids = ids.to(self.device, non_blocking=True)
                labels = labels.to(self.device, non_blocking=True)
            return dense, ids, labels
        sparse = []
        k = self.multi_hot or 1
        for f in range(NUM_SPARSE):
            ids = self._sample_ids(f, b * k)
            offsets = torch.arange(0, (b + 1) * k, k, dtype=torch.int32)
            sparse.append(RaggedIds(ids, offsets))
        labels = (torch.rand(b, generator=self.gen) < 0.3).float()
        if self.device.type != "cpu":
            dense = dense.to(self.device, non_blocking=True)
            sparse = [s.to(self.device, non_blocking=True) for s in sparse]
            labels = labels.to(self.device, non_blocking=True)
        return dense, sparse, labels

    def next_seq_batch(self, seq_len: int = 50, item_cardinality: int = 1 << 20):
        """Behavior-sequence batch for DIN/DIEN/BST: (dense, sparse ids
        [B, 26], seq_ids [B, T] zero-padded, target_ids [B], labels)."""
        b = self.batch_size
        dense = torch.randn(b, NUM_DENSE, generator=self.gen)
        ids = torch.stack([self._sample_ids(f, b)
                           for f in range(NUM_SPARSE)], dim=1)
        lens = torch.randint(1, seq_len + 1, (b,), generator=self.gen)
        seq = torch.randint(1, item_cardinality, (b, seq_len),
                            generator=self.gen)
        pos = torch.arange(seq_len).unsqueeze(0)
        seq = seq * (pos < lens.unsqueeze(1))
        target = torch.randint(1, item_cardinality, (b,),
                               generator=self.gen)
        labels = (torch.rand(b, generator=self.gen) < 0.3).float()
        if self.device.type != "cpu":
            dense, ids, seq, target, labels = (
                t.to(self.device, non_blocking=True)
                for t in (dense, ids, seq, target, labels))
        return dense, ids, seq, target, labels

    def __iter__(self):
        n = 0
        while self.num_batches is None or n < self.num_batches:
            yield self.next_batch()
            n += 1
