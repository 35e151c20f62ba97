"""Ragged sparse-id batches.

The framework's sparse-feature input format: a flat int64 id tensor plus
row offsets (CSR style). This replaces the reference's tf.SparseTensor
input to embedding_lookup_sparse — on GPU a (values, offsets) pair is the
layout the fused lookup kernels consume directly, with no COO→CSR
conversion on the hot path.
"""
from __future__ import annotations

import torch


class RaggedIds:
    """Batch of variable-length id lists: values[nnz], offsets[batch+1]."""

    __slots__ = ("values", "offsets", "weights")

    def __init__(self, values: torch.Tensor, offsets: torch.Tensor,
                 weights: torch.Tensor = None):
        assert values.dtype in (torch.int64, torch.int32)
        assert offsets.dtype in (torch.int64, torch.int32)
        self.values = values
        self.offsets = offsets
        self.weights = weights

    @property
    def batch_size(self) -> int:
        return self.offsets.numel() - 1

    @property
    def nnz(self) -> int:
        return self.values.numel()

    def to(self, device, non_blocking: bool = False) -> "RaggedIds":
        return RaggedIds(
            self.values.to(device, non_blocking=non_blocking),
            self.offsets.to(device, non_blocking=non_blocking),
            None if self.weights is None
            else self.weights.to(device, non_blocking=non_blocking),
        )

    def pin_memory(self) -> "RaggedIds":
        return RaggedIds(
            self.values.pin_memory(), self.offsets.pin_memory(),
            None if self.weights is None else self.weights.pin_memory())

    @staticmethod
    def from_dense(ids: torch.Tensor) -> "RaggedIds":
        """[batch, k] dense id matrix -> ragged with uniform row length."""
        b, k = ids.shape
        offsets = torch.arange(0, (b + 1) * k, k, dtype=torch.int32,
                               device=ids.device)
        return RaggedIds(ids.reshape(-1).contiguous(), offsets)

    @staticmethod
    def from_lists(lists, device="cpu") -> "RaggedIds":
        lens = [len(l) for l in lists]
        offsets = torch.zeros(len(lists) + 1, dtype=torch.int32)
        offsets[1:] = torch.cumsum(torch.tensor(lens, dtype=torch.int32), 0)
        values = torch.tensor([x for l in lists for x in l], dtype=torch.int64)
        return RaggedIds(values.to(device), offsets.to(device))

    def row_ids(self) -> torch.Tensor:
        """int32[nnz] tensor mapping each value to its batch row."""
        lengths = (self.offsets[1:] - self.offsets[:-1]).to(torch.int64)
        return torch.repeat_interleave(
            torch.arange(self.batch_size, device=self.values.device,
                         dtype=torch.int32), lengths)
