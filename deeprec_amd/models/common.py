"""Shared model-zoo infrastructure.

Every model mirrors its reference counterpart's architecture
(modelzoo/<model>/train.py) on the Criteo-shaped input schema:
13 continuous + 26 categorical features (matrix layout [B, 26] or ragged),
EmbeddingCollection-backed sparse side, bf16 fused MLPs on GPU.
"""
from __future__ import annotations

from typing import List, Optional, Sequence

import torch
import torch.nn as nn

from deeprec_amd.data.synthetic import NUM_SPARSE
from deeprec_amd.embedding import EmbeddingVariableOption
from deeprec_amd.embedding.collection import EmbeddingCollection


def make_mlp(sizes: Sequence[int], in_dim: int, device,
             bf16: bool, final_activation: bool = True) -> nn.Sequential:
    if bf16 and torch.device(device).type == "cuda":
        from deeprec_amd.ops.fused_mlp import fused_mlp
        return fused_mlp(list(sizes), in_dim, final_activation)
    layers: List[nn.Module] = []
    d = in_dim
    for i, h in enumerate(sizes):
        layers.append(nn.Linear(d, h))
        if final_activation or i + 1 < len(sizes):
            layers.append(nn.ReLU(inplace=True))
        d = h
    return nn.Sequential(*layers)


class Dice(nn.Module):
    """Dice activation (reference: grappler dice_fusion.cc + DIN paper):
    x * p + alpha * x * (1 - p), p = sigmoid(batchnorm(x))."""

    def __init__(self, dim: int, eps: float = 1e-8):
        super().__init__()
        self.bn = nn.BatchNorm1d(dim, eps=eps, affine=False)
        self.alpha = nn.Parameter(torch.zeros(dim))

    def forward(self, x):
        shape = x.shape
        p = torch.sigmoid(self.bn(x.reshape(-1, shape[-1]).float()))
        p = p.reshape(shape).to(x.dtype)
        return x * p + self.alpha.to(x.dtype) * x * (1 - p)


class RecModelBase(nn.Module):
    """Criteo-schema recommendation model base: one EmbeddingCollection for
    the 26 categorical features + helpers shared by the zoo."""

    def __init__(self, embedding_dim: int = 16, device="cpu",
                 bf16: bool = True, num_sparse: int = NUM_SPARSE,
                 ev_option: Optional[EmbeddingVariableOption] = None,
                 name: str = "model", sharded: bool = False,
                 combiners: Optional[List[str]] = None):
        super().__init__()
        self.device_ = torch.device(device)
        self.embedding_dim = embedding_dim
        self.num_sparse = num_sparse
        self.bf16 = bf16 and self.device_.type == "cuda"
        cls = EmbeddingCollection
        if sharded:
            from deeprec_amd.parallel.sharded_collection import (
                ShardedEmbeddingCollection)
            cls = ShardedEmbeddingCollection
        self.collection = cls(
            f"{name}/sparse", [f"C{i + 1}" for i in range(num_sparse)],
            embedding_dim, ev_option=ev_option,
            combiners=combiners or ["mean"] * num_sparse,
            device=self.device_)

    @property
    def compute_dtype(self):
        return torch.bfloat16 if self.bf16 else torch.float32

    def embedding_variables(self):
        return [self.collection]

    def sparse_feats(self, sparse_ids, train=True) -> torch.Tensor:
        """[B, num_sparse, dim]."""
        if isinstance(sparse_ids, torch.Tensor):
            cat = self.collection.lookup_matrix(
                sparse_ids, out_dtype=self.compute_dtype, train=train)
        else:
            cat = self.collection.lookup(
                sparse_ids, out_dtype=self.compute_dtype, train=train)
        return cat.view(-1, self.num_sparse, self.embedding_dim)

    def amp(self):
        import contextlib
        if self.bf16:
            return torch.autocast(device_type="cuda", dtype=torch.bfloat16)
        return contextlib.nullcontext()

    def loss_fn(self, logits: torch.Tensor, labels: torch.Tensor):
        return nn.functional.binary_cross_entropy_with_logits(
            logits.float(), labels.float())
