"""DLRM — deep learning recommendation model.

Capability parity with the reference model (modelzoo/dlrm/train.py:70-283):
bottom MLP over 13 dense features -> [*, 16]; 26 EmbeddingVariable lookups
(dim 16); pairwise-dot (or concat) feature interaction; top MLP
[512, 256, 1]; BCE loss. bf16 compute with fp32 master weights mirrors the
reference's keep_weights bf16 scope.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn

from deeprec_amd.data.synthetic import NUM_SPARSE
from deeprec_amd.embedding import (
    EmbeddingVariable, EmbeddingVariableOption,
    group_embedding_lookup_sparse,
)


def _mlp(sizes, in_dim, final_activation=True):
    layers: List[nn.Module] = []
    d = in_dim
    for i, h in enumerate(sizes):
        layers.append(nn.Linear(d, h))
        if final_activation or i + 1 < len(sizes):
            layers.append(nn.ReLU(inplace=True))
        d = h
    return nn.Sequential(*layers)


class DLRM(nn.Module):
    def __init__(self, embedding_dim: int = 16,
                 mlp_bot=(512, 256, 64, 16), mlp_top=(512, 256),
                 interaction_op: str = "dot", bf16: bool = True,
                 device="cpu", ev_option: Optional[EmbeddingVariableOption] = None,
                 num_sparse: int = NUM_SPARSE, name_prefix: str = "dlrm",
                 sharded: bool = False, use_collection: bool = True):
        super().__init__()
        assert mlp_bot[-1] == embedding_dim, \
            "bottom MLP output must match embedding dim (dot interaction)"
        self.embedding_dim = embedding_dim
        self.interaction_op = interaction_op
        self.bf16 = bf16
        self.device_ = torch.device(device)
        self.num_sparse = num_sparse

        # dense input padded 13 -> 16: K=13 GEMMs pick poor hipBLASLt tiles
        # (measured 148us for 8192x512x13); zero-padding is math-identical
        self.dense_in = 16
        use_fused = bf16 and torch.device(device).type == "cuda"
        n_f = num_sparse + 1
        n_pairs = n_f * (n_f - 1) // 2
        # pad pair count to a multiple of 16 so the fused top-MLP GEMM gets
        # aligned K fragments (351 -> 352 for the 26-feature config)
        self.n_pairs_pad = (n_pairs + 15) & ~15
        top_in = (embedding_dim + self.n_pairs_pad
                  if interaction_op == "dot" else n_f * embedding_dim)
        if use_fused:
            from deeprec_amd.ops.fused_mlp import fused_mlp
            self.mlp_bot = fused_mlp(mlp_bot, self.dense_in)
            self.mlp_top = fused_mlp(list(mlp_top) + [1], top_in,
                                     final_activation=False)
        else:
            self.mlp_bot = _mlp(mlp_bot, self.dense_in)
            self.mlp_top = _mlp(list(mlp_top) + [1], top_in,
                                final_activation=False)
        self.to(self.device_)

        self.collection = None
        if sharded:
            from deeprec_amd.parallel.sharded_collection import (
                ShardedEmbeddingCollection)
            self.collection = ShardedEmbeddingCollection(
                f"{name_prefix}/sparse",
                [f"C{i+1}" for i in range(num_sparse)], embedding_dim,
                ev_option=ev_option, combiners=["mean"] * num_sparse,
                device=self.device_,
                comm_dtype=torch.bfloat16 if self.bf16 else None)
            self.evs = []
        elif use_collection:
            from deeprec_amd.embedding.collection import EmbeddingCollection
            self.collection = EmbeddingCollection(
                f"{name_prefix}/sparse",
                [f"C{i+1}" for i in range(num_sparse)], embedding_dim,
                ev_option=ev_option, combiners=["mean"] * num_sparse,
                device=self.device_)
            self.evs = []
        else:
            self.evs = [
                EmbeddingVariable(f"{name_prefix}/C{i+1}", embedding_dim,
                                  ev_option=ev_option, device=self.device_)
                for i in range(num_sparse)]

    def embedding_variables(self):
        return [self.collection] if self.collection is not None else self.evs

    def _interact(self, feats: torch.Tensor) -> torch.Tensor:
        """feats: [B, F, D] -> pairwise dots, upper triangle (i<j), padded
        (reference: _dot_op, modelzoo/dlrm/train.py:121-132)."""
        from deeprec_amd.ops.fused_mlp import dot_interaction
        return dot_interaction(feats, self.n_pairs_pad)

    def forward(self, dense: torch.Tensor, sparse_ids,
                train: bool = True) -> torch.Tensor:
        compute_dtype = torch.bfloat16 if (
            self.bf16 and dense.device.type == "cuda") else torch.float32
        if self.collection is not None:
            if isinstance(sparse_ids, torch.Tensor):
                emb_cat = self.collection.lookup_matrix(
                    sparse_ids, out_dtype=compute_dtype, train=train)
            else:
                emb_cat = self.collection.lookup(sparse_ids,
                                                 out_dtype=compute_dtype,
                                                 train=train)
            emb_feats = emb_cat.view(-1, self.num_sparse, self.embedding_dim)
        else:
            emb_list = group_embedding_lookup_sparse(
                self.evs, sparse_ids, combiners=["mean"] * len(self.evs),
                out_dtype=compute_dtype, train=train)
            emb_feats = torch.stack(list(emb_list), dim=1)
        import contextlib
        amp = (torch.autocast(device_type="cuda", dtype=torch.bfloat16)
               if compute_dtype == torch.bfloat16 else contextlib.nullcontext())
        if dense.shape[1] < self.dense_in:
            dense = nn.functional.pad(
                dense, (0, self.dense_in - dense.shape[1]))
        with amp:
            bot = self.mlp_bot(dense)
            if self.interaction_op == "dot":
                # cat-fused interaction: [bot | pair dots] in one kernel
                # per direction (no feats assembly, no top-input cat)
                from deeprec_amd.ops.fused_mlp import dot_interaction_cat
                top_in = dot_interaction_cat(bot, emb_feats,
                                             self.n_pairs_pad)
            else:
                feats = torch.cat(
                    [bot.unsqueeze(1), emb_feats.to(bot.dtype)], dim=1)
                top_in = feats.flatten(1)
            logits = self.mlp_top(top_in)
        return logits.squeeze(1).float()

    def loss_fn(self, logits: torch.Tensor, labels: torch.Tensor):
        return nn.functional.binary_cross_entropy_with_logits(logits, labels)
