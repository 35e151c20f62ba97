"""Model zoo — capability parity with the reference's modelzoo/ (16 models,
SURVEY.md §2.7). Each class mirrors the reference model's architecture
(cited per class) on the Criteo-shaped schema; all use the
EmbeddingCollection sparse path and fused bf16 MLPs on GPU.
"""
from __future__ import annotations

import math
from typing import List

import torch
import torch.nn as nn

from deeprec_amd.data.synthetic import NUM_DENSE
from deeprec_amd.models.common import RecModelBase, make_mlp


class WDL(RecModelBase):
    """Wide & Deep (reference: modelzoo/wide_and_deep/train.py): wide =
    linear over sparse ids (dim-1 embeddings) + dense; deep = MLP over
    [dense, embeddings]."""

    def __init__(self, embedding_dim=16, deep_sizes=(1024, 512, 256),
                 device="cpu", bf16=True, **kw):
        super().__init__(embedding_dim, device, bf16, name="wdl", **kw)
        from deeprec_amd.embedding.collection import EmbeddingCollection
        self.wide = EmbeddingCollection(
            "wdl/wide", [f"W{i}" for i in range(self.num_sparse)], 16,
            combiners=["sum"] * self.num_sparse, device=self.device_)
        deep_in = NUM_DENSE + self.num_sparse * embedding_dim
        self.deep = make_mlp(deep_sizes, deep_in, device, self.bf16)
        self.head = nn.Linear(deep_sizes[-1], 1)
        self.wide_head = nn.Linear(16 + NUM_DENSE, 1)
        self.to(self.device_)

    def embedding_variables(self):
        return [self.collection, self.wide]

    def forward(self, dense, sparse_ids, train=True):
        emb = self.sparse_feats(sparse_ids, train)      # [B, 26, D]
        if isinstance(sparse_ids, torch.Tensor):
            wide_cat = self.wide.lookup_matrix(sparse_ids,
                                               out_dtype=torch.float32,
                                               train=train)
        else:
            wide_cat = self.wide.lookup(sparse_ids, out_dtype=torch.float32,
                                        train=train)
        wide_sum = wide_cat.view(-1, self.num_sparse, 16).sum(1)
        with self.amp():
            deep_in = torch.cat(
                [dense.to(emb.dtype), emb.flatten(1)], dim=1)
            deep_out = self.deep(deep_in)
        logit = (self.head(deep_out.float()).squeeze(1)
                 + self.wide_head(
                     torch.cat([wide_sum, dense], 1)).squeeze(1))
        return logit


class DeepFM(RecModelBase):
    """DeepFM (reference: modelzoo/deepfm/train.py): FM first+second order
    over embeddings + deep MLP."""

    def __init__(self, embedding_dim=16, deep_sizes=(1024, 512, 256),
                 device="cpu", bf16=True, **kw):
        super().__init__(embedding_dim, device, bf16, name="deepfm", **kw)
        deep_in = NUM_DENSE + self.num_sparse * embedding_dim
        self.deep = make_mlp(deep_sizes, deep_in, device, self.bf16)
        self.head = nn.Linear(deep_sizes[-1] + 1 + NUM_DENSE, 1)
        self.to(self.device_)

    def forward(self, dense, sparse_ids, train=True):
        emb = self.sparse_feats(sparse_ids, train).float()  # [B, F, D]
        # FM 2nd order: 0.5 * (sum^2 - sum of squares), reduced over D
        s = emb.sum(1)
        fm2 = 0.5 * (s * s - (emb * emb).sum(1)).sum(1, keepdim=True)
        with self.amp():
            deep_out = self.deep(torch.cat(
                [dense.to(self.compute_dtype),
                 emb.flatten(1).to(self.compute_dtype)], 1))
        logit = self.head(torch.cat([deep_out.float(), fm2, dense], 1))
        return logit.squeeze(1)


class _CrossNet(nn.Module):
    """DCN cross layers: x_{l+1} = x0 * (w^T x_l) + b + x_l
    (reference: modelzoo/dcn). v2 uses a full matrix W per layer;
    low_rank > 0 factorizes it as U @ V^T (the MLPerf DLRM-DCN recipe,
    reference: modelzoo/mlperf)."""

    def __init__(self, dim, n_layers=3, v2=False, low_rank=0):
        super().__init__()
        self.v2 = v2
        if v2 and low_rank > 0:
            self.ws = nn.ModuleList([
                nn.Sequential(nn.Linear(dim, low_rank, bias=False),
                              nn.Linear(low_rank, dim))
                for _ in range(n_layers)])
        elif v2:
            self.ws = nn.ModuleList(
                [nn.Linear(dim, dim) for _ in range(n_layers)])
        else:
            self.ws = nn.ParameterList(
                [nn.Parameter(torch.randn(dim) / math.sqrt(dim))
                 for _ in range(n_layers)])
            self.bs = nn.ParameterList(
                [nn.Parameter(torch.zeros(dim)) for _ in range(n_layers)])

    def forward(self, x0):
        x = x0
        if self.v2:
            for w in self.ws:
                x = x0 * w(x) + x
        else:
            for w, b in zip(self.ws, self.bs):
                x = x0 * (x @ w).unsqueeze(1) + b + x
        return x


class DCN(RecModelBase):
    """Deep & Cross (reference: modelzoo/dcn, dcnv2)."""

    def __init__(self, embedding_dim=16, deep_sizes=(1024, 512, 256),
                 cross_layers=3, v2=False, device="cpu", bf16=True, **kw):
        name = "dcnv2" if v2 else "dcn"
        super().__init__(embedding_dim, device, bf16, name=name, **kw)
        in_dim = NUM_DENSE + self.num_sparse * embedding_dim
        self.cross = _CrossNet(in_dim, cross_layers, v2)
        self.deep = make_mlp(deep_sizes, in_dim, device, self.bf16)
        self.head = nn.Linear(in_dim + deep_sizes[-1], 1)
        self.to(self.device_)

    def forward(self, dense, sparse_ids, train=True):
        emb = self.sparse_feats(sparse_ids, train)
        x0 = torch.cat([dense, emb.flatten(1).float()], 1)
        cross_out = self.cross(x0)
        with self.amp():
            deep_out = self.deep(x0.to(self.compute_dtype))
        return self.head(
            torch.cat([cross_out, deep_out.float()], 1)).squeeze(1)


class MLPerfDLRMDCN(RecModelBase):
    """MLPerf DLRM-DCNv2 (reference: modelzoo/mlperf): bottom MLP over
    dense features, concat with the embeddings, low-rank DCNv2 cross
    interaction, top MLP. Scaled to the synthetic harness shapes (the
    MLPerf config uses dim 128 / rank 512 on real Criteo)."""

    def __init__(self, embedding_dim=16, bot_sizes=(128, 64, 16),
                 top_sizes=(512, 256), cross_layers=3, low_rank=32,
                 device="cpu", bf16=True, **kw):
        super().__init__(embedding_dim, device, bf16,
                         name="mlperf_dlrm_dcn", **kw)
        assert bot_sizes[-1] == embedding_dim
        self.bot = make_mlp(bot_sizes, NUM_DENSE, device, self.bf16)
        in_dim = (self.num_sparse + 1) * embedding_dim
        self.cross = _CrossNet(in_dim, cross_layers, v2=True,
                               low_rank=low_rank)
        self.top = make_mlp(list(top_sizes) + [1], in_dim, device,
                            self.bf16, final_activation=False)
        self.to(self.device_)

    def forward(self, dense, sparse_ids, train=True):
        emb = self.sparse_feats(sparse_ids, train)
        with self.amp():
            dense_out = self.bot(dense.to(self.compute_dtype))
        x0 = torch.cat([dense_out.float(), emb.flatten(1).float()], 1)
        x = self.cross(x0)
        with self.amp():
            out = self.top(x.to(self.compute_dtype))
        return out.float().squeeze(1)


class DSSM(RecModelBase):
    """Two-tower DSSM (reference: modelzoo/dssm): user tower over half the
    sparse features + dense, item tower over the rest; cosine head."""

    def __init__(self, embedding_dim=16, tower_sizes=(256, 128, 64),
                 device="cpu", bf16=True, **kw):
        super().__init__(embedding_dim, device, bf16, name="dssm", **kw)
        self.n_user = self.num_sparse // 2
        user_in = NUM_DENSE + self.n_user * embedding_dim
        item_in = (self.num_sparse - self.n_user) * embedding_dim
        self.user_tower = make_mlp(tower_sizes, user_in, device, self.bf16,
                                   final_activation=False)
        self.item_tower = make_mlp(tower_sizes, item_in, device, self.bf16,
                                   final_activation=False)
        self.to(self.device_)

    def forward(self, dense, sparse_ids, train=True):
        emb = self.sparse_feats(sparse_ids, train)
        u_in = torch.cat([dense.to(emb.dtype),
                          emb[:, :self.n_user].flatten(1)], 1)
        i_in = emb[:, self.n_user:].flatten(1)
        with self.amp():
            u = self.user_tower(u_in)
            v = self.item_tower(i_in)
        u, v = u.float(), v.float()
        cos = (u * v).sum(1) / (u.norm(dim=1) * v.norm(dim=1) + 1e-8)
        return cos * 5.0  # temperature, logits for BCE


class MMoE(RecModelBase):
    """Multi-gate mixture of experts, 2 tasks (reference: modelzoo/mmoe)."""

    def __init__(self, embedding_dim=16, n_experts=8, expert_sizes=(256, 128),
                 tower_sizes=(64,), n_tasks=2, device="cpu", bf16=True, **kw):
        super().__init__(embedding_dim, device, bf16, name="mmoe", **kw)
        in_dim = NUM_DENSE + self.num_sparse * embedding_dim
        self.n_tasks = n_tasks
        self.experts = nn.ModuleList(
            [make_mlp(expert_sizes, in_dim, device, self.bf16)
             for _ in range(n_experts)])
        self.gates = nn.ModuleList(
            [nn.Linear(in_dim, n_experts) for _ in range(n_tasks)])
        self.towers = nn.ModuleList(
            [make_mlp(list(tower_sizes) + [1], expert_sizes[-1], device,
                      self.bf16, final_activation=False)
             for _ in range(n_tasks)])
        self.to(self.device_)

    def forward(self, dense, sparse_ids, train=True):
        emb = self.sparse_feats(sparse_ids, train)
        x = torch.cat([dense.to(emb.dtype), emb.flatten(1)], 1)
        with self.amp():
            ex = torch.stack([e(x) for e in self.experts], 1)  # [B,E,H]
        ex = ex.float()
        logits = []
        for t in range(self.n_tasks):
            g = torch.softmax(self.gates[t](x.float()), dim=1)  # [B,E]
            mix = torch.einsum("be,beh->bh", g, ex)
            with self.amp():
                logits.append(
                    self.towers[t](mix.to(self.compute_dtype)).float()
                    .squeeze(1))
        return logits

    def loss_fn(self, logits: List[torch.Tensor], labels):
        if not isinstance(labels, (list, tuple)):
            labels = [labels] * len(logits)
        return sum(nn.functional.binary_cross_entropy_with_logits(
            lg.float(), lb.float()) for lg, lb in zip(logits, labels))


class ESMM(RecModelBase):
    """Entire-space multi-task (CTR & CTCVR) (reference: modelzoo/esmm)."""

    def __init__(self, embedding_dim=16, tower_sizes=(256, 128, 64),
                 device="cpu", bf16=True, **kw):
        super().__init__(embedding_dim, device, bf16, name="esmm", **kw)
        in_dim = NUM_DENSE + self.num_sparse * embedding_dim
        self.ctr = make_mlp(list(tower_sizes) + [1], in_dim, device,
                            self.bf16, final_activation=False)
        self.cvr = make_mlp(list(tower_sizes) + [1], in_dim, device,
                            self.bf16, final_activation=False)
        self.to(self.device_)

    def forward(self, dense, sparse_ids, train=True):
        emb = self.sparse_feats(sparse_ids, train)
        x = torch.cat([dense.to(emb.dtype), emb.flatten(1)], 1)
        with self.amp():
            ctr = self.ctr(x).float().squeeze(1)
            cvr = self.cvr(x).float().squeeze(1)
        return ctr, cvr

    def loss_fn(self, logits, labels):
        ctr_logit, cvr_logit = logits
        if isinstance(labels, (list, tuple)):
            click, convert = labels
        else:
            click = convert = labels
        p_ctr = torch.sigmoid(ctr_logit)
        p_ctcvr = p_ctr * torch.sigmoid(cvr_logit)
        l1 = nn.functional.binary_cross_entropy_with_logits(
            ctr_logit, click.float())
        l2 = nn.functional.binary_cross_entropy(
            p_ctcvr.clamp(1e-7, 1 - 1e-7), (click * convert).float())
        return l1 + l2


class SimpleMultiTask(RecModelBase):
    """Shared-bottom two-task model (reference: modelzoo/simple_multitask)."""

    def __init__(self, embedding_dim=16, bottom_sizes=(256, 128),
                 tower_sizes=(64,), device="cpu", bf16=True, **kw):
        super().__init__(embedding_dim, device, bf16, name="smt", **kw)
        in_dim = NUM_DENSE + self.num_sparse * embedding_dim
        self.bottom = make_mlp(bottom_sizes, in_dim, device, self.bf16)
        self.towers = nn.ModuleList([
            make_mlp(list(tower_sizes) + [1], bottom_sizes[-1], device,
                     self.bf16, final_activation=False) for _ in range(2)])
        self.to(self.device_)

    def forward(self, dense, sparse_ids, train=True):
        emb = self.sparse_feats(sparse_ids, train)
        x = torch.cat([dense.to(emb.dtype), emb.flatten(1)], 1)
        with self.amp():
            h = self.bottom(x)
            return [t(h).float().squeeze(1) for t in self.towers]

    loss_fn = MMoE.loss_fn


class DBMTL(RecModelBase):
    """DBMTL: bayesian-style task dependency — task2 consumes task1's
    hidden (reference: modelzoo/dbmtl)."""

    def __init__(self, embedding_dim=16, bottom_sizes=(512, 256),
                 task_sizes=(128, 64), device="cpu", bf16=True, **kw):
        super().__init__(embedding_dim, device, bf16, name="dbmtl", **kw)
        in_dim = NUM_DENSE + self.num_sparse * embedding_dim
        self.bottom = make_mlp(bottom_sizes, in_dim, device, self.bf16)
        self.t1 = make_mlp(task_sizes, bottom_sizes[-1], device, self.bf16)
        self.t2 = make_mlp(task_sizes, bottom_sizes[-1], device, self.bf16)
        self.h1 = nn.Linear(task_sizes[-1], 1)
        self.h2 = nn.Linear(task_sizes[-1] * 2, 1)
        self.to(self.device_)

    def forward(self, dense, sparse_ids, train=True):
        emb = self.sparse_feats(sparse_ids, train)
        x = torch.cat([dense.to(emb.dtype), emb.flatten(1)], 1)
        with self.amp():
            h = self.bottom(x)
            h1 = self.t1(h).float()
            h2 = self.t2(h).float()
        l1 = self.h1(h1).squeeze(1)
        l2 = self.h2(torch.cat([h1, h2], 1)).squeeze(1)
        return [l1, l2]

    loss_fn = MMoE.loss_fn


class PLE(RecModelBase):
    """Progressive layered extraction (CGC single level, 2 tasks)
    (reference: modelzoo/ple)."""

    def __init__(self, embedding_dim=16, expert_sizes=(256, 128),
                 tower_sizes=(64,), n_shared=2, n_spec=2, device="cpu",
                 bf16=True, **kw):
        super().__init__(embedding_dim, device, bf16, name="ple", **kw)
        in_dim = NUM_DENSE + self.num_sparse * embedding_dim
        mk = lambda: make_mlp(expert_sizes, in_dim, device, self.bf16)
        self.shared = nn.ModuleList([mk() for _ in range(n_shared)])
        self.spec = nn.ModuleList(
            [nn.ModuleList([mk() for _ in range(n_spec)]) for _ in range(2)])
        self.gates = nn.ModuleList(
            [nn.Linear(in_dim, n_shared + n_spec) for _ in range(2)])
        self.towers = nn.ModuleList(
            [make_mlp(list(tower_sizes) + [1], expert_sizes[-1], device,
                      self.bf16, final_activation=False) for _ in range(2)])
        self.to(self.device_)

    def forward(self, dense, sparse_ids, train=True):
        emb = self.sparse_feats(sparse_ids, train)
        x = torch.cat([dense.to(emb.dtype), emb.flatten(1)], 1)
        with self.amp():
            shared = [e(x) for e in self.shared]
            outs = []
            for t in range(2):
                experts = torch.stack(
                    shared + [e(x) for e in self.spec[t]], 1).float()
                g = torch.softmax(self.gates[t](x.float()), 1)
                mix = torch.einsum("be,beh->bh", g, experts)
                outs.append(self.towers[t](
                    mix.to(self.compute_dtype)).float().squeeze(1))
        return outs

    loss_fn = MMoE.loss_fn


class MaskNet(RecModelBase):
    """MaskNet serial model: instance-guided masks over feature embedding
    (reference: modelzoo/masknet)."""

    def __init__(self, embedding_dim=16, n_blocks=3, hidden=256,
                 device="cpu", bf16=True, **kw):
        super().__init__(embedding_dim, device, bf16, name="masknet", **kw)
        in_dim = NUM_DENSE + self.num_sparse * embedding_dim
        self.ln = nn.LayerNorm(in_dim)
        self.mask_gens = nn.ModuleList()
        self.blocks = nn.ModuleList()
        d = in_dim
        for _ in range(n_blocks):
            self.mask_gens.append(nn.Sequential(
                nn.Linear(in_dim, hidden), nn.ReLU(),
                nn.Linear(hidden, d)))
            self.blocks.append(nn.Sequential(
                nn.Linear(d, hidden), nn.LayerNorm(hidden), nn.ReLU()))
            d = hidden
        self.head = nn.Linear(d, 1)
        self.to(self.device_)

    def forward(self, dense, sparse_ids, train=True):
        emb = self.sparse_feats(sparse_ids, train).float()
        x0 = self.ln(torch.cat([dense, emb.flatten(1)], 1))
        h = x0
        for gen, blk in zip(self.mask_gens, self.blocks):
            h = blk(gen(x0) * h)
        return self.head(h).squeeze(1)
