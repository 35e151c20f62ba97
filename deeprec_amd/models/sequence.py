"""Sequence (user-behavior) models: DIN, DIEN, BST.

Reference: modelzoo/din, modelzoo/dien (GRU + attention + AUGRU),
modelzoo/bst (transformer block). Behavior sequences are whole per-sample
sequences (~tens of items) — the reference never shards sequence length
(SURVEY.md §5 "long context: not present"); capability = sequence feature
columns + these architectures.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from deeprec_amd.data.synthetic import NUM_DENSE
from deeprec_amd.embedding import EmbeddingVariable, embedding_lookup
from deeprec_amd.models.common import RecModelBase, make_mlp


class _SeqBase(RecModelBase):
    """Adds an item EmbeddingVariable for behavior sequences + target id."""

    def __init__(self, embedding_dim=16, item_dim=32, device="cpu",
                 bf16=True, name="seq", num_sparse=10, **kw):
        super().__init__(embedding_dim, device, bf16, name=name,
                         num_sparse=num_sparse, **kw)
        self.item_dim = item_dim
        self.item_ev = EmbeddingVariable(f"{name}/items", item_dim,
                                         device=self.device_)

    def embedding_variables(self):
        return [self.collection, self.item_ev]

    def seq_emb(self, seq_ids: torch.Tensor, train=True) -> torch.Tensor:
        """[B, T] -> [B, T, item_dim]; id 0 = padding -> zero embedding.

        Padding is excluded from the lookup entirely (reference
        semantics: masked positions train nothing). Measured both ways:
        looking up every position and masking afterwards is 7x SLOWER —
        the ~200k-occurrence padding key dominates the unpooled lookup
        and its CSR grad scatter even with 32-way hot-key splitting."""
        flat = seq_ids.reshape(-1)
        mask = flat > 0
        emb = torch.zeros(flat.numel(), self.item_dim,
                          device=flat.device)
        emb = emb.index_put(
            (mask.nonzero().squeeze(1),),
            embedding_lookup(self.item_ev, flat[mask],
                             train=train).float())
        return emb.reshape(*seq_ids.shape, self.item_dim)


class DIN(_SeqBase):
    """Deep Interest Network (reference: modelzoo/din/train.py):
    target-conditioned attention over the behavior sequence."""

    def __init__(self, embedding_dim=16, item_dim=32, att_hidden=64,
                 mlp_sizes=(256, 128, 64), device="cpu", bf16=True, **kw):
        super().__init__(embedding_dim, item_dim, device, bf16, name="din",
                         **kw)
        if self.bf16:
            from deeprec_amd.ops.fused_mlp import FusedLinear
            self.att = nn.Sequential(
                FusedLinear(item_dim * 4, att_hidden, activation="sigmoid"),
                FusedLinear(att_hidden, 1, activation=None))
        else:
            self.att = nn.Sequential(
                nn.Linear(item_dim * 4, att_hidden), nn.Sigmoid(),
                nn.Linear(att_hidden, 1))
        in_dim = NUM_DENSE + self.num_sparse * embedding_dim + item_dim * 2
        self.mlp = make_mlp(list(mlp_sizes) + [1], in_dim, device, self.bf16,
                            final_activation=False)
        self.to(self.device_)

    def attend(self, seq, target, mask):
        """seq [B,T,D], target [B,D] -> [B,D] attention-pooled. The
        [s, t, s-t, s*t] feature build is ONE fused bf16 kernel on GPU
        (the torch expand/sub/mul/cat path was 5 kernels at ~2x the
        [B,T,4D] HBM traffic)."""
        b, t, d = seq.shape
        if seq.device.type == "cuda" and self.bf16:
            from deeprec_amd.ops.fused_attention import din_att_features
            att_in = din_att_features(seq, target)          # [B*T, 4D]
            scores = self.att(att_in).float().reshape(b, t)
        else:
            cd = self.compute_dtype if self.bf16 else seq.dtype
            s16 = seq.to(cd)
            t16 = target.to(cd).unsqueeze(1).expand_as(s16)
            att_in = torch.cat([s16, t16, s16 - t16, s16 * t16], dim=2)
            scores = self.att(att_in).float().squeeze(2)    # [B, T]
        if seq.device.type == "cuda":
            from deeprec_amd.ops.fused_attention import masked_softmax_pool
            return masked_softmax_pool(scores, seq, mask)
        scores = scores.masked_fill(~mask, -1e9)
        w = torch.softmax(scores, dim=1)
        return (w.unsqueeze(2) * seq).sum(1)

    def forward(self, dense, sparse_ids, seq_ids, target_ids, train=True):
        emb = self.sparse_feats(sparse_ids, train)
        seq = self.seq_emb(seq_ids, train)
        target = embedding_lookup(self.item_ev, target_ids,
                                  train=train).float()
        mask = seq_ids > 0
        interest = self.attend(seq, target, mask)
        x = torch.cat([dense, emb.flatten(1).float(), interest, target], 1)
        with self.amp():
            out = self.mlp(x.to(self.compute_dtype))
        return out.float().squeeze(1)


class DIEN(_SeqBase):
    """Deep Interest Evolution Network (reference: modelzoo/dien/
    train.py:207-253): GRU interest extractor + attention-gated GRU
    (AUGRU) interest evolution."""

    def __init__(self, embedding_dim=16, item_dim=32, gru_hidden=32,
                 att_hidden=64, mlp_sizes=(256, 128, 64), device="cpu",
                 bf16=True, use_aux_loss=True, aux_alpha=1.0, **kw):
        super().__init__(embedding_dim, item_dim, device, bf16, name="dien",
                         **kw)
        # auxiliary next-item supervision on the GRU hidden states
        # (reference: _auxiliary_loss, modelzoo/dien/train.py:231-251:
        # click pair [h_t; e_{t+1}] vs negative pair, masked log-loss
        # added to the main objective). Negatives = within-batch
        # permutation of the real next items (no fabricated ids).
        self.use_aux_loss = use_aux_loss
        self.aux_alpha = aux_alpha
        self._aux_loss = None
        self.aux_net = nn.Sequential(
            nn.Linear(gru_hidden + item_dim, 64), nn.Sigmoid(),
            nn.Linear(64, 1))
        from deeprec_amd.ops.fused_gru import FusedGRU
        # fused single-kernel recurrences (MIOpen's RNN path ran this at
        # 44 ms/step; the python AUGRU loop added ~600 launches)
        self.gru = FusedGRU(item_dim, gru_hidden)
        self.augru = FusedGRU(gru_hidden, gru_hidden)
        if self.bf16:
            from deeprec_amd.ops.fused_mlp import FusedLinear
            self.att = nn.Sequential(
                FusedLinear(gru_hidden * 2, att_hidden,
                            activation="sigmoid"),
                FusedLinear(att_hidden, 1, activation=None))
        else:
            self.att = nn.Sequential(
                nn.Linear(gru_hidden * 2, att_hidden), nn.Sigmoid(),
                nn.Linear(att_hidden, 1))
        self.target_proj = nn.Linear(item_dim, gru_hidden)
        in_dim = (NUM_DENSE + self.num_sparse * embedding_dim
                  + gru_hidden + item_dim)
        self.mlp = make_mlp(list(mlp_sizes) + [1], in_dim, device, self.bf16,
                            final_activation=False)
        self.to(self.device_)

    def forward(self, dense, sparse_ids, seq_ids, target_ids, train=True):
        emb = self.sparse_feats(sparse_ids, train)
        seq = self.seq_emb(seq_ids, train)               # [B,T,Di]
        target = embedding_lookup(self.item_ev, target_ids,
                                  train=train).float()   # [B,Di]
        mask = (seq_ids > 0).float()
        h_seq = self.gru(seq)                            # [B,T,H]
        if train and self.use_aux_loss and seq.shape[1] > 1:
            self._aux_loss = self._auxiliary_loss(h_seq, seq, mask)
        tgt_h = self.target_proj(target)                 # [B,H]
        cd = self.compute_dtype if self.bf16 else h_seq.dtype
        att_in = torch.cat(
            [h_seq.to(cd), tgt_h.to(cd).unsqueeze(1).expand_as(h_seq)], 2)
        scores = self.att(att_in).float().squeeze(2)
        scores = scores.masked_fill(mask == 0, -1e9)
        alpha = torch.softmax(scores, 1) * mask          # [B,T]
        # AUGRU: attention gates the update — h'=(1-a)h + a*GRU(h,x);
        # a=0 at padding leaves h untouched, so [:, -1] is the final state
        h = self.augru(h_seq, alpha)[:, -1]
        x = torch.cat([dense, emb.flatten(1).float(), h, target], 1)
        with self.amp():
            out = self.mlp(x.to(self.compute_dtype))
        return out.float().squeeze(1)

    def _auxiliary_loss(self, h_seq, seq, mask):
        """Masked BCE over click pairs [h_t; e_{t+1}] vs no-click pairs
        [h_t; e_shuffled_{t+1}] (reference semantics; sigmoid+BCE is the
        numerically-stable form of its softmax+log)."""
        h = h_seq[:, :-1].float()                        # [B,T-1,H]
        click = seq[:, 1:].float()                       # real next items
        perm = torch.randperm(seq.shape[0], device=seq.device)
        noclick = click[perm]                            # in-batch negatives
        m = (mask[:, :-1] * mask[:, 1:])                 # both steps valid
        m = m * (perm != torch.arange(seq.shape[0],
                                      device=seq.device)).float() \
            .unsqueeze(1)                                # drop self-pairs
        if m.sum() == 0:
            return None
        logit_c = self.aux_net(torch.cat([h, click], -1)).squeeze(2)
        logit_n = self.aux_net(torch.cat([h, noclick], -1)).squeeze(2)
        bce = nn.functional.binary_cross_entropy_with_logits
        lc = bce(logit_c, torch.ones_like(logit_c), reduction="none")
        ln = bce(logit_n, torch.zeros_like(logit_n), reduction="none")
        return ((lc + ln) * m).sum() / m.sum()

    def loss_fn(self, logits, labels):
        loss = super().loss_fn(logits, labels)
        if self._aux_loss is not None:
            loss = loss + self.aux_alpha * self._aux_loss
            self._aux_loss = None
        return loss


class BST(_SeqBase):
    """Behavior Sequence Transformer (reference: modelzoo/bst): transformer
    encoder over [behavior sequence; target item] with learned positions."""

    def __init__(self, embedding_dim=16, item_dim=32, n_heads=4,
                 ff_dim=128, n_layers=1, max_len=65,
                 mlp_sizes=(256, 64), device="cpu", bf16=True, **kw):
        super().__init__(embedding_dim, item_dim, device, bf16, name="bst",
                         **kw)
        self.pos = nn.Parameter(torch.zeros(max_len, item_dim))
        from deeprec_amd.ops.fused_attention import FusedTransformerEncoder
        # per-sample LDS attention + bf16 MFMA projections: torch's
        # nn.TransformerEncoder at these shapes (T~50, d=32) ran fp32
        # launch-bound batched GEMMs at 9.9 ms/step (batch 8192)
        self.encoder = FusedTransformerEncoder(item_dim, n_heads, ff_dim,
                                               n_layers)
        in_dim = NUM_DENSE + self.num_sparse * embedding_dim + item_dim
        self.mlp = make_mlp(list(mlp_sizes) + [1], in_dim, device, self.bf16,
                            final_activation=False)
        self.to(self.device_)

    def forward(self, dense, sparse_ids, seq_ids, target_ids, train=True):
        emb = self.sparse_feats(sparse_ids, train)
        seq = self.seq_emb(seq_ids, train)               # [B,T,Di]
        target = embedding_lookup(self.item_ev, target_ids,
                                  train=train).float()
        x_seq = torch.cat([seq, target.unsqueeze(1)], 1)  # [B,T+1,Di]
        t1 = x_seq.shape[1]
        x_seq = x_seq + self.pos[:t1]
        pad = torch.cat(
            [seq_ids <= 0,
             torch.zeros(seq.shape[0], 1, dtype=torch.bool,
                         device=seq.device)], 1)
        enc = self.encoder(x_seq, pad)
        pooled = enc.mean(1)
        x = torch.cat([dense, emb.flatten(1).float(), pooled], 1)
        with self.amp():
            out = self.mlp(x.to(self.compute_dtype))
        return out.float().squeeze(1)
