"""Fused multi-head attention + transformer encoder layer.

GPU path: the per-sample LDS attention kernels (attention_kernels.hip) —
built for the sequence models' shapes (T ≈ 50-100, d_model ≈ 32-64)
where torch's batched-GEMM attention is launch-bound (BST measured
9.9 ms/step on nn.TransformerEncoder at batch 8192). CPU path: plain
torch attention with identical semantics (the numerics reference).

Capability ≙ the reference transformer block (modelzoo/bst/train.py) and
attention blocks (modelzoo/din/train.py:207-253).
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn

from deeprec_amd.ops.fused_mlp import FusedLinear


class _FusedMHA(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, key_pad_u8, n_heads, scale):
        from deeprec_amd.ops.build_ext import require_extension
        ext = require_extension()
        out, stats = ext.mha_fwd(q, k, v, key_pad_u8, n_heads, scale)
        ctx.ext = ext
        ctx.n_heads = n_heads
        ctx.scale = scale
        ctx.save_for_backward(q, k, v, key_pad_u8, stats)
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, key_pad_u8, stats = ctx.saved_tensors
        dq, dk, dv = ctx.ext.mha_bwd(dout, q, k, v, key_pad_u8, stats,
                                     ctx.n_heads, ctx.scale)
        return dq, dk, dv, None, None, None


def _torch_mha(q, k, v, key_pad, n_heads, scale):
    """CPU/reference attention: softmax(QK^T * scale) V with key padding
    (pad rows attend nothing; padded queries still produce outputs)."""
    b, t, d = q.shape
    dh = d // n_heads
    qs = q.view(b, t, n_heads, dh).transpose(1, 2).float()
    ks = k.view(b, t, n_heads, dh).transpose(1, 2).float()
    vs = v.view(b, t, n_heads, dh).transpose(1, 2).float()
    scores = torch.matmul(qs, ks.transpose(-1, -2)) * scale
    scores = scores.masked_fill(key_pad[:, None, None, :], -1e30)
    p = torch.softmax(scores, dim=-1)
    out = torch.matmul(p, vs)
    return out.transpose(1, 2).reshape(b, t, d).to(q.dtype)


def fused_mha(q, k, v, key_pad: torch.Tensor, n_heads: int,
              scale: float = None):
    """q/k/v [B, T, D]; key_pad [B, T] bool (True = masked key)."""
    d = q.shape[-1]
    scale = scale if scale is not None else 1.0 / math.sqrt(d // n_heads)
    if q.device.type == "cuda":
        return _FusedMHA.apply(q.to(torch.bfloat16).contiguous(),
                               k.to(torch.bfloat16).contiguous(),
                               v.to(torch.bfloat16).contiguous(),
                               key_pad.to(torch.uint8).contiguous(),
                               n_heads, scale)
    return _torch_mha(q, k, v, key_pad, n_heads, scale)


class _DinFeat(torch.autograd.Function):
    @staticmethod
    def forward(ctx, seq, tgt):
        from deeprec_amd.ops.build_ext import require_extension
        ext = require_extension()
        ctx.ext = ext
        ctx.save_for_backward(seq, tgt)
        return ext.din_feat_fwd(seq.contiguous(), tgt.contiguous())

    @staticmethod
    def backward(ctx, g):
        seq, tgt = ctx.saved_tensors
        dseq, dtgt = ctx.ext.din_feat_bwd(g, seq, tgt)
        return dseq, dtgt


def din_att_features(seq: torch.Tensor, tgt: torch.Tensor):
    """[B,T,D] seq + [B,D] target -> bf16 [B*T, 4D] attention features
    [s, t, s-t, s*t] in one fused pass on GPU (torch path elsewhere)."""
    if seq.device.type == "cuda":
        return _DinFeat.apply(seq.float(), tgt.float())
    t = tgt.unsqueeze(1).expand_as(seq)
    return torch.cat([seq, t, seq - t, seq * t],
                     dim=2).reshape(seq.shape[0] * seq.shape[1], -1)


class _ResLN(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x16, a16, gamma, beta, eps):
        from deeprec_amd.ops.build_ext import require_extension
        ext = require_extension()
        y, z, stats = ext.resln_fwd(x16, a16, gamma, beta, eps)
        ctx.ext = ext
        ctx.save_for_backward(z, stats, gamma)
        return y

    @staticmethod
    def backward(ctx, dy):
        z, stats, gamma = ctx.saved_tensors
        dz, dgamma, dbeta = ctx.ext.resln_bwd(dy, z, stats, gamma)
        # the residual add distributes dz to both inputs
        return dz, dz, dgamma, dbeta, None


class FusedResidualLN(nn.Module):
    """y = LayerNorm(x + a) fused into two kernels (torch's LN pipeline
    measured ~930 us/step on [B*T, 32] rows; these are ~30 us). The
    residual stream stays bf16 on GPU; CPU path is the fp32 reference."""

    def __init__(self, dim: int, eps: float = 1e-5):
        super().__init__()
        self.dim = dim
        self.eps = eps
        self.weight = nn.Parameter(torch.ones(dim))
        self.bias = nn.Parameter(torch.zeros(dim))

    def forward(self, x, a):
        if x.device.type == "cuda":
            return _ResLN.apply(x.to(torch.bfloat16).contiguous(),
                                a.to(torch.bfloat16).contiguous(),
                                self.weight, self.bias, self.eps)
        z = x.float() + a.float()
        return nn.functional.layer_norm(z, (self.dim,), self.weight,
                                        self.bias, self.eps)


class FusedTransformerLayer(nn.Module):
    """Post-norm transformer encoder layer (the nn.TransformerEncoderLayer
    default): x = LN(x + MHA(x)); x = LN(x + FFN(x)). QKV/out/FFN run on
    the bf16 MFMA FusedLinear path on GPU; attention on the per-sample
    LDS kernel."""

    def __init__(self, d_model: int, n_heads: int, ff_dim: int):
        super().__init__()
        self.d_model = d_model
        self.n_heads = n_heads
        self.qkv = FusedLinear(d_model, 3 * d_model, activation=None)
        self.out = FusedLinear(d_model, d_model, activation=None)
        self.ff1 = FusedLinear(d_model, ff_dim, activation="relu")
        self.ff2 = FusedLinear(ff_dim, d_model, activation=None)
        self.norm1 = FusedResidualLN(d_model)
        self.norm2 = FusedResidualLN(d_model)

    def forward(self, x: torch.Tensor, key_pad: torch.Tensor):
        b, t, d = x.shape
        qkv = self.qkv(x.reshape(b * t, d)).reshape(b, t, 3 * d)
        q, k, v = qkv.chunk(3, dim=2)
        a = fused_mha(q.contiguous(), k.contiguous(), v.contiguous(),
                      key_pad, self.n_heads)
        a = self.out(a.reshape(b * t, d)).reshape(b, t, d)
        x = self.norm1(x, a)
        f = self.ff2(self.ff1(x.reshape(b * t, d))).reshape(b, t, d)
        x = self.norm2(x, f)
        return x


class FusedTransformerEncoder(nn.Module):
    def __init__(self, d_model: int, n_heads: int, ff_dim: int,
                 n_layers: int):
        super().__init__()
        self.layers = nn.ModuleList(
            FusedTransformerLayer(d_model, n_heads, ff_dim)
            for _ in range(n_layers))

    def forward(self, x, key_pad):
        for layer in self.layers:
            x = layer(x, key_pad)
        return x


class _MaskedSoftmaxPool(torch.autograd.Function):
    @staticmethod
    def forward(ctx, scores, seq, mask_u8):
        from deeprec_amd.ops.build_ext import require_extension
        ext = require_extension()
        out, w = ext.msm_pool_fwd(scores, seq, mask_u8)
        ctx.ext = ext
        ctx.save_for_backward(seq, w, mask_u8)
        return out

    @staticmethod
    def backward(ctx, dout):
        seq, w, mask_u8 = ctx.saved_tensors
        dscores, dseq = ctx.ext.msm_pool_bwd(dout, seq, w, mask_u8)
        return dscores, dseq, None


def masked_softmax_pool(scores: torch.Tensor, seq: torch.Tensor,
                        mask: torch.Tensor) -> torch.Tensor:
    """softmax(scores masked to the valid positions) @ seq, fused.

    scores [B, T] fp32, seq [B, T, D] fp32, mask [B, T] bool (True =
    valid). A fully-masked row pools to ZEROS (masked positions train
    nothing — torch softmax of an all -inf row would give uniform 1/T
    instead). One kernel per direction (attention_kernels.hip) replaces
    the masked_fill / softmax / broadcast-mul / sum chain and its
    backward — the dominant torch-glue cost of the DIN step."""
    if scores.is_cuda:
        return _MaskedSoftmaxPool.apply(
            scores.float().contiguous(), seq.float().contiguous(),
            mask.to(torch.uint8).contiguous())
    mf = mask.to(scores.dtype)
    s = scores.float() - scores.float().amax(dim=1, keepdim=True)
    e = torch.exp(s) * mf
    w = e / (e.sum(dim=1, keepdim=True) + 1e-20)
    return (w.unsqueeze(2) * seq.float()).sum(1)
