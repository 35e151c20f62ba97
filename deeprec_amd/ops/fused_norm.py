"""Fused normalization ops.

fused_l2_normalize — row-wise y = x / sqrt(max(sum(x^2), eps)) with a
single-wave HIP kernel per row on GPU (reference capability:
FusedL2Normalize[Grad], kernels/fused_l2_normalize/fused_l2_normalize_op.cc
— CPU AVX there). CPU path is the plain-torch formula (numerics oracle).

fused_layer_norm — torch's native layer_norm (ROCm composite is already a
single fused kernel on gfx950); exported for API parity with the
reference's fused_layer_normalize_ops.cc.
"""
from __future__ import annotations

import torch


class _FusedL2Normalize(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, eps):
        from deeprec_amd.ops.build_ext import require_extension
        ext = require_extension()
        shp = x.shape
        x2 = x.reshape(-1, shp[-1]).to(torch.bfloat16).contiguous()
        y, inv = ext.l2norm_fwd(x2, eps)
        ctx.ext = ext
        ctx.shp = shp
        ctx.in_dtype = x.dtype
        ctx.save_for_backward(y, inv)
        return y.reshape(shp)

    @staticmethod
    def backward(ctx, dy):
        y, inv = ctx.saved_tensors
        dx = ctx.ext.l2norm_bwd(
            dy.reshape(y.shape).to(torch.bfloat16), y, inv)
        return dx.reshape(ctx.shp).to(ctx.in_dtype), None


def fused_l2_normalize(x: torch.Tensor, epsilon: float = 1e-12,
                       axis: int = -1) -> torch.Tensor:
    """L2-normalize along the last axis (axis kept for API parity)."""
    assert axis in (-1, x.dim() - 1), "fused path normalizes the last axis"
    if x.device.type == "cuda":
        return _FusedL2Normalize.apply(x, epsilon)
    denom = x.float().pow(2).sum(-1, keepdim=True).clamp_min(epsilon).sqrt()
    return (x.float() / denom).to(x.dtype)


def fused_layer_norm(x: torch.Tensor, weight: torch.Tensor,
                     bias: torch.Tensor, epsilon: float = 1e-5):
    return torch.nn.functional.layer_norm(
        x, x.shape[-1:], weight, bias, epsilon)
