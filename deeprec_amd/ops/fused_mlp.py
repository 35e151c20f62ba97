"""FusedMLP — bf16 MFMA linear layers with fp32 master weights.

GPU path uses the hand-written gfx950 kernels (ops/hip/dense_kernels.hip);
CPU path falls back to torch (the numerics reference). This mirrors the
reference's keep_weights bf16 scope (fp32 master weights, bf16 compute —
reference: python/ops/variable_scope.py:3008) with the cast fused into the
layer instead of graph-level cast nodes.
"""
from __future__ import annotations

import math
from typing import List, Optional

import torch
import torch.nn as nn

_ACT_ID = {None: 0, "none": 0, "relu": 1, "sigmoid": 2}


class _FusedLinear(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, act_id, w16_cache=None, dw_ws=None):
        from deeprec_amd.ops.build_ext import require_extension
        ext = require_extension()
        # w16_cache: bf16 shadow refreshed once per optimizer step by
        # enable_weight_cache() — replaces a per-layer cast kernel
        w16 = (w16_cache if w16_cache is not None
               else weight.detach().to(torch.bfloat16))
        x16 = x.to(torch.bfloat16).contiguous()
        out = ext.linear_fwd(x16, w16, bias.detach().float(), act_id)
        ctx.ext = ext
        ctx.act_id = act_id
        ctx.x_dtype = x.dtype
        ctx.dw_ws = dw_ws
        ctx.save_for_backward(x16, w16, out)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        x16, w16, out = ctx.saved_tensors
        ext = ctx.ext
        g = grad_out.to(torch.bfloat16).contiguous()
        if ctx.act_id != 0:
            g = ext.act_bwd(g, out, ctx.act_id)
        dx = ext.linear_dx(g, w16)
        # direct col-fragment dW: measured faster end-to-end than the
        # transpose-then-row-load variant (torch .t().contiguous() costs
        # ~12us/copy, more than the strided-fragment penalty it removes)
        if ctx.dw_ws is not None:
            # FlatDenseAdam mode: dW/db land in the optimizer's flat
            # gradient buffer (zero-filled + accumulated by the kernel)
            dw, db = ext.linear_dw_out(g, x16, ctx.dw_ws, True)
        else:
            dw, db = ext.linear_dw(g, x16, True)
        return dx.to(ctx.x_dtype), dw, db, None, None, None


class FusedLinear(nn.Module):
    """Linear(+activation) with fp32 master weight, bf16 MFMA compute on
    GPU. activation: None | 'relu' | 'sigmoid'. 3D inputs are flattened
    over the leading dims."""

    def __init__(self, in_features: int, out_features: int,
                 relu: bool = True, activation: Optional[str] = "unset"):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.activation = ("relu" if relu else None) \
            if activation == "unset" else activation
        self.weight = nn.Parameter(torch.empty(out_features, in_features))
        self.bias = nn.Parameter(torch.zeros(out_features))
        # torch Linear default init (kaiming uniform)
        nn.init.kaiming_uniform_(self.weight, a=math.sqrt(5))
        bound = 1.0 / math.sqrt(in_features)
        nn.init.uniform_(self.bias, -bound, bound)
        self.w16_cache = None  # set by enable_weight_cache()
        self.dw_ws = None      # set by FlatDenseAdam (flat grad slice)

    def forward(self, x):
        lead = x.shape[:-1]
        flat = x.reshape(-1, x.shape[-1])
        if x.device.type == "cuda" and flat.shape[0] % 16 == 0:
            out = _FusedLinear.apply(flat, self.weight, self.bias,
                                     _ACT_ID[self.activation],
                                     self.w16_cache, self.dw_ws)
        else:
            out = nn.functional.linear(flat, self.weight.to(flat.dtype),
                                       self.bias.to(flat.dtype))
            if self.activation == "relu":
                out = nn.functional.relu(out)
            elif self.activation == "sigmoid":
                out = torch.sigmoid(out)
        return out.reshape(*lead, self.out_features)


class _DotInteraction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, feats, p_pad):
        from deeprec_amd.ops.build_ext import require_extension
        ext = require_extension()
        f16 = feats.to(torch.bfloat16).contiguous()
        out = ext.interact_fwd(f16, p_pad)
        ctx.ext = ext
        ctx.f_dtype = feats.dtype
        ctx.save_for_backward(f16)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        (f16,) = ctx.saved_tensors
        df = ctx.ext.interact_bwd(grad_out.to(torch.bfloat16), f16)
        return df.to(ctx.f_dtype), None


def dot_interaction(feats: torch.Tensor, p_pad: int = None) -> torch.Tensor:
    """feats [B, F, D] -> pairwise dots [B, p_pad] (i<j upper triangle,
    zero-padded). Fused gfx950 kernel on GPU; torch fallback elsewhere."""
    b, f, d = feats.shape
    p = f * (f - 1) // 2
    p_pad = p_pad or p
    if feats.device.type == "cuda" and (f * d) % 8 == 0:
        return _DotInteraction.apply(feats, p_pad)
    z = torch.bmm(feats.float(), feats.float().transpose(1, 2))
    iu = torch.triu_indices(f, f, offset=1, device=feats.device)
    out = z[:, iu[0], iu[1]]
    if p_pad > p:
        out = torch.nn.functional.pad(out, (0, p_pad - p))
    return out.to(feats.dtype)


class _DotInteractionCat(torch.autograd.Function):
    @staticmethod
    def forward(ctx, bot, emb, p_pad):
        from deeprec_amd.ops.build_ext import require_extension
        ext = require_extension()
        b16 = bot.to(torch.bfloat16).contiguous()
        e16 = emb.to(torch.bfloat16).contiguous()
        out = ext.interact_cat_fwd(b16, e16, p_pad)
        ctx.ext = ext
        ctx.dtypes = (bot.dtype, emb.dtype)
        ctx.save_for_backward(b16, e16)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        b16, e16 = ctx.saved_tensors
        dbot, demb = ctx.ext.interact_cat_bwd(
            grad_out.to(torch.bfloat16), b16, e16)
        bd, ed = ctx.dtypes
        return dbot.to(bd), demb.to(ed), None


def dot_interaction_cat(bot: torch.Tensor, emb: torch.Tensor,
                        p_pad: int) -> torch.Tensor:
    """bot [B,D] + emb [B,F,D] -> [B, D + p_pad] = [bot | pairwise dots
    over [bot; emb] rows, i<j upper triangle, zero-padded] in ONE fused
    kernel per direction — removes the feats-assembly cat AND the
    top-MLP input cat. Torch fallback composes the same math."""
    if bot.device.type == "cuda" and bot.shape[1] % 8 == 0:
        out = _DotInteractionCat.apply(bot, emb, p_pad)
        # keep the caller's compute dtype (the fp32 debug path feeds
        # plain nn.Linear layers)
        return out if out.dtype == bot.dtype else out.to(bot.dtype)
    feats = torch.cat([bot.unsqueeze(1), emb.to(bot.dtype)], dim=1)
    return torch.cat([bot, dot_interaction(feats, p_pad)], dim=1)


def fused_mlp(sizes: List[int], in_dim: int,
              final_activation: bool = True) -> nn.Sequential:
    layers = []
    d = in_dim
    for i, h in enumerate(sizes):
        layers.append(FusedLinear(d, h,
                                  relu=final_activation or i + 1 < len(sizes)))
        d = h
    return nn.Sequential(*layers)


def enable_weight_cache(model: nn.Module):
    """Give every FusedLinear a bf16 weight shadow and return a refresh()
    callable (ONE multi-tensor cast) to run after each optimizer step —
    set it as the optimizer's post_step_hook. Replaces the per-layer
    fp32->bf16 cast kernels (~4.6 us each) in the hot loop. Call refresh()
    manually after any out-of-band weight mutation (checkpoint restore,
    broadcast)."""
    mods = [m for m in model.modules() if isinstance(m, FusedLinear)]
    if not mods:
        return None
    shadows, masters = [], []
    for m in mods:
        m.w16_cache = m.weight.detach().to(torch.bfloat16).contiguous()
        shadows.append(m.w16_cache)
        masters.append(m.weight.detach())

    def refresh():
        torch._foreach_copy_(shadows, masters)

    return refresh
