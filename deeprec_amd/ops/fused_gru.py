"""FusedGRU / FusedAUGRU — single-kernel recurrence on gfx950.

See ops/hip/gru_kernels.hip. torch-GRU-compatible weights (weight_ih
[3H,D] rows [r;z;n], weight_hh [3H,H], bias_ih, bias_hh). The x-side
projection for all timesteps is one MFMA GEMM; the recurrence is one
kernel; all weight/input grads are three GEMM calls over the stored
per-gate pre-activation grads. AUGRU takes a per-(sample, step) attention
factor a: h' = (1-a) h + a * GRU(h, x) and returns its gradient.
CPU path: reference loop with identical math (the numerics oracle).
"""
from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn


class _FusedGRUFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w_ih, b_ih, w_hh, b_hh, alpha):
        from deeprec_amd.ops.build_ext import require_extension
        ext = require_extension()
        B, T, D = x.shape
        H = w_hh.shape[1]
        x16 = x.reshape(B * T, D).to(torch.bfloat16).contiguous()
        w_ih16 = w_ih.detach().to(torch.bfloat16)
        x3 = ext.linear_fwd(x16, w_ih16, b_ih.detach().float(), 0)
        u16 = w_hh.detach().to(torch.bfloat16).contiguous()
        bh = b_hh.detach().float()
        a = alpha.float().contiguous() if alpha is not None \
            else torch.Tensor()
        h_out, gates = ext.gru_fwd(x3.view(B, T, 3 * H), u16, bh, a, B, T, H)
        ctx.ext = ext
        ctx.dims = (B, T, D, H)
        ctx.x_dtype = x.dtype
        ctx.has_alpha = alpha is not None
        ctx.save_for_backward(x16, w_ih16, u16, bh, h_out, gates,
                              a if alpha is not None else torch.empty(0))
        return h_out

    @staticmethod
    def backward(ctx, dh_out):
        ext = ctx.ext
        B, T, D, H = ctx.dims
        x16, w_ih16, u16, bh, h_out, gates, a = ctx.saved_tensors
        dpre_x, dpre_h, dalpha = ext.gru_bwd(
            dh_out.float(), h_out, gates, u16, bh,
            a if ctx.has_alpha else torch.Tensor(), B, T, H)
        dpx = dpre_x.view(B * T, 3 * H)
        dx = ext.linear_dx(dpx, w_ih16).view(B, T, D).to(ctx.x_dtype)
        dw_ih, db_ih = ext.linear_dw(dpx, x16, True)
        h_prev = torch.cat(
            [torch.zeros(B, 1, H, device=h_out.device), h_out[:, :-1]],
            dim=1).reshape(B * T, H).to(torch.bfloat16).contiguous()
        dw_hh, db_hh = ext.linear_dw(dpre_h.view(B * T, 3 * H), h_prev, True)
        return dx, dw_ih, db_ih, dw_hh, db_hh, \
            (dalpha if ctx.has_alpha else None)


def _cpu_gru(x, w_ih, b_ih, w_hh, b_hh, alpha):
    """fp32 reference loop (torch GRU math + attention gate)."""
    B, T, D = x.shape
    H = w_hh.shape[1]
    h = torch.zeros(B, H, device=x.device, dtype=torch.float32)
    outs = []
    x3 = x.float() @ w_ih.float().t() + b_ih.float()
    for t in range(T):
        hp = h @ w_hh.float().t() + b_hh.float()
        xr, xz, xn = x3[:, t].split(H, dim=1)
        hr, hz, hn = hp.split(H, dim=1)
        r = torch.sigmoid(xr + hr)
        z = torch.sigmoid(xz + hz)
        n = torch.tanh(xn + r * hn)
        h_new = (1 - z) * n + z * h
        if alpha is not None:
            a = alpha[:, t:t + 1].float()
            h = (1 - a) * h + a * h_new
        else:
            h = h_new
        outs.append(h)
    return torch.stack(outs, dim=1)


class FusedGRU(nn.Module):
    """batch_first GRU (optionally attention-gated) with torch-compatible
    parameters. forward(x [B,T,D], alpha=None) -> h_seq [B,T,H]."""

    def __init__(self, input_size: int, hidden_size: int):
        super().__init__()
        assert hidden_size <= 32, "fused GRU supports hidden size <= 32"
        self.input_size = input_size
        self.hidden_size = hidden_size
        H = hidden_size
        self.weight_ih_l0 = nn.Parameter(torch.empty(3 * H, input_size))
        self.weight_hh_l0 = nn.Parameter(torch.empty(3 * H, H))
        self.bias_ih_l0 = nn.Parameter(torch.empty(3 * H))
        self.bias_hh_l0 = nn.Parameter(torch.empty(3 * H))
        stdv = 1.0 / math.sqrt(H)
        for p in self.parameters():
            nn.init.uniform_(p, -stdv, stdv)

    def forward(self, x: torch.Tensor,
                alpha: Optional[torch.Tensor] = None) -> torch.Tensor:
        if x.device.type == "cuda":
            return _FusedGRUFunction.apply(
                x, self.weight_ih_l0, self.bias_ih_l0, self.weight_hh_l0,
                self.bias_hh_l0, alpha)
        return _cpu_gru(x, self.weight_ih_l0, self.bias_ih_l0,
                        self.weight_hh_l0, self.bias_hh_l0, alpha)
