"""OCP fp8 (e4m3fn) inference path.

Capability parity with the reference's post-training low-precision tool
(tools/low_precision_optimize/ — int8/fp16 quantization of a trained
model for serving); the MI355X-native form is OCP fp8: CDNA4 multiplies
non-scaled e4m3 at the full bf16 MFMA rate, so an fp8 serving MLP halves
weight/activation bytes and L2 footprint at zero math-rate cost
(ops/hip/fp8_kernels.hip). Training stays bf16-with-fp32-masters; this
is a serving/inference conversion, applied AFTER training like the
reference tool.

Scheme: per-row dynamic scales (amax/448):
  weights     quantized once per output channel at conversion;
  activations quantized per batch row on the fly;
  y = sa[m] * sw[n] * (qA . qW) + bias, fp32 accumulation, bf16 out.

CPU fallback emulates the exact same arithmetic with
``torch.float8_e4m3fn`` so numerics tests run without a GPU.

Usage:
    from deeprec_amd.ops.fp8 import convert_mlp_to_fp8
    n = convert_mlp_to_fp8(model)         # swaps nn.Linear -> Fp8Linear
    ev_q, ev_scale = quantize_fp8_rows(ev.gather(keys))  # fp8 gather out
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn

_E4M3_MAX = 448.0

_ACT = {None: 0, "none": 0, "relu": 1, "sigmoid": 2}


def _quant_rows_torch(x: torch.Tensor):
    """Reference implementation of the row quantizer (also CPU path)."""
    xf = x.float()
    amax = xf.abs().amax(dim=1, keepdim=True)
    scale = torch.where(amax > 0, amax / _E4M3_MAX,
                        torch.ones_like(amax))
    q = (xf / scale).clamp(-_E4M3_MAX, _E4M3_MAX).to(torch.float8_e4m3fn)
    return q.view(torch.uint8), scale.squeeze(1)


def quantize_fp8_rows(x: torch.Tensor):
    """[n, d] fp32/bf16 -> (uint8 e4m3 [n, d], fp32 scales [n]).

    On GPU this is ONE kernel (amax reduce + convert per row); rows
    dequantize as ``q.view(torch.float8_e4m3fn).float() * scale[:,None]``.
    Use on EV gather output for fp8 serving payloads (halves the bytes of
    a bf16 row with ~2 decimal digits of precision).
    """
    if x.is_cuda:
        from deeprec_amd.ops.build_ext import require_extension
        return require_extension().quant_rows_e4m3(x.contiguous())
    return _quant_rows_torch(x)


def dequantize_fp8_rows(q: torch.Tensor, scale: torch.Tensor,
                        dtype=torch.float32):
    return (q.view(torch.float8_e4m3fn).to(torch.float32)
            * scale.unsqueeze(1)).to(dtype)


class Fp8Linear(nn.Module):
    """Inference-only linear with e4m3 weights + per-channel scales.

    Built from a trained fp32/bf16 ``nn.Linear`` (or raw weights) via
    :meth:`from_linear`. Forward quantizes the activation rows
    dynamically and runs the fp8 MFMA kernel; the CPU path emulates the
    identical arithmetic (same quantizer, fp32 accumulation).
    """

    def __init__(self, qw: torch.Tensor, sw: torch.Tensor,
                 bias: Optional[torch.Tensor], act: Optional[str] = None):
        super().__init__()
        self.register_buffer("qw", qw)          # uint8 [N, K]
        self.register_buffer("sw", sw)          # fp32 [N]
        self.register_buffer("bias", None if bias is None
                             else bias.detach().float())
        self.act = act
        self.out_features, self.in_features = qw.shape

    @classmethod
    def from_weight(cls, weight: torch.Tensor,
                    bias: Optional[torch.Tensor],
                    act: Optional[str] = None):
        with torch.no_grad():
            qw, sw = quantize_fp8_rows(weight.detach())
        return cls(qw, sw, bias, act)

    @classmethod
    def from_linear(cls, lin: nn.Linear, act: Optional[str] = None):
        return cls.from_weight(lin.weight, lin.bias, act)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        shape = x.shape[:-1]
        x2 = x.reshape(-1, self.in_features)
        if x.is_cuda:
            from deeprec_amd.ops.build_ext import require_extension
            ext = require_extension()
            qx, sx = ext.quant_rows_e4m3(
                x2.contiguous() if x2.dtype in (torch.float32,
                                                torch.bfloat16)
                else x2.float().contiguous())
            y = ext.linear_fwd_fp8(qx, sx, self.qw, self.sw, self.bias,
                                   _ACT[self.act])
        else:
            qx, sx = _quant_rows_torch(x2)
            a = qx.view(torch.float8_e4m3fn).float() * sx.unsqueeze(1)
            w = (self.qw.view(torch.float8_e4m3fn).float()
                 * self.sw.unsqueeze(1))
            y = a @ w.t()
            if self.bias is not None:
                y = y + self.bias
            if self.act == "relu":
                y = torch.relu(y)
            elif self.act == "sigmoid":
                y = torch.sigmoid(y)
            y = y.to(torch.bfloat16)
        y = y.reshape(*shape, self.out_features)
        # keep the caller's compute dtype (fp32 debug/serving models feed
        # plain torch ops downstream)
        return y if y.dtype == x.dtype else y.to(x.dtype)

    def extra_repr(self):
        return (f"in={self.in_features}, out={self.out_features}, "
                f"e4m3 per-channel")


class Fp8MlpConverter:
    """Reversible fp8 conversion for serving hot-updates: a full model
    update restores fp32 weights into the ORIGINAL linear modules, so
    the Predictor reverts before a restore and re-quantizes after
    (reference flow: FullModelUpdate rebuilds the serving session;
    quantization there is a separate offline tool — here it is part of
    the online-update cycle)."""

    def __init__(self, module: nn.Module):
        self.module = module
        self._sites = []  # (parent, attr_name, original_module)

    def convert(self) -> int:
        from deeprec_amd.ops.fused_mlp import FusedLinear
        assert not self._sites, "already converted — revert() first"

        def walk(mod):
            n = 0
            for name, child in list(mod.named_children()):
                if isinstance(child, nn.Linear):
                    repl = Fp8Linear.from_linear(child)
                elif isinstance(child, FusedLinear):
                    repl = Fp8Linear.from_weight(child.weight, child.bias,
                                                 child.activation)
                else:
                    n += walk(child)
                    continue
                self._sites.append((mod, name, child))
                setattr(mod, name, repl)
                n += 1
            return n

        return walk(self.module)

    def revert(self):
        for parent, name, orig in reversed(self._sites):
            setattr(parent, name, orig)
        self._sites.clear()


def convert_mlp_to_fp8(module: nn.Module) -> int:
    """Swap every ``nn.Linear`` / ``FusedLinear`` under ``module`` for an
    :class:`Fp8Linear` (post-training, inference-only — the activation
    fused into a FusedLinear carries over). Returns the number
    converted."""
    from deeprec_amd.ops.fused_mlp import FusedLinear
    n = 0
    for name, child in list(module.named_children()):
        if isinstance(child, nn.Linear):
            setattr(module, name, Fp8Linear.from_linear(child))
            n += 1
        elif isinstance(child, FusedLinear):
            setattr(module, name,
                    Fp8Linear.from_weight(child.weight, child.bias,
                                          child.activation))
            n += 1
        else:
            n += convert_mlp_to_fp8(child)
    return n
