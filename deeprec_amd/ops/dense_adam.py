"""FlatDenseAdam — fused dense optimizer over ONE flat parameter buffer.

The MI355X replacement for torch's fused Adam in the captured step
(≙ reference dense ApplyAdamAsync, training_ali_ops_gpu.cu.cc:534).
Eliminates the remaining torch glue around the dense update:

- parameters are REBOUND as views of one flat fp32 buffer; each
  FusedLinear's dW/db land directly in the matching slice of one flat
  gradient buffer (linear_dw_out), so there is nothing to flatten,
  unflatten or copy around the data-parallel all-reduce — the reducer
  all-reduces the flat buffer in place and the 1/world averaging folds
  into the Adam update's gradient scale;
- ONE kernel updates w/m/v AND emits the bf16 weight shadows the
  FusedLinear forward consumes (replaces torch fused Adam + the
  multi-tensor bf16 cast);
- bias correction reads device-resident beta powers (capture-safe).

Opt-in: use_flat_dense_adam(opt, model) swaps the torch dense delegate
after construction. Requires every trainable dense parameter to belong
to a FusedLinear that fires at most once per step (true for the zoo's
MLP towers); returns False and leaves the optimizer untouched otherwise.
"""
from __future__ import annotations

from typing import List

import torch

from deeprec_amd.ops.build_ext import require_extension
from deeprec_amd.ops.fused_mlp import FusedLinear


class FlatDenseAdam:
    def __init__(self, layers: List[FusedLinear], lr: float,
                 beta1: float = 0.9, beta2: float = 0.999,
                 epsilon: float = 1e-8):
        assert layers, "no FusedLinear layers"
        self.ext = require_extension()
        self.lr = lr
        self.beta1, self.beta2, self.epsilon = beta1, beta2, epsilon
        self.grad_scale = 1.0  # set to 1/world by the flat allreducer
        dev = layers[0].weight.device
        self.layers = layers
        sizes = []
        for m in layers:
            n, k = m.out_features, m.in_features
            sizes.append(n * k + n)  # [dW | db] — linear_dw_out layout
        total = sum(sizes)
        self.w = torch.empty(total, dtype=torch.float32, device=dev)
        self.g = torch.zeros(total, dtype=torch.float32, device=dev)
        self.m = torch.zeros(total, dtype=torch.float32, device=dev)
        self.v = torch.zeros(total, dtype=torch.float32, device=dev)
        self.w16 = torch.empty(total, dtype=torch.bfloat16, device=dev)
        self.powers = torch.tensor([beta1, beta2], dtype=torch.float32,
                                   device=dev)
        off = 0
        self._slices = []
        with torch.no_grad():
            for mod, sz in zip(layers, sizes):
                n, k = mod.out_features, mod.in_features
                wv = self.w[off: off + n * k].view(n, k)
                bv = self.w[off + n * k: off + sz]
                wv.copy_(mod.weight.data)
                bv.copy_(mod.bias.data)
                mod.weight.data = wv
                mod.bias.data = bv
                mod.w16_cache = self.w16[off: off + n * k].view(n, k)
                mod.dw_ws = self.g[off: off + sz]
                self._slices.append((off, sz))
                off += sz
        self.refresh_shadows()

    def refresh_shadows(self):
        """One cast of the flat buffer — run after any out-of-band weight
        mutation (checkpoint restore, parameter broadcast)."""
        with torch.no_grad():
            self.w16.copy_(self.w)

    def zero_grad(self, set_to_none: bool = True):
        # dW/db kernels zero-then-accumulate their workspace every
        # backward, so per-step zeroing of the flat buffer is not needed;
        # autograd's .grad pointers are cleared so AccumulateGrad assigns
        # (never doubles into) the persistent views
        for m in self.layers:
            m.weight.grad = None
            m.bias.grad = None

    def step(self):
        self.ext.dense_adam(self.w, self.g, self.m, self.v, self.w16,
                            self.powers, self.lr, self.beta1, self.beta2,
                            self.epsilon, self.grad_scale)
        self.ext.update_powers(self.powers, self.beta1, self.beta2)

    # -- checkpoint integration (torch-optimizer-shaped enough for Saver)
    def state_dict(self) -> dict:
        return {"flat": True, "m": self.m.cpu(), "v": self.v.cpu(),
                "powers": self.powers.cpu(), "lr": self.lr}

    def load_state_dict(self, sd: dict):
        with torch.no_grad():
            self.m.copy_(sd["m"].to(self.m.device))
            self.v.copy_(sd["v"].to(self.v.device))
            self.powers.copy_(sd["powers"].to(self.powers.device))


class FlatGradAllreducer:
    """Dense-grad all-reduce over the flat gradient buffer: ONE in-place
    collective, no flatten/unflatten, no div (averaging folds into the
    Adam grad scale)."""

    def __init__(self, adam: FlatDenseAdam):
        from deeprec_amd.parallel import comm
        self.adam = adam
        self.buf = adam.g
        self._pending = None
        if comm.is_initialized():
            adam.grad_scale = 1.0 / comm.world_size()

    def allreduce(self, async_op: bool = False):
        from deeprec_amd.parallel import comm
        if not comm.is_initialized():
            return
        import torch.distributed as dist
        work = dist.all_reduce(self.buf, async_op=async_op)
        if async_op:
            self._pending = work

    def wait(self):
        if self._pending is not None:
            self._pending.wait()
            self._pending = None


def use_flat_dense_adam(opt, model: torch.nn.Module) -> bool:
    """Swap `opt`'s dense delegate for a FlatDenseAdam over `model`'s
    FusedLinear layers. Returns False (no change) unless every trainable
    parameter of `model` belongs to a FusedLinear."""
    layers = [m for m in model.modules() if isinstance(m, FusedLinear)]
    if not layers:
        return False
    owned = set()
    for m in layers:
        owned.add(id(m.weight))
        owned.add(id(m.bias))
    for p in model.parameters():
        if p.requires_grad and id(p) not in owned:
            return False
    if any(p.device.type != "cuda" or p.dtype != torch.float32
           for m in layers for p in (m.weight, m.bias)):
        return False
    opt._dense = FlatDenseAdam(layers, lr=opt.lr,
                               beta1=getattr(opt, "beta1", 0.9),
                               beta2=getattr(opt, "beta2", 0.999),
                               epsilon=getattr(opt, "epsilon", 1e-8))
    return True
