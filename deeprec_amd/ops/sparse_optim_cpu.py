"""CPU / fp32-reference sparse optimizer applies on EV slots.

These are the semantics the fused HIP kernels
(ops/hip/ev_kernels.hip: sparse_apply_* ) must reproduce; GPU numerics
tests compare against this module.

Reference kernels: kernels/training_ali_ops.cc (CPU) and
training_ali_ops_gpu.cu.cc (kv_sparse_apply_{adagrad,ftrl,adam_async}).
All applies take the slot indices from the step's single hash probe and
skip un-admitted keys (slot < 0).
"""
from __future__ import annotations

import math

import torch


def _adm(storage, slots, grad):
    mask = slots >= 0
    return slots[mask].long(), grad[mask].float()


def apply_sgd(storage, slots, grad, lr, **_):
    s, g = _adm(storage, slots, grad)
    storage.values[s] -= lr * g


def apply_adagrad(storage, slots, grad, lr, initial_accumulator=0.1,
                  epsilon=0.0, **_):
    s, g = _adm(storage, slots, grad)
    accum = storage.get_slab("adagrad_accum", storage.dim, initial_accumulator)
    a = accum[s] + g * g
    accum[s] = a
    storage.values[s] -= lr * g / (a.sqrt() + epsilon)


def apply_adagrad_decay(storage, slots, grad, lr, global_step,
                        initial_accumulator=0.1,
                        accumulator_decay_step=100000,
                        accumulator_decay_rate=0.9,
                        accumulator_baseline=0.0, epsilon=0.0, **_):
    """Periodically-decayed Adagrad (reference: AdagradDecayOptimizer,
    python/training/adagrad_decay.py): every accumulator_decay_step global
    steps the accumulator is multiplied by accumulator_decay_rate, floored
    at accumulator_baseline. Per-key decay bookkeeping lives in a 1-wide
    slab so sparsely-touched keys decay by the right number of periods."""
    s, g = _adm(storage, slots, grad)
    accum = storage.get_slab("adagrad_accum", storage.dim, initial_accumulator)
    period_slab = storage.get_slab("adagrad_decay_period", 1, 0.0)
    cur_period = float(global_step // max(1, accumulator_decay_step))
    dp = (cur_period - period_slab[s, 0]).clamp(min=0)
    decay = torch.pow(torch.tensor(accumulator_decay_rate), dp).unsqueeze(1)
    a = (accum[s] * decay).clamp(min=accumulator_baseline) + g * g
    accum[s] = a
    period_slab[s, 0] = cur_period
    storage.values[s] -= lr * g / (a.sqrt() + epsilon)


def apply_adam(storage, slots, grad, lr, step_t, beta1=0.9, beta2=0.999,
               epsilon=1e-8, **_):
    """step_t: 1-based global apply count for bias correction."""
    s, g = _adm(storage, slots, grad)
    m = storage.get_slab("adam_m", storage.dim, 0.0)
    v = storage.get_slab("adam_v", storage.dim, 0.0)
    mn = beta1 * m[s] + (1 - beta1) * g
    vn = beta2 * v[s] + (1 - beta2) * g * g
    m[s], v[s] = mn, vn
    lr_t = lr * math.sqrt(1 - beta2 ** step_t) / (1 - beta1 ** step_t)
    storage.values[s] -= lr_t * mn / (vn.sqrt() + epsilon)


def apply_adam_async(storage, slots, grad, lr, beta1_power, beta2_power,
                     beta1=0.9, beta2=0.999, epsilon=1e-8,
                     sparse_rmsprop=False, **_):
    """Lock-free async Adam (reference: AdamAsyncOptimizer,
    python/training/adam_async.py:40 and
    training_ali_ops_gpu.cu.cc:KvSparseApplyAdamAsyncKernel). beta powers
    are per-variable scalars maintained by the caller. With sparse_rmsprop
    the m accumulator is skipped and v behaves like RMSProp."""
    s, g = _adm(storage, slots, grad)
    v = storage.get_slab("adam_v", storage.dim, 0.0)
    if sparse_rmsprop:
        vn = beta2 * v[s] + (1 - beta2) * g * g
        v[s] = vn
        storage.values[s] -= lr * g / (vn.sqrt() + epsilon)
        return
    m = storage.get_slab("adam_m", storage.dim, 0.0)
    mn = beta1 * m[s] + (1 - beta1) * g
    vn = beta2 * v[s] + (1 - beta2) * g * g
    m[s], v[s] = mn, vn
    lr_t = lr * math.sqrt(1 - beta2_power) / (1 - beta1_power)
    storage.values[s] -= lr_t * mn / (vn.sqrt() + epsilon)


def apply_adamw(storage, slots, grad, lr, step_t, weight_decay=0.01,
                beta1=0.9, beta2=0.999, epsilon=1e-8, **_):
    s, g = _adm(storage, slots, grad)
    m = storage.get_slab("adam_m", storage.dim, 0.0)
    v = storage.get_slab("adam_v", storage.dim, 0.0)
    mn = beta1 * m[s] + (1 - beta1) * g
    vn = beta2 * v[s] + (1 - beta2) * g * g
    m[s], v[s] = mn, vn
    lr_t = lr * math.sqrt(1 - beta2 ** step_t) / (1 - beta1 ** step_t)
    w = storage.values[s]
    storage.values[s] = w - lr_t * mn / (vn.sqrt() + epsilon) - lr * weight_decay * w


def apply_ftrl(storage, slots, grad, lr, l1=0.0, l2=0.0,
               lr_power=-0.5, l2_shrinkage=0.0, **_):
    """FTRL-proximal (reference: KvSparseApplyFtrlOp,
    training_ali_ops.cc:431); l2_shrinkage != 0 is the FtrlV2 variant."""
    s, g = _adm(storage, slots, grad)
    n = storage.get_slab("ftrl_accum", storage.dim, 0.1)
    z = storage.get_slab("ftrl_linear", storage.dim, 0.0)
    w = storage.values[s].float()
    n_old = n[s]
    n_new = n_old + g * g
    sigma = (n_new.pow(-lr_power) - n_old.pow(-lr_power)) / lr
    z_new = z[s] + (g + 2.0 * l2_shrinkage * w) - sigma * w
    n[s], z[s] = n_new, z_new
    quad = n_new.pow(-lr_power) / lr + 2.0 * l2
    w_new = torch.where(
        z_new.abs() > l1,
        (torch.sign(z_new) * l1 - z_new) / quad,
        torch.zeros_like(w))
    storage.values[s] = w_new.to(storage.values.dtype)
