"""Optimizers with EmbeddingVariable branches.

Capability parity with the reference's optimizer surface: every optimizer
handles dense torch parameters AND EmbeddingVariables, applying sparse
updates to only the slots touched this step (reference:
KvResourceSparseApply* in ops/training_ali_ops.cc:110-763; the EV branch
every stock optimizer gained, e.g. python/training/adagrad.py:141-154).

Dense updates delegate to torch.optim (hipBLASLt-backed fused kernels);
sparse EV updates go through the fused HIP applies
(ops/hip/ev_kernels.hip) on GPU or ops/sparse_optim_cpu.py on CPU.
"""
from __future__ import annotations

from typing import Iterable, List, Optional

import torch

from deeprec_amd.embedding.variable import (
    GLOBAL_STEP, EmbeddingVariable)
from deeprec_amd.ops import sparse_optim_cpu as cpu_apply


class Optimizer:
    """Base: dense delegate + sparse EV apply loop."""

    sparse_name: str = None

    def __init__(self, params: Iterable = None,
                 embedding_variables: Optional[List[EmbeddingVariable]] = None,
                 learning_rate: float = 0.01, clip_norm: float = None):
        self.lr = learning_rate
        # per-gradient norm clip (reference: the DIN/DIEN train.py
        # tf.clip_by_norm(grad, 5) wrap around compute_gradients)
        self.clip_norm = clip_norm
        self.evs = list(embedding_variables or [])
        params = [p for p in (params or []) if p.requires_grad]
        self._params = params
        self._dense = self._make_dense(params) if params else None
        # distributed seam: set to DenseGradAllreducer.wait to overlap an
        # async dense-grad all-reduce with the sparse applies
        self.pre_dense_step = None
        # post-step seam (e.g. fused_mlp.enable_weight_cache's refresh)
        self.post_step_hook = None
        self._step_count = 0

    # -- subclass hooks --
    def _make_dense(self, params):
        raise NotImplementedError

    def _sparse_hyper(self, ev) -> dict:
        raise NotImplementedError

    def set_learning_rate(self, lr: float):
        """Update the learning rate in place (LR schedules — reference:
        tf.train.exponential_decay fed into optimizer constructors).
        Takes effect on the NEXT eager step for both sparse applies
        (hyper is rebuilt per step) and the dense optimizer (param
        groups updated). A hipGraph-captured step bakes the lr it was
        captured with — re-capture (or run eager) to change it."""
        self.lr = lr
        if self._dense is not None:
            if hasattr(self._dense, "param_groups"):
                for g in self._dense.param_groups:
                    g["lr"] = lr
            else:  # FlatDenseAdam delegate reads .lr per step
                self._dense.lr = lr

    def zero_grad(self, set_to_none: bool = True):
        if self._dense is not None:
            self._dense.zero_grad(set_to_none=set_to_none)
        for ev in self.evs:
            ev._pending_grads.clear()

    def _apply_sparse(self, ev, slots, grad, hyper, keys=None):
        if hasattr(ev.storage, "apply_split"):  # multi-tier HBM_DRAM[_SSD]
            if keys is not None and hasattr(ev.storage, "set_apply_keys"):
                ev.storage.set_apply_keys(keys)  # SSD tier is key-addressed
            ev.storage.apply_split(self.sparse_name, slots, grad, hyper)
        elif ev.device.type == "cuda":
            from deeprec_amd.ops import hip_backend
            hip_backend.sparse_apply(self.sparse_name, ev.storage, slots,
                                     grad, hyper)
        else:
            getattr(cpu_apply, f"apply_{self.sparse_name}")(
                ev.storage, slots, grad, **hyper)

    def _clip_grads(self):
        """tf.clip_by_norm semantics: each gradient tensor (and each
        sparse grad-rows tensor) independently rescaled to norm <=
        clip_norm."""
        cn = self.clip_norm
        if self._dense is not None and hasattr(self._dense, "_slices"):
            # FlatDenseAdam: dense grads live in the flat buffer, one
            # [dW|db] slice per layer — clip per slice (same
            # per-gradient semantics)
            g = self._dense.g
            for off, sz in self._dense._slices:
                sl = g[off:off + sz]
                n = sl.norm()
                if n > cn:
                    sl.mul_(cn / (n + 1e-12))
        else:
            for p in self._params:
                if p.grad is not None:
                    n = p.grad.norm()
                    if n > cn:
                        p.grad.mul_(cn / (n + 1e-12))
        for ev in self.evs:
            for _, _, grad in ev._pending_grads:
                n = grad.norm()
                if n > cn:
                    grad.mul_(cn / (n + 1e-12))

    def step(self, increment_global_step: bool = True):
        self._step_count += 1
        if self.clip_norm is not None:
            self._clip_grads()
        # sparse applies first: they are independent of the dense grads,
        # so an in-flight async all-reduce overlaps with them
        for ev in self.evs:
            if not ev.trainable:
                ev.consume_grads()
                continue
            hyper = self._sparse_hyper(ev)
            for slots, keys, grad in ev.consume_grads():
                self._apply_sparse(ev, slots, grad, hyper, keys=keys)
        if self.pre_dense_step is not None:
            self.pre_dense_step()
        if self._dense is not None:
            self._dense.step()
        if self.post_step_hook is not None:
            self.post_step_hook()
        self._post_step()
        if increment_global_step:
            GLOBAL_STEP.increment()

    def _post_step(self):
        pass

    # ---- checkpoint integration ----
    def state_dict(self) -> dict:
        return {
            "dense": self._dense.state_dict() if self._dense else None,
            "step_count": self._step_count,
            "beta_powers": getattr(self, "_beta_powers", None),
            "global_step": GLOBAL_STEP.value,
        }

    def load_state_dict(self, sd: dict):
        if sd.get("dense") is not None and self._dense is not None:
            self._dense.load_state_dict(sd["dense"])
        self._step_count = sd.get("step_count", 0)
        if sd.get("beta_powers") is not None:
            self._beta_powers = sd["beta_powers"]
        GLOBAL_STEP.value = sd.get("global_step", GLOBAL_STEP.value)


class GradientDescentOptimizer(Optimizer):
    sparse_name = "sgd"

    def _make_dense(self, params):
        return torch.optim.SGD(params, lr=self.lr)

    def _sparse_hyper(self, ev):
        return dict(lr=self.lr)


class AdagradOptimizer(Optimizer):
    sparse_name = "adagrad"

    def __init__(self, params=None, embedding_variables=None,
                 learning_rate=0.01, initial_accumulator_value=0.1,
                 epsilon=0.0):
        self.initial_accumulator_value = initial_accumulator_value
        self.epsilon = epsilon
        super().__init__(params, embedding_variables, learning_rate)

    def _make_dense(self, params):
        return torch.optim.Adagrad(
            params, lr=self.lr,
            initial_accumulator_value=self.initial_accumulator_value,
            eps=max(self.epsilon, 1e-10))

    def _sparse_hyper(self, ev):
        return dict(lr=self.lr,
                    initial_accumulator=self.initial_accumulator_value,
                    epsilon=self.epsilon)


class AdagradDecayOptimizer(AdagradOptimizer):
    """Reference: python/training/adagrad_decay.py:35."""

    sparse_name = "adagrad_decay"

    def __init__(self, params=None, embedding_variables=None,
                 learning_rate=0.01, initial_accumulator_value=0.1,
                 accumulator_decay_step=100000, accumulator_decay_rate=0.9,
                 accumulator_baseline=0.0, epsilon=0.0):
        self.accumulator_decay_step = accumulator_decay_step
        self.accumulator_decay_rate = accumulator_decay_rate
        self.accumulator_baseline = accumulator_baseline
        super().__init__(params, embedding_variables, learning_rate,
                         initial_accumulator_value, epsilon)

    def _sparse_hyper(self, ev):
        from deeprec_amd.embedding.variable import get_global_step
        return dict(lr=self.lr, global_step=get_global_step(),
                    initial_accumulator=self.initial_accumulator_value,
                    accumulator_decay_step=self.accumulator_decay_step,
                    accumulator_decay_rate=self.accumulator_decay_rate,
                    accumulator_baseline=self.accumulator_baseline,
                    epsilon=self.epsilon)


class AdamOptimizer(Optimizer):
    sparse_name = "adam"

    def __init__(self, params=None, embedding_variables=None,
                 learning_rate=0.001, beta1=0.9, beta2=0.999, epsilon=1e-8):
        self.beta1, self.beta2, self.epsilon = beta1, beta2, epsilon
        super().__init__(params, embedding_variables, learning_rate)

    def _make_dense(self, params):
        return torch.optim.Adam(params, lr=self.lr,
                                betas=(self.beta1, self.beta2),
                                eps=self.epsilon)

    def _sparse_hyper(self, ev):
        return dict(lr=self.lr, step_t=self._step_count, beta1=self.beta1,
                    beta2=self.beta2, epsilon=self.epsilon)


class AdamAsyncOptimizer(AdamOptimizer):
    """Per-variable beta powers, lock-free apply semantics
    (reference: python/training/adam_async.py:40)."""

    sparse_name = "adam_async"

    def __init__(self, params=None, embedding_variables=None,
                 learning_rate=0.001, beta1=0.9, beta2=0.999, epsilon=1e-8,
                 apply_sparse_rmsprop=False, graph_safe=False):
        self.apply_sparse_rmsprop = apply_sparse_rmsprop
        self.graph_safe = graph_safe  # hipGraph capture: powers on device
        self._beta_powers = {}
        super().__init__(params, embedding_variables, learning_rate,
                         beta1, beta2, epsilon)

    def _sparse_hyper(self, ev):
        if self.graph_safe:
            powers = self._beta_powers.get(ev.name)
            if not torch.is_tensor(powers):
                powers = torch.tensor([self.beta1, self.beta2],
                                      device=ev.device)
                self._beta_powers[ev.name] = powers
            return dict(lr=self.lr, powers_dev=powers, beta1=self.beta1,
                        beta2=self.beta2, epsilon=self.epsilon,
                        sparse_rmsprop=self.apply_sparse_rmsprop)
        b1p, b2p = self._beta_powers.setdefault(
            ev.name, [self.beta1, self.beta2])
        return dict(lr=self.lr, beta1_power=b1p, beta2_power=b2p,
                    beta1=self.beta1, beta2=self.beta2, epsilon=self.epsilon,
                    sparse_rmsprop=self.apply_sparse_rmsprop)

    def _make_dense(self, params):
        if self.graph_safe:
            # fused multi-tensor adam: one kernel, capture-safe (the
            # capturable=True eager path cost ~150us/step in elementwise
            # bias-correction kernels)
            try:
                return torch.optim.Adam(params, lr=self.lr,
                                        betas=(self.beta1, self.beta2),
                                        eps=self.epsilon, fused=True,
                                        capturable=True)
            except (RuntimeError, ValueError):
                return torch.optim.Adam(params, lr=self.lr,
                                        betas=(self.beta1, self.beta2),
                                        eps=self.epsilon, capturable=True)
        return torch.optim.Adam(params, lr=self.lr,
                                betas=(self.beta1, self.beta2),
                                eps=self.epsilon)

    def _post_step(self):
        for ev in self.evs:
            p = self._beta_powers.get(ev.name)
            if p is None:
                # NOTE: constructed lazily OUTSIDE any graph capture (the
                # sparse-hyper call during warmup creates it first)
                p = (torch.tensor([self.beta1, self.beta2],
                                  device=ev.device)
                     if self.graph_safe else [self.beta1, self.beta2])
                self._beta_powers[ev.name] = p
            if torch.is_tensor(p):
                from deeprec_amd.ops.build_ext import require_extension
                require_extension().update_powers(p, self.beta1, self.beta2)
            else:
                p[0] *= self.beta1
                p[1] *= self.beta2


class AdamWOptimizer(AdamOptimizer):
    """Reference: python/training/weight_decay_optimizers.py:297."""

    sparse_name = "adamw"

    def __init__(self, params=None, embedding_variables=None,
                 learning_rate=0.001, weight_decay=0.01, beta1=0.9,
                 beta2=0.999, epsilon=1e-8):
        self.weight_decay = weight_decay
        super().__init__(params, embedding_variables, learning_rate,
                         beta1, beta2, epsilon)

    def _make_dense(self, params):
        return torch.optim.AdamW(params, lr=self.lr,
                                 betas=(self.beta1, self.beta2),
                                 eps=self.epsilon,
                                 weight_decay=self.weight_decay)

    def _sparse_hyper(self, ev):
        h = super()._sparse_hyper(ev)
        h["weight_decay"] = self.weight_decay
        return h


class FtrlOptimizer(Optimizer):
    sparse_name = "ftrl"

    def __init__(self, params=None, embedding_variables=None,
                 learning_rate=0.01, learning_rate_power=-0.5,
                 initial_accumulator_value=0.1,
                 l1_regularization_strength=0.0,
                 l2_regularization_strength=0.0,
                 l2_shrinkage_regularization_strength=0.0):
        self.lr_power = learning_rate_power
        self.initial_accumulator_value = initial_accumulator_value
        self.l1 = l1_regularization_strength
        self.l2 = l2_regularization_strength
        self.l2_shrinkage = l2_shrinkage_regularization_strength
        super().__init__(params, embedding_variables, learning_rate)

    def _make_dense(self, params):
        # dense FTRL is rare; SGD fallback keeps dense params training
        return torch.optim.SGD(params, lr=self.lr)

    def _sparse_hyper(self, ev):
        return dict(lr=self.lr, l1=self.l1, l2=self.l2,
                    lr_power=self.lr_power, l2_shrinkage=self.l2_shrinkage)


def make_optimizer(name: str, params=None, embedding_variables=None,
                   learning_rate=0.01, **kw) -> Optimizer:
    table = {
        "sgd": GradientDescentOptimizer,
        "gradientdescent": GradientDescentOptimizer,
        "adagrad": AdagradOptimizer,
        "adagraddecay": AdagradDecayOptimizer,
        "adam": AdamOptimizer,
        "adamasync": AdamAsyncOptimizer,
        "adamw": AdamWOptimizer,
        "ftrl": FtrlOptimizer,
    }
    return table[name.lower().replace("_", "")](
        params, embedding_variables, learning_rate, **kw)
