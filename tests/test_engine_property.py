"""Randomized model-based test: CpuStorage vs a plain-dict oracle.

Applies a random op sequence (lookup/create, grad-apply, export/import,
shrink, frequency queries) and checks the engine agrees with an obviously
correct dict implementation at every step. The same op semantics are
cross-checked against the HIP engine by tests/test_gpu_engine.py, so this
pins the CPU reference the GPU numerics tests compare against.
"""
import random

import pytest
import torch

from deeprec_amd.embedding.options import (EmbeddingVariableOption,
                                           GlobalStepEvict,
                                           InitializerOption)
from deeprec_amd.ops.cpu_backend import CpuStorage


class DictOracle:
    def __init__(self, dim, default):
        self.dim = dim
        self.default = default
        self.rows = {}
        self.freq = {}
        self.version = {}

    def lookup(self, keys, step):
        out = torch.empty(len(keys), self.dim)
        for i, k in enumerate(keys):
            if k not in self.rows:
                self.rows[k] = self.default[k % self.default.shape[0]] \
                    .clone()
            self.freq[k] = self.freq.get(k, 0) + 1
            self.version[k] = step
            out[i] = self.rows[k]
        return out

    def apply_sgd(self, keys, grads, lr):
        for k, g in zip(keys, grads):
            self.rows[k] -= lr * g

    def evict_older_than(self, min_step):
        dead = [k for k, v in self.version.items() if v < min_step]
        for k in dead:
            del self.rows[k], self.freq[k], self.version[k]
        return len(dead)


@pytest.mark.parametrize("seed", [0, 1, 2])
def test_cpu_engine_matches_dict_oracle(seed):
    rng = random.Random(seed)
    torch.manual_seed(seed)
    dim = 6
    steps_to_live = 12
    opt = EmbeddingVariableOption(
        init_option=InitializerOption(default_value_dim=8),
        evict_option=GlobalStepEvict(steps_to_live=steps_to_live))
    st = CpuStorage(dim, opt)
    oracle = DictOracle(dim, st.default_values)

    for step in range(40):
        keys = [rng.randrange(0, 60) for _ in range(rng.randrange(1, 9))]
        kt = torch.tensor(sorted(set(keys)), dtype=torch.int64)
        counts = torch.ones(kt.numel(), dtype=torch.int64)
        slots = st.lookup_or_create(kt, counts, step=step)
        got = st.gather(kt, slots)
        want = oracle.lookup(kt.tolist(), step)
        torch.testing.assert_close(got, want, rtol=1e-5, atol=1e-6)

        if rng.random() < 0.7:  # train the looked-up keys
            grads = torch.randn(kt.numel(), dim)
            from deeprec_amd.ops.sparse_optim_cpu import apply_sgd
            apply_sgd(st, slots, grads, lr=0.1)
            oracle.apply_sgd(kt.tolist(), grads, 0.1)

        if rng.random() < 0.15:  # eviction by staleness
            n = st.shrink(step)
            n_o = oracle.evict_older_than(step - steps_to_live)
            assert n == n_o

        if rng.random() < 0.1:  # export/import round trip
            k, v, f, ver = st.export()
            order = torch.argsort(k)
            assert sorted(k.tolist()) == sorted(oracle.rows)
            for kk, vv, ff, vver in zip(k.tolist(), v, f.tolist(),
                                        ver.tolist()):
                torch.testing.assert_close(vv, oracle.rows[kk],
                                           rtol=1e-5, atol=1e-6)
                assert ff == oracle.freq[kk]
                assert vver == oracle.version[kk]
            st2 = CpuStorage(dim, opt)
            st2.import_(k, v, f, ver)
            k2, v2, f2, ver2 = st2.export()
            assert sorted(k2.tolist()) == sorted(k.tolist())

    assert st.size() == len(oracle.rows)


class FilteredOracle:
    """Dict oracle for counter-filter admission: a key is admitted (row
    allocated, default-initialized) in the SAME call where its
    accumulated count reaches filter_freq; non-admitted lookups return
    default_value_no_permission (0.0) and never train."""

    def __init__(self, dim, default, filter_freq):
        self.dim = dim
        self.default = default
        self.F = filter_freq
        self.counts = {}
        self.rows = {}

    def lookup(self, keys, counts):
        out = torch.zeros(len(keys), self.dim)
        admitted = []
        for i, (k, c) in enumerate(zip(keys, counts)):
            self.counts[k] = self.counts.get(k, 0) + c
            if self.counts[k] >= self.F:
                if k not in self.rows:
                    self.rows[k] = self.default[
                        k % self.default.shape[0]].clone()
                out[i] = self.rows[k]
                admitted.append(i)
        return out, admitted


@pytest.mark.parametrize("seed", [0, 1, 2])
def test_cpu_engine_counter_filter_matches_oracle(seed):
    """Counter-filter admission under random lookup/train sequences
    (reference semantics: CounterFilterPolicy, counter_filter_policy.h
    — metadata accumulates pre-admission, values only post-admission)."""
    from deeprec_amd.embedding.options import CounterFilter
    from deeprec_amd.ops.sparse_optim_cpu import apply_sgd

    rng = random.Random(100 + seed)
    torch.manual_seed(seed)
    dim, F = 5, 3
    opt = EmbeddingVariableOption(
        init_option=InitializerOption(default_value_dim=8),
        filter_option=CounterFilter(filter_freq=F))
    st = CpuStorage(dim, opt)
    oracle = FilteredOracle(dim, st.default_values, F)

    for step in range(50):
        ks = sorted(set(rng.randrange(0, 40)
                        for _ in range(rng.randrange(1, 7))))
        cs = [rng.randrange(1, 3) for _ in ks]
        kt = torch.tensor(ks, dtype=torch.int64)
        ct = torch.tensor(cs, dtype=torch.int64)
        slots = st.lookup_or_create(kt, ct, step=step)
        got = st.gather(kt, slots)
        want, admitted = oracle.lookup(ks, cs)
        torch.testing.assert_close(got, want, rtol=1e-5, atol=1e-6)
        # engine and oracle agree on WHO is admitted
        assert (slots >= 0).nonzero().squeeze(1).tolist() == admitted

        if rng.random() < 0.7 and admitted:
            grads = torch.randn(kt.numel(), dim)
            apply_sgd(st, slots, grads, lr=0.1)
            for i in admitted:
                oracle.rows[ks[i]] -= 0.1 * grads[i]

    # sub-threshold keys were counted but never materialized
    assert st.size() == len(oracle.rows)
    every = torch.arange(40, dtype=torch.int64)
    freqs = st.frequencies(every)
    for k in range(40):
        assert int(freqs[k]) == oracle.counts.get(k, 0), k
