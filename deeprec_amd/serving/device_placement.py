"""Serving-time device placement: embeddings on CPU, dense on GPU.

Capability parity with the reference's device-placement optimization
(docs/docs_en/Device-Placement.md: the serving pass auto-places the
embedding-layer subgraph on CPU so per-request H2D/D2H of the sparse
weights disappears and GPU memory holds only the dense model). Here the
"pass" is a storage swap: move_embeddings_to_cpu() exports each EV's
rows into a CPU storage wrapped by an adapter that converts tensors at
the boundary (GPU ids in, GPU rows out); the pooled path then runs the
generic torch gather+pool. Inference-only (training lookups raise).
"""
from __future__ import annotations

import torch


class CpuOffloadStorage:
    """CPU-resident rows behind a CUDA model: lookups land on the inner
    CpuStorage; gathered rows return on `out_device`."""

    def __init__(self, inner, out_device):
        self.inner = inner
        self.out_device = torch.device(out_device)
        self.dim = inner.dim
        self.key_bits = inner.key_bits
        self.dvd_per_table = inner.dvd_per_table
        self.slabs = {}
        self._last_rank = None

    def lookup(self, keys):
        return self.inner.lookup(keys.cpu()).to(keys.device)

    def lookup_or_create(self, keys, counts, step, train=True):
        if train:
            raise RuntimeError("CpuOffloadStorage is serving-only")
        return self.lookup(keys)

    def gather(self, keys, slots, out_dtype=None):
        rows = self.inner.gather(keys.cpu(), slots.cpu(), out_dtype)
        return rows.to(self.out_device)

    def prefers_dedup(self):
        return False

    def observe_uniq_ratio(self, m, nnz):
        pass

    def frequencies(self, keys):
        return self.inner.frequencies(keys.cpu()).to(keys.device)

    def versions(self, keys):
        return self.inner.versions(keys.cpu()).to(keys.device)

    def size(self):
        return self.inner.size()

    def total_count(self):
        return self.inner.total_count()

    def export(self, include_filtered=False):
        return self.inner.export(include_filtered)

    def import_(self, *a, **kw):
        return self.inner.import_(*a, **kw)

    def memory_usage(self):
        out = self.inner.memory_usage()
        out["cpu_offloaded"] = True
        return out


def move_embeddings_to_cpu(model) -> int:
    """Swap every EV/collection storage for a CPU-resident copy (GPU
    memory freed for the dense model; rows stream per request). Returns
    the number of rows moved."""
    from deeprec_amd.ops.cpu_backend import CpuStorage
    moved = 0
    for ev in model.embedding_variables():
        base = getattr(ev, "local", ev)
        st = base.storage
        keys, values, freqs, versions = st.export()
        inner = CpuStorage(base.dim, st.ev_option)
        inner.key_bits = st.key_bits
        inner.dvd_per_table = st.dvd_per_table
        inner.default_values = st.default_values.cpu()
        inner.import_(keys.cpu(), values.cpu(),
                      freqs.cpu().to(torch.int32), versions.cpu())
        base.storage = CpuOffloadStorage(inner, base.device)
        moved += int(keys.numel())
    return moved
