"""Single-node collective training: one process per GPU over RCCL/xGMI.

Capability parity with the reference's CollectiveStrategy
(python/distribute/group_embedding_collective_strategy.py:29): dense
gradients ride a bucketed all-reduce; embedding shards ride all-to-all
(parallel/sharded_embedding.py); a launcher spawns one process per GPU
(reference: python/distribute/launch.py:55-97 — here torch.distributed.run
fills that role, see bench.py).
"""
from __future__ import annotations

import os
from typing import Iterable, List

import torch
import torch.distributed as dist

from deeprec_amd.parallel import comm  # noqa: F401
from deeprec_amd.parallel.comm import is_initialized, rank, world_size  # noqa: F401
from deeprec_amd.parallel.sharded_embedding import (  # noqa: F401
    ShardedEmbeddingVariable, sharded_embedding_lookup_sparse)
from deeprec_amd.parallel.sharded_collection import (  # noqa: F401
    ShardedEmbeddingCollection)


def init_distributed(backend: str = None, device=None):
    """Initialize torch.distributed from torchrun env vars."""
    if is_initialized():
        return
    ws = int(os.environ.get("WORLD_SIZE", "1"))
    if ws <= 1:
        return
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    dist.init_process_group(backend=backend)
    if torch.cuda.is_available():
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))


class DenseGradAllreducer:
    """Bucketed dense-gradient all-reduce.

    xGMI sizing note: ring all-reduce is per-link bound (~153 GB/s), and
    DLRM-class dense MLPs are only a few MB — one flat bucket per step is
    the right shape; bucketing exists for larger dense towers. Buckets are
    averaged (sum / world_size).
    """

    def __init__(self, params: Iterable[torch.Tensor],
                 bucket_cap_mb: float = 25.0):
        self.params: List[torch.Tensor] = [
            p for p in params if p.requires_grad]
        self.bucket_cap = int(bucket_cap_mb * (1 << 20))
        self._pending = []

    def allreduce(self, async_op: bool = False):
        """Reduce all dense grads. With async_op=True the collectives are
        launched without waiting — call wait() (or set it as the
        optimizer's pre_dense_step) before the dense optimizer step, so
        the all-reduce overlaps with the sparse applies."""
        if not is_initialized():
            return
        # world=1 still reduces: a single-GPU lease then exercises the
        # exact RCCL call pattern (incl. under hipGraph capture) that a
        # full node will run
        w = world_size()
        bucket, nbytes = [], 0
        for p in self.params:
            if p.grad is None:
                continue
            bucket.append(p.grad)
            nbytes += p.grad.numel() * p.grad.element_size()
            if nbytes >= self.bucket_cap:
                self._reduce_bucket(bucket, w, async_op)
                bucket, nbytes = [], 0
        if bucket:
            self._reduce_bucket(bucket, w, async_op)

    def wait(self):
        for handle, flat, grads, w in self._pending:
            if handle is not None:
                handle.wait()
            flat.div_(w)
            for g, r in zip(
                    grads, torch._utils._unflatten_dense_tensors(flat,
                                                                 grads)):
                g.copy_(r)
        self._pending = []

    def _reduce_bucket(self, grads, w, async_op=False):
        flat = torch._utils._flatten_dense_tensors(grads)
        handle = dist.all_reduce(flat, async_op=async_op)
        if async_op:
            self._pending.append((handle, flat, grads, w))
            return
        flat.div_(w)
        for g, r in zip(grads,
                        torch._utils._unflatten_dense_tensors(flat, grads)):
            g.copy_(r)


def broadcast_parameters(params: Iterable[torch.Tensor], src: int = 0):
    """Sync initial dense weights (reference: hvd broadcast hook,
    hvd_strategy.py:472)."""
    if not is_initialized():
        return
    for p in params:
        dist.broadcast(p.detach(), src=src)
