"""Collective communication helpers.

Single communication stack by design: torch.distributed with the "nccl"
backend, which on ROCm IS RCCL over xGMI (the reference's four coexisting
stacks — gRPC/seastar/NCCL/MPI, SURVEY.md §5 — collapse to this one for
single-node collective training). The gloo backend backs CPU-only tests;
gloo lacks all_to_all, so an equivalent point-to-point emulation keeps one
code path testable without GPUs.
"""
from __future__ import annotations

import torch
import torch.distributed as dist

_ALLTOALL_SUPPORTED = None


def is_initialized() -> bool:
    return dist.is_available() and dist.is_initialized()


def shutdown(after_capture: bool = False):
    """Tear down the process group. after_capture=True when collectives
    were recorded into a hipGraph this process: their host-side work
    bookkeeping never completes, so a clean destroy_process_group waits
    forever — abort the communicator instead (safe at teardown: training
    is done and every rank aborts at the same point)."""
    if not is_initialized():
        return
    if after_capture:
        try:
            from torch.distributed.distributed_c10d import (
                _abort_process_group)
            _abort_process_group()
            return
        except Exception:  # noqa: BLE001 — fall through to bounded destroy
            pass
    import threading
    th = threading.Thread(target=dist.destroy_process_group, daemon=True)
    th.start()
    th.join(timeout=30)
    if th.is_alive():
        import os
        os._exit(0)  # teardown wedged; the job itself already finished


def world_size() -> int:
    return dist.get_world_size() if is_initialized() else 1


def rank() -> int:
    return dist.get_rank() if is_initialized() else 0


def _supports_all_to_all() -> bool:
    global _ALLTOALL_SUPPORTED
    if _ALLTOALL_SUPPORTED is None:
        _ALLTOALL_SUPPORTED = dist.get_backend() != "gloo"
    return _ALLTOALL_SUPPORTED


def all_to_all_single(inp: torch.Tensor, in_splits, out_splits) -> torch.Tensor:
    """Variable-size all-to-all along dim 0. in_splits/out_splits: python
    lists of per-peer element counts (dim-0 rows)."""
    w = world_size()
    if w == 1 and not is_initialized():
        return inp.clone()
    # NOTE: world=1 with an initialized group still goes through the
    # backend (self-exchange) so the RCCL code path is exercised on a
    # single-GPU lease exactly as it will run at world=8
    out_shape = (sum(out_splits),) + tuple(inp.shape[1:])
    out = torch.empty(out_shape, dtype=inp.dtype, device=inp.device)
    if _supports_all_to_all():
        dist.all_to_all_single(out, inp.contiguous(),
                               output_split_sizes=list(out_splits),
                               input_split_sizes=list(in_splits))
        return out
    # gloo emulation: pairwise send/recv
    r = rank()
    in_off = [0]
    for s in in_splits:
        in_off.append(in_off[-1] + s)
    out_off = [0]
    for s in out_splits:
        out_off.append(out_off[-1] + s)
    reqs = []
    for peer in range(w):
        if peer == r:
            out[out_off[r]:out_off[r + 1]] = inp[in_off[r]:in_off[r + 1]]
            continue
        chunk = inp[in_off[peer]:in_off[peer + 1]].contiguous()
        reqs.append(dist.isend(chunk, dst=peer, tag=0))
    for peer in range(w):
        if peer == r:
            continue
        buf = torch.empty((out_splits[peer],) + tuple(inp.shape[1:]),
                          dtype=inp.dtype, device=inp.device)
        dist.recv(buf, src=peer, tag=0)
        out[out_off[peer]:out_off[peer + 1]] = buf
    for q in reqs:
        q.wait()
    return out


def exchange_counts(counts: torch.Tensor) -> torch.Tensor:
    """counts int64[w] (rows destined to each peer) -> int64[w] rows
    arriving from each peer. The two-phase count-then-payload exchange of
    the reference's all-to-all dispatcher."""
    w = world_size()
    if w == 1 and not is_initialized():
        return counts.clone()
    return all_to_all_single(counts.contiguous(), [1] * w, [1] * w)


def padded_all_to_all(inp: torch.Tensor, in_splits, cap: int):
    """Fixed-shape all-to-all: every peer slot padded to `cap` rows.

    The building block for capturing the DISTRIBUTED step in a hipGraph
    (RCCL collectives are capturable, but only with static shapes): both
    the payload (w*cap rows each way) and the true counts (w ints) have
    shapes independent of the data. Rows beyond a slot's true count are
    zeros on the wire and must be ignored via the returned counts.

    Returns (out [w*cap, ...] with peer p's rows at [p*cap, p*cap+n_p),
    out_counts int64 [w]). Raises if any split exceeds cap.
    """
    w = world_size()
    if max(in_splits, default=0) > cap:
        raise ValueError(f"split {max(in_splits)} exceeds pad cap {cap}")
    if w == 1 and not is_initialized():
        out = torch.zeros((cap,) + tuple(inp.shape[1:]), dtype=inp.dtype,
                          device=inp.device)
        out[: inp.shape[0]] = inp
        return out, torch.tensor(list(in_splits), dtype=torch.int64)
    send = torch.zeros((w * cap,) + tuple(inp.shape[1:]),
                       dtype=inp.dtype, device=inp.device)
    off = 0
    for p, n in enumerate(in_splits):
        send[p * cap: p * cap + n] = inp[off: off + n]
        off += n
    # counts ride the collective too, so they must live where the
    # backend wants its tensors (CUDA for nccl/RCCL, CPU for gloo)
    cdev = inp.device if dist.get_backend() != "gloo" else "cpu"
    counts = torch.tensor(list(in_splits), dtype=torch.int64, device=cdev)
    out_counts = exchange_counts(counts)
    out = all_to_all_single(send, [cap] * w, [cap] * w)
    return out, out_counts.cpu()
