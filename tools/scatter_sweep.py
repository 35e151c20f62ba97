"""SPLITS/CHUNK sweep for the pooled-grad scatter kernel.

Builds the exact DLRM matrix-path workload (zipf ids, identity rows) and
times group_pooled_bwd_strided at each split setting, plus the csr_order
build. Run on a GPU box: python tools/scatter_sweep.py
"""
import torch

from deeprec_amd.data.synthetic import CriteoSyntheticDataset
from deeprec_amd.ops.build_ext import require_extension


def time_fn(fn, iters=100):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1000.0  # us


def main():
    ext = require_extension()
    dev = torch.device("cuda")
    batch, n_tables, dim = 8192, 26, 16
    ds = CriteoSyntheticDataset(batch_size=batch, device=dev, seed=7,
                                matrix_format=True)
    _, ids, _ = ds.next_batch()          # [B, 26] raw ids
    # composite keys as the collection builds them
    keys = (torch.arange(n_tables, device=dev, dtype=torch.int64)
            .unsqueeze(0) << 48) | ids
    flat = keys.t().reshape(-1)          # table-major like lookup_matrix
    uniq, inverse, counts = torch.unique(flat, return_inverse=True,
                                         return_counts=True)
    m = uniq.numel()
    nnz = flat.numel()
    c32 = counts.to(torch.int32)
    bounds = torch.zeros(m + 1, dtype=torch.int32, device=dev)
    bounds[1:] = c32.cumsum(0)
    inverse = inverse.to(torch.int32)
    order = ext.csr_order(inverse, bounds, m)
    row_ids = torch.arange(nnz, dtype=torch.int32, device=dev)
    row_coeff = torch.ones(nnz, device=dev)
    grad = torch.randn(batch, n_tables * dim, device=dev,
                       dtype=torch.bfloat16)
    print(f"m={m} nnz={nnz} max_count={int(counts.max())} "
          f"top5={counts.topk(5).values.tolist()}")

    t_order = time_fn(lambda: ext.csr_order(inverse, bounds, m))
    print(f"csr_order: {t_order:.1f} us")
    for splits in (1, 8, 16, 32, 64):
        t = time_fn(lambda: ext.group_pooled_bwd_strided(
            grad, order, bounds, row_ids, torch.Tensor(), row_coeff,
            m, torch.Tensor(), batch, n_tables, dim, True, splits))
        print(f"splits={splits:>2}: {t:7.1f} us")


if __name__ == "__main__":
    main()
