"""Device-generic (torch) segment pooling used by the CPU path and as the
fp32 numerics reference for the fused HIP kernels.

Combiner semantics match the reference's embedding_lookup_sparse
(reference: python/ops/embedding_ops.py):
  sum   -> sum_i w_i e_i
  mean  -> sum_i w_i e_i / sum_i w_i
  sqrtn -> sum_i w_i e_i / sqrt(sum_i w_i^2)
Empty rows produce zeros.
"""
from __future__ import annotations

import torch

COMBINERS = ("sum", "mean", "sqrtn")


def _row_coeff(offsets: torch.Tensor, combiner: str,
               weights: torch.Tensor, row_ids: torch.Tensor) -> torch.Tensor:
    """Per-row normalization coefficient; [batch] float."""
    b = offsets.numel() - 1
    dev = offsets.device
    if combiner == "sum":
        return torch.ones(b, device=dev)
    if weights is None:
        lengths = (offsets[1:] - offsets[:-1]).to(torch.float32)
        denom = lengths if combiner == "mean" else lengths.sqrt()
    else:
        acc = torch.zeros(b, device=dev, dtype=torch.float32)
        w = weights.float() if combiner == "mean" else weights.float() ** 2
        acc.index_add_(0, row_ids.long(), w)
        denom = acc if combiner == "mean" else acc.sqrt()
    return 1.0 / denom.clamp(min=1e-12)


def pooled_forward(emb: torch.Tensor, inverse: torch.Tensor,
                   offsets: torch.Tensor, row_ids: torch.Tensor,
                   combiner: str = "mean", weights: torch.Tensor = None,
                   out_dtype=None) -> torch.Tensor:
    """emb: [m, D] unique-key embeddings; returns [batch, D]."""
    assert combiner in COMBINERS
    b = offsets.numel() - 1
    d = emb.shape[1]
    gathered = emb.float()[inverse.long()]           # [nnz, D]
    if weights is not None:
        gathered = gathered * weights.float().unsqueeze(1)
    out = torch.zeros(b, d, device=emb.device, dtype=torch.float32)
    out.index_add_(0, row_ids.long(), gathered)
    coeff = _row_coeff(offsets, combiner, weights, row_ids)
    out = out * coeff.unsqueeze(1)
    return out.to(out_dtype or emb.dtype)


def pooled_backward(grad_out: torch.Tensor, inverse: torch.Tensor,
                    offsets: torch.Tensor, row_ids: torch.Tensor, m: int,
                    combiner: str = "mean",
                    weights: torch.Tensor = None) -> torch.Tensor:
    """Gradient w.r.t. the [m, D] unique-key embeddings."""
    assert combiner in COMBINERS
    coeff = _row_coeff(offsets, combiner, weights, row_ids)
    g = grad_out.float() * coeff.unsqueeze(1)        # [batch, D]
    g_nnz = g[row_ids.long()]                        # [nnz, D]
    if weights is not None:
        g_nnz = g_nnz * weights.float().unsqueeze(1)
    grad_unique = torch.zeros(m, grad_out.shape[1],
                              device=grad_out.device, dtype=torch.float32)
    grad_unique.index_add_(0, inverse.long(), g_nnz)
    return grad_unique
