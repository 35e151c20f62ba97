"""Sample-aware batch compression.

Capability parity with the reference's sample-awared graph compression
(python/graph_optimizer/sample_awared_graph_compression.py): ranking
batches score many candidate items for ONE user, so most per-sample
feature rows repeat; deduplicating identical samples before the forward
and scattering the outputs back is result-identical with a fraction of
the compute. The reference rewrites the graph; the eager analog
deduplicates the batch at the model boundary (serving path — the
Predictor exposes it as `compress=True`).
"""
from __future__ import annotations

from typing import List, Sequence, Tuple

import torch


def _row_key(tensors: Sequence[torch.Tensor]) -> torch.Tensor:
    """One hashable row per sample: concatenated raw bytes viewed as
    int64 words (padded)."""
    parts: List[torch.Tensor] = []
    b = tensors[0].shape[0]
    for t in tensors:
        flat = t.reshape(b, -1)
        raw = flat.contiguous().cpu().view(torch.uint8).reshape(b, -1)
        pad = (-raw.shape[1]) % 8
        if pad:
            raw = torch.nn.functional.pad(raw, (0, pad))
        parts.append(raw.view(torch.int64).reshape(b, -1))
    return torch.cat(parts, dim=1)


def compress_batch(*tensors: torch.Tensor
                   ) -> Tuple[List[torch.Tensor], torch.Tensor]:
    """-> (unique-sample tensors, inverse [B]) such that
    outputs_full = outputs_unique[inverse]."""
    assert tensors and all(t.shape[0] == tensors[0].shape[0]
                           for t in tensors)
    keys = _row_key(tensors)
    uniq_rows, inverse = torch.unique(keys, dim=0, return_inverse=True)
    m = uniq_rows.shape[0]
    # representative index per unique row
    rep = torch.full((m,), -1, dtype=torch.int64)
    order = torch.arange(keys.shape[0] - 1, -1, -1)
    rep[inverse[order]] = order  # last write wins -> lowest index
    uniq = [t[rep.to(t.device)] for t in tensors]
    return uniq, inverse


def compressed_forward(fn, *tensors: torch.Tensor):
    """Run `fn` on the deduplicated batch, expand back to full size.
    Returns (out_full, unique_fraction)."""
    uniq, inverse = compress_batch(*tensors)
    out = fn(*uniq)
    inv = inverse.to(out.device if torch.is_tensor(out)
                     else out[0].device)
    if torch.is_tensor(out):
        return out[inv], uniq[0].shape[0] / tensors[0].shape[0]
    return [o[inv] for o in out], uniq[0].shape[0] / tensors[0].shape[0]
