"""Post-training embedding quantization (reference capability:
tools/low_precision_optimize): converts a checkpoint's EV value tensors to
int8 (per-row scales, ~4x smaller) or OCP fp8 e4m3 (--format fp8, ~4x
smaller and directly consumable by the fp8 serving kernels, ops/fp8.py);
also verifies reconstruction error.

Usage: python tools/quantize_embeddings.py <ckpt_dir>/ckpt-N [--apply]
       [--format int8|fp8]
"""
import argparse
import glob
import os
import sys

import torch

# runnable from anywhere: the repo root is the import root
_REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if _REPO_ROOT not in sys.path:
    sys.path.insert(0, _REPO_ROOT)
from safetensors.torch import load_file, save_file


def quantize_rows(values: torch.Tensor):
    scale = values.abs().amax(dim=1, keepdim=True).clamp(min=1e-8) / 127.0
    q = torch.clamp((values / scale).round(), -127, 127).to(torch.int8)
    return q, scale.squeeze(1)


def dequantize_rows(q: torch.Tensor, scale: torch.Tensor):
    return q.float() * scale.unsqueeze(1)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("ckpt")
    p.add_argument("--apply", action="store_true",
                   help="write <file>.<fmt>.safetensors alongside originals")
    p.add_argument("--format", choices=["int8", "fp8"], default="int8")
    args = p.parse_args()
    files = sorted(glob.glob(os.path.join(args.ckpt, "ev-*.safetensors")))
    if not files:
        print("no EV files found", file=sys.stderr)
        sys.exit(1)
    for fn in files:
        if fn.endswith((".int8.safetensors", ".fp8.safetensors")):
            continue
        data = load_file(fn)
        v = data["values"]
        if args.format == "fp8":
            from deeprec_amd.ops.fp8 import (dequantize_fp8_rows,
                                             quantize_fp8_rows)
            q, scale = quantize_fp8_rows(v)
            err = (dequantize_fp8_rows(q, scale) - v).abs().max()
        else:
            q, scale = quantize_rows(v)
            err = (dequantize_rows(q, scale) - v).abs().max()
        orig_b = v.numel() * 4
        new_b = q.numel() + scale.numel() * 4
        print(f"{os.path.basename(fn)}: rows={v.shape[0]} dim={v.shape[1]} "
              f"{orig_b / 1e6:.2f}MB -> {new_b / 1e6:.2f}MB "
              f"max_abs_err={float(err):.5f}")
        if args.apply:
            out = {k: t for k, t in data.items() if k != "values"}
            out[f"values_{args.format}"] = q
            out["values_scale"] = scale
            save_file(out, fn.replace(".safetensors",
                                      f".{args.format}.safetensors"))


if __name__ == "__main__":
    main()


def load_quantized(fn: str):
    """Load a `.int8.safetensors` / `.fp8.safetensors` EV file back into
    servable tensors: (keys, values fp32 dequantized, freqs, versions).
    Feed straight into `EmbeddingVariable.restore(...)` (or a remote KV
    publish) — completes the low-precision pipeline: quantize offline,
    serve the shrunken artifact."""
    from safetensors.torch import load_file
    data = load_file(fn)
    scale = data["values_scale"]
    if "values_int8" in data:
        values = dequantize_rows(data["values_int8"], scale)
    elif "values_fp8" in data:
        from deeprec_amd.ops.fp8 import dequantize_fp8_rows
        values = dequantize_fp8_rows(data["values_fp8"], scale)
    else:
        raise ValueError(f"{fn}: no quantized values tensor")
    return (data["keys"], values, data.get("freqs"),
            data.get("versions"))
