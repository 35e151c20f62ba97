"""Post-training embedding quantization (reference capability:
tools/low_precision_optimize): converts a checkpoint's EV value tensors to
int8 with per-row scales, shrinking serving checkpoints ~4x; also verifies
reconstruction error.

Usage: python tools/quantize_embeddings.py <ckpt_dir>/ckpt-N [--apply]
"""
import argparse
import glob
import os
import sys

import torch
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
                   help="write <file>.int8.safetensors alongside originals")
    args = p.parse_args()
    files = sorted(glob.glob(os.path.join(args.ckpt, "ev-*.safetensors")))
    if not files:
        print("no EV files found", file=sys.stderr)
        sys.exit(1)
    for fn in files:
        if fn.endswith(".int8.safetensors"):
            continue
        data = load_file(fn)
        v = data["values"]
        q, scale = quantize_rows(v)
        err = (dequantize_rows(q, scale) - v).abs().max()
        orig_b = v.numel() * 4
        new_b = q.numel() + scale.numel() * 4
        print(f"{os.path.basename(fn)}: rows={v.shape[0]} dim={v.shape[1]} "
              f"{orig_b / 1e6:.2f}MB -> {new_b / 1e6:.2f}MB "
              f"max_abs_err={float(err):.5f}")
        if args.apply:
            out = {k: t for k, t in data.items() if k != "values"}
            out["values_int8"] = q
            out["values_scale"] = scale
            save_file(out, fn.replace(".safetensors", ".int8.safetensors"))


if __name__ == "__main__":
    main()
