"""Per-shape microbenchmark of the fused-MLP kernels vs torch (rocBLAS).

Times linear_fwd / linear_dx / linear_dw on each DLRM layer shape with
hipEvents. Run on a GPU box:
    python tools/gemm_sweep.py
"""
import torch

from deeprec_amd.ops.build_ext import require_extension

SHAPES = [  # (M, N, K) of the DLRM MLP layers at batch 8192
    (8192, 512, 16),
    (8192, 256, 512),
    (8192, 64, 256),
    (8192, 16, 64),
    (8192, 512, 480),
    (8192, 256, 512),
    (8192, 1, 256),
]


def time_fn(fn, iters=200):
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
    print(f"{'M':>6} {'N':>4} {'K':>4} | {'fwd0':>7} {'fwd1':>7} "
          f"{'dx':>7} {'dw16':>7} {'dw64':>7} {'dw64t':>7}  (us)")
    tot = [0.0] * 6
    for M, N, K in SHAPES:
        A = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
        W = torch.randn(N, K, device=dev, dtype=torch.bfloat16)
        b = torch.randn(N, device=dev, dtype=torch.float32)
        G = torch.randn(M, N, device=dev, dtype=torch.bfloat16)
        dW = torch.zeros(N, K + 1, device=dev, dtype=torch.float32)

        fwd = time_fn(lambda: ext.linear_fwd(A, W, b, 1, 0))
        fwd1 = time_fn(lambda: ext.linear_fwd(A, W, b, 1, 1))
        dx = time_fn(lambda: ext.linear_dx(G, W))
        dw0 = time_fn(lambda: ext.linear_dw(G, A, True, 0))
        dw1 = time_fn(lambda: ext.linear_dw(G, A, True, 1))
        dw2 = time_fn(lambda: ext.linear_dw(G, A, True, 2))
        print(f"{M:>6} {N:>4} {K:>4} | {fwd:7.1f} {fwd1:7.1f} {dx:7.1f} "
              f"{dw0:7.1f} {dw1:7.1f} {dw2:7.1f}")
        for i, v in enumerate((fwd, fwd1, dx, dw0, dw1, dw2)):
            tot[i] += v
    print(f"{'TOTAL':>16} | {tot[0]:7.1f} {tot[1]:7.1f} {tot[2]:7.1f} "
          f"{tot[3]:7.1f} {tot[4]:7.1f} {tot[5]:7.1f}")


if __name__ == "__main__":
    main()
