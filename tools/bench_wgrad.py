"""hipBLASLt baseline for the wgrad GEMM shapes (round-2 comparison
target for tools/wgrad_gemm.hip).

The step's weight gradients are dW = dY^T X with K = global tokens
(B*N = 65536 at the bench operating point) — K-major operands, fp32
accumulate, bf16 out. Prints us/TF per shape.

Usage (GPU box):  python tools/bench_wgrad.py
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from progen_amd.tuning import enable_tuned_gemms

# ProGen-1.2B wgrad shapes at B=64, seq 1024 (K = 65536):
#   to_qkv   dW: (3*h*dh, dim)   = (4608, 1536)
#   to_out   dW: (dim, h*dh)     = (1536, 1536)
#   proj_in  dW: (2*4*dim, dim)  = (12288, 1536)  [GLU layers]
#   proj_out dW: (dim, 4*dim)    = (1536, 6144)
SHAPES = [(4608, 1536), (1536, 1536), (12288, 1536), (1536, 6144)]
K = 65536


def main():
    enable_tuned_gemms()
    dev = "cuda"
    for M, N in SHAPES:
        dy = torch.randn(K, M, dtype=torch.bfloat16, device=dev)
        x = torch.randn(K, N, dtype=torch.bfloat16, device=dev)
        for _ in range(5):
            dw = dy.t() @ x
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        iters = 50
        for _ in range(iters):
            dw = dy.t() @ x
        torch.cuda.synchronize()
        us = (time.perf_counter() - t0) / iters * 1e6
        tf = 2.0 * M * N * K / (us * 1e-6) / 1e12
        print(f"hipBLASLt wgrad {M}x{N}x{K}: {us:.1f} us  {tf:.1f} TF/s")
    del dw  # noqa: F841


if __name__ == "__main__":
    main()
