"""fp8 (e4m3, hipBLASLt via torch._scaled_mm) vs bf16 GEMM timing on the
ProGen-6B and 1.2B projection shapes — the VERDICT r1 item 5 "measured
fp8 ladder rung".

Usage (GPU box):  python tools/bench_fp8.py
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from progen_amd.ops import fp8
from progen_amd.tuning import enable_tuned_gemms

# (M, N, K): M = tokens (B x seq), y = x @ W^T with W (N, K)
SHAPES = [
    # ProGen-6B (dim 4096, seq 2048, B=8 -> M=16384)
    ("6B qkv", 16384, 12288, 4096),
    ("6B out", 16384, 4096, 4096),
    ("6B proj_in(GLU)", 16384, 32768, 4096),
    ("6B proj_out", 16384, 4096, 16384),
    # ProGen-1.2B (dim 1536, seq 1024, B=64 -> M=65536)
    ("1.2B qkv", 65536, 4608, 1536),
    ("1.2B proj_in", 65536, 12288, 1536),
]


def bench(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    enable_tuned_gemms()
    dev = "cuda"
    for name, M, N, K in SHAPES:
        x = torch.randn(M, K, dtype=torch.bfloat16, device=dev)
        w = torch.randn(N, K, dtype=torch.bfloat16, device=dev) * 0.02
        qx, sx = fp8.quantize_e4m3(x)
        qw, sw = fp8.quantize_e4m3(w)
        qwt = qw.t()

        us16 = bench(lambda: torch.nn.functional.linear(x, w))
        us8 = bench(lambda: torch._scaled_mm(qx, qwt, scale_a=sx, scale_b=sw,
                                             out_dtype=torch.bfloat16))
        # end-to-end including quantization (what fp8_linear pays)
        us8q = bench(lambda: fp8.fp8_linear(x, w))
        fl = 2.0 * M * N * K
        print(f"{name:18s} {M}x{N}x{K}: bf16 {us16:8.1f} us ({fl/us16/1e6:6.1f} TF/s)"
              f" | fp8 {us8:8.1f} us ({fl/us8/1e6:6.1f} TF/s, {us16/us8:4.2f}x)"
              f" | fp8+quant {us8q:8.1f} us ({us16/us8q:4.2f}x)")


if __name__ == "__main__":
    main()
