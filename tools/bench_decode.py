"""Measure single-stream decode latency/throughput of the cached decoder
on the flagship ProGen-1.2B config (serving path).

Usage (GPU box):  python tools/bench_decode.py [--tokens 128]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from progen_amd import ProGenBase
from progen_amd.config import ProGenConfig
from progen_amd.decode import DecodeCache, forward_step


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--tokens", type=int, default=128)
    ap.add_argument("--warmup", type=int, default=16)
    args = ap.parse_args()

    cfg = ProGenConfig(num_tokens=256, dim=2048, depth=24, heads=16,
                       dim_head=128, window_size=512, seq_len=1024,
                       ff_glu=True, global_mlp_depth=2)
    torch.manual_seed(0)
    m = ProGenBase(cfg).to(device="cuda", dtype=torch.bfloat16).eval()
    m.rotary_sin = m.rotary_sin.float()
    m.rotary_cos = m.rotary_cos.float()

    cache = DecodeCache(m, batch=1)
    tok = torch.randint(1, 256, (1,), device="cuda")
    for _ in range(args.warmup):
        logits = forward_step(m, tok, cache)
        tok = logits.argmax(dim=-1)
    torch.cuda.synchronize()

    t0 = time.perf_counter()
    for _ in range(args.tokens):
        logits = forward_step(m, tok, cache)
        tok = logits.argmax(dim=-1)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"decode: {args.tokens} tokens in {dt:.3f}s "
          f"-> {args.tokens / dt:.1f} tok/s, {1e3 * dt / args.tokens:.2f} ms/tok "
          f"(ProGen-1.2B bf16, batch 1, cached)")


if __name__ == "__main__":
    main()
