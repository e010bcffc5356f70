"""Measure decode latency/throughput of the cached decoder on the
FLAGSHIP ProGen-1.2B config (dim 1536, depth 36, heads 24, dh 64,
wsz 256 — the same architecture as bench.py's headline; the r01 decode
numbers were mislabeled on a different 2048/24/dh128 model, VERDICT r1
weak #8).

Modes: eager per-token step, hipGraph-captured step (--graph), batched
(--batch N). Serving metric: ms/token (batch 1) and tokens/s aggregate.

Usage (GPU box):  python tools/bench_decode.py [--tokens 128] [--graph] [--batch 8]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from progen_amd import ProGenBase
from progen_amd.config import ProGenConfig
from progen_amd.decode import (DecodeCache, GraphedDecodeStep, forward_step,
                               forward_step_static)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--tokens", type=int, default=128)
    ap.add_argument("--warmup", type=int, default=16)
    ap.add_argument("--graph", action="store_true")
    ap.add_argument("--batch", type=int, default=1)
    args = ap.parse_args()

    # flagship ProGen-1.2B (configs/model/progen_1b.toml)
    cfg = ProGenConfig(num_tokens=256, dim=1536, depth=36, heads=24,
                       dim_head=64, window_size=256, seq_len=1024,
                       global_mlp_depth=2)
    torch.manual_seed(0)
    m = ProGenBase(cfg).to(device="cuda", dtype=torch.bfloat16).eval()
    m.rotary_sin = m.rotary_sin.float()
    m.rotary_cos = m.rotary_cos.float()

    B = args.batch
    cache = DecodeCache(m, batch=B)
    tok = torch.randint(1, 256, (B,), device="cuda")

    if args.graph:
        # a couple of eager steps as prefill, then capture
        for _ in range(2):
            logits = forward_step(m, tok, cache)
            tok = logits.argmax(dim=-1)
        tok = tok.cpu()
        g = GraphedDecodeStep(m, cache, start_pos=cache.pos)
        # pure-replay rule: sampling runs on the HOST (memcpys only
        # between replays — a device argmax kernel would poison replay)
        for _ in range(args.warmup):
            logits = g.step(tok)
            tok = logits.cpu().float().argmax(dim=-1)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.tokens):
            logits = g.step(tok)
            tok = logits.cpu().float().argmax(dim=-1)
        torch.cuda.synchronize()
        mode = "graphed"
    else:
        for _ in range(args.warmup):
            logits = forward_step(m, tok, cache)
            tok = logits.argmax(dim=-1)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.tokens):
            logits = forward_step(m, tok, cache)
            tok = logits.argmax(dim=-1)
        torch.cuda.synchronize()
        mode = "eager"
    dt = time.perf_counter() - t0
    n = args.tokens * B
    print(f"decode[{mode}, batch {B}]: {args.tokens} steps in {dt:.3f}s -> "
          f"{n / dt:.1f} tok/s aggregate, {1e3 * dt / args.tokens:.2f} ms/step "
          f"(ProGen-1.2B flagship config, bf16, cached)")


if __name__ == "__main__":
    main()
