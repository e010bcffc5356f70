"""Isolate the graphed-train NaN seen via train.py --graph on real data.

Variants (graphed vs eager, 14 steps, ProGen-small):
  A: unpadded batches, pre-allocated on device (== the passing GPU test)
  B: PADDED batches (zero tails of varying length), pre-allocated
  C: unpadded, freshly H2D-copied each step (train.py's my_shard pattern)
  D: padded + fresh H2D  (== train.py)
"""
import copy
import sys

import torch

from progen_amd import ProGenBase, ProGenConfig
from progen_amd.optim import ProGenAdamW
from progen_amd.runtime import GraphedTrainStep
from progen_amd.utils import compute_loss

dev = torch.device("cuda:0")
STEPS = 14


def make_batches(padded: bool):
    g = torch.Generator().manual_seed(123)
    out = []
    for i in range(STEPS):
        d = torch.randint(1, 256, (32, 1025), generator=g)
        d[:, 0] = 0
        if padded:
            # uniref-like ragged tails: row r keeps 64..1024 real tokens
            lens = torch.randint(64, 1024, (32,), generator=g)
            for r in range(32):
                d[r, lens[r]:] = 0
        out.append(d)
    return out


def run(padded: bool, fresh: bool, graphed: bool):
    torch.manual_seed(21)
    cfg = ProGenConfig(num_tokens=256, dim=512, depth=12, dim_head=64,
                       heads=8, window_size=256, seq_len=1024,
                       global_mlp_depth=2)
    m = ProGenBase(cfg).to(device=dev, dtype=torch.bfloat16)
    m.rotary_sin = m.rotary_sin.float()
    m.rotary_cos = m.rotary_cos.float()
    o = ProGenAdamW(m, lr=2e-4, weight_decay=1e-3, max_grad_norm=0.5)
    batches = make_batches(padded)
    if not fresh:
        batches = [b.to(dev) for b in batches]
    losses = []
    if graphed:
        g = GraphedTrainStep(m, o, None, 32, 1024, dev)
        for b in batches:
            bb = b.to(dev) if fresh else b
            losses.append(g.run(bb).item())
    else:
        for b in batches:
            bb = b.to(dev) if fresh else b
            o.zero_grad()
            loss = compute_loss(m, bb)
            loss.backward()
            o.step()
            losses.append(loss.item())
    return losses


for name, padded, fresh in [("A", False, False), ("B", True, False),
                            ("C", False, True), ("D", True, True)]:
    le = run(padded, fresh, graphed=False)
    lg = run(padded, fresh, graphed=True)
    bad = any(x != x for x in lg)
    print(f"variant {name} padded={padded} fresh={fresh} "
          f"NaN={'YES' if bad else 'no'}", flush=True)
    print("  eager :", " ".join(f"{x:.4f}" for x in le), flush=True)
    print("  graph :", " ".join(f"{x:.4f}" for x in lg), flush=True)
    if bad:
        sys.stdout.flush()
