"""Bisect the replay race: graphed variant-A run with PROGEN_FORCE_EAGER=1
(torch-native ops captured) vs default (HIP kernels captured)."""
import os

import torch

from progen_amd import ProGenBase, ProGenConfig
from progen_amd.optim import ProGenAdamW
from progen_amd.runtime import GraphedTrainStep

dev = torch.device("cuda:0")
STEPS = 14


def build():
    torch.manual_seed(21)
    cfg = ProGenConfig(num_tokens=256, dim=512, depth=12, dim_head=64,
                       heads=8, window_size=256, seq_len=1024,
                       global_mlp_depth=2)
    m = ProGenBase(cfg).to(device=dev, dtype=torch.bfloat16)
    m.rotary_sin = m.rotary_sin.float()
    m.rotary_cos = m.rotary_cos.float()
    o = ProGenAdamW(m, lr=2e-4, weight_decay=1e-3, max_grad_norm=0.5)
    return m, o


def batches():
    g = torch.Generator().manual_seed(123)
    out = []
    for _ in range(STEPS):
        d = torch.randint(1, 256, (32, 1025), generator=g)
        d[:, 0] = 0
        out.append(d.to(dev))
    return out


for force in ("1", "0"):
    os.environ["PROGEN_FORCE_EAGER"] = force
    m, o = build()
    g = GraphedTrainStep(m, o, None, 32, 1024, dev)
    losses = [g.run(b).item() for b in batches()]
    tag = "torch-eager-ops" if force == "1" else "HIP-ops"
    bad = any(x != x for x in losses)
    print(f"graphed [{tag}] NaN={'YES' if bad else 'no'}:",
          " ".join(f"{x:.4f}" for x in losses), flush=True)
