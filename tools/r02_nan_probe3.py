"""Bisect the graphed-replay corruption per kernel family
(profiles/r02_graphed_nan_investigation.md): for each PROGEN_EAGER_OPS
setting, run the SAME 14 steps eager and graphed and compare
trajectories. Since the AdamW non-finite skip guard now swallows the
NaN explosions, the corruption signal is (a) trajectory divergence vs
the same-mode eager run, (b) skipped steps visible as a lagging device
step counter. If routing one op family to torch makes replays track
eager, that family is implicated; if ALL-torch still diverges, the
issue is runtime/hardware. Failure is probabilistic (~25%/replay) — run
the sweep a few times.
"""
import os

import torch

from progen_amd import ProGenBase, ProGenConfig
from progen_amd.optim import ProGenAdamW
from progen_amd.runtime import GraphedTrainStep
from progen_amd.utils import compute_loss

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


def run_eager():
    m, o = build()
    losses = []
    for b in batches():
        o.zero_grad()
        loss = compute_loss(m, b)
        loss.backward()
        o.step()
        losses.append(loss.item())
    return losses


def run_graphed():
    m, o = build()
    g = GraphedTrainStep(m, o, None, 32, 1024, dev)
    losses = [g.run(b).item() for b in batches()]
    torch.cuda.synchronize()
    return losses, int(o.step_dev.item())


CASES = [
    ("all-HIP", {}),
    ("eager:attn", {"PROGEN_EAGER_OPS": "attn"}),
    ("eager:sgu", {"PROGEN_EAGER_OPS": "sgu"}),
    ("eager:ln", {"PROGEN_EAGER_OPS": "ln"}),
    ("eager:glu", {"PROGEN_EAGER_OPS": "glu"}),
    ("eager:ce", {"PROGEN_EAGER_OPS": "ce"}),
    ("eager:adamw", {"PROGEN_EAGER_OPS": "adamw"}),
    ("all-torch", {"PROGEN_FORCE_EAGER": "1"}),
]

for tag, env in CASES:
    for k in ("PROGEN_EAGER_OPS", "PROGEN_FORCE_EAGER"):
        os.environ.pop(k, None)
    os.environ.update(env)
    try:
        le = run_eager()
        lg, applied = run_graphed()
        bad = any(x != x for x in lg)
        div = next((i for i, (a, b) in enumerate(zip(le, lg))
                    if abs(a - b) > 0.05 * max(1.0, abs(a))), None)
        verdict = ("NaN@" + str(next(i for i, x in enumerate(lg) if x != x))
                   if bad else
                   f"DIVERGED@{div}" if div is not None else "ok")
        print(f"[{tag}] {verdict} applied={applied}/{STEPS}", flush=True)
        print("  eager :", " ".join(f"{x:.4f}" for x in le), flush=True)
        print("  graph :", " ".join(f"{x:.4f}" for x in lg), flush=True)
    except Exception as e:  # noqa: BLE001 — one failure shouldn't end the sweep
        print(f"[{tag}] ERROR: {type(e).__name__}: {str(e)[:200]}", flush=True)
    torch.cuda.synchronize()
    torch.cuda.empty_cache()
