"""Instrumented repro: after each replay, D2H-copy (replay-safe) the
flat grad / master / params and analyze on CPU — find WHICH parameter's
gradient first goes non-finite in the graphed run."""
import torch

from progen_amd import ProGenBase, ProGenConfig
from progen_amd.optim import ProGenAdamW
from progen_amd.runtime import GraphedTrainStep
from progen_amd.utils import compute_loss

dev = torch.device("cuda:0")
STEPS = 7


def build():
    torch.manual_seed(21)
    cfg = ProGenConfig(num_tokens=256, dim=512, depth=12, dim_head=64,
                       heads=8, window_size=256, seq_len=1024,
                       global_mlp_depth=2)
    m = ProGenBase(cfg).to(device=dev, dtype=torch.bfloat16)
    m.rotary_sin = m.rotary_sin.float()
    m.rotary_cos = m.rotary_cos.float()
    o = ProGenAdamW(m, lr=2e-4, weight_decay=1e-3, max_grad_norm=0.5)
    name_of = {id(p): n for n, p in m.named_parameters()}
    names = [name_of[id(p)] for p in o.space.params]
    return m, o, names


def batches():
    g = torch.Generator().manual_seed(123)
    out = []
    for _ in range(STEPS):
        d = torch.randint(1, 256, (32, 1025), generator=g)
        d[:, 0] = 0
        out.append(d.to(dev))
    return out


def report(o, names, tag):
    g = o.space.flat_grad.detach().cpu().float()
    p = o.space.flat.detach().cpu().float()
    mast = o.master.detach().cpu()
    va = o.exp_avg_sq.detach().cpu()
    bad = ~torch.isfinite(g)
    print(f"  {tag}: gmax={g.abs().max():.3e} gnorm={g.norm():.3e} "
          f"nonfin={int(bad.sum())} pmax={p.abs().max():.3e} "
          f"mmax={mast.abs().max():.3e} vmax={va.max():.3e}", flush=True)
    if bad.any():
        per = []
        for (off, n), nm in zip(o.space.offsets, names):
            c = int(bad[off:off + n].sum())
            if c:
                per.append((c, nm))
        per.sort(reverse=True)
        for c, nm in per[:8]:
            print(f"    nonfinite {c:8d}  {nm}", flush=True)
        return True
    return False


print("=== eager ===", flush=True)
m, o, names = build()
for i, b in enumerate(batches()):
    o.zero_grad()
    loss = compute_loss(m, b)
    loss.backward()
    o.step()
    print(f"step {i} loss {loss.item():.4f}", flush=True)
    report(o, names, "post-step")

print("=== graphed ===", flush=True)
m, o, names = build()
g = GraphedTrainStep(m, o, None, 32, 1024, dev)
for i, b in enumerate(batches()):
    loss = g.run(b)
    torch.cuda.synchronize()
    print(f"step {i} loss {loss.item():.4f}", flush=True)
    if report(o, names, "post-replay"):
        break
