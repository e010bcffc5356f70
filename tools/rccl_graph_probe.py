"""Probe: RCCL all_reduce INSIDE a hipGraph capture (1-rank group)."""
import os
import torch
import torch.distributed as dist

os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29511")
dist.init_process_group("nccl", rank=0, world_size=1)
torch.cuda.set_device(0)

x = torch.randn(1 << 20, device="cuda")
buf = torch.zeros_like(x)

side = torch.cuda.Stream()
side.wait_stream(torch.cuda.current_stream())
with torch.cuda.stream(side):
    for _ in range(3):
        buf.copy_(x)
        dist.all_reduce(buf)
        buf.mul_(0.5)
torch.cuda.current_stream().wait_stream(side)
torch.cuda.synchronize()

g = torch.cuda.CUDAGraph()
try:
    with torch.cuda.graph(g):
        buf.copy_(x)
        dist.all_reduce(buf)
        buf.mul_(0.5)
    print("CAPTURE OK", flush=True)
    for i in range(5):
        g.replay()
    torch.cuda.synchronize()
    err = (buf - 0.5 * x).abs().max().item()
    print("REPLAY OK err", err, flush=True)
except Exception as e:
    print("CAPTURE FAILED:", type(e).__name__, str(e)[:200], flush=True)
dist.destroy_process_group()
print("DONE", flush=True)
