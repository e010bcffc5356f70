"""TP=8 training bench for ProGen-6B (BASELINE.json config #5).

Round-2 tool: needs a multi-GPU box (TP over RCCL on the xGMI mesh).
Launch:
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \\
        --master-addr 127.0.0.1 tools/bench_tp.py --steps 20 --warmup 5

Every rank holds the full replicated activations and a 1/8 shard of the
projection weights (parallel/tp_model.py); the per-block all-reduces run
on RCCL. Prints one JSON line (rank 0) in the same shape as bench.py.
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist

from progen_amd.config import ProGenConfig
from progen_amd.models.progen import ProGenBase
from progen_amd.optim import ProGenAdamW
from progen_amd.parallel import tp, tp_model
from progen_amd.utils import compute_loss


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--model", default="progen_6b",
                    help="configs/model/<name>.toml")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    on_gpu = torch.cuda.is_available()
    if world > 1:
        dist.init_process_group("nccl" if on_gpu else "gloo")
    if on_gpu:
        torch.cuda.set_device(local_rank)
    device = torch.device("cuda", local_rank) if on_gpu else torch.device("cpu")

    try:
        import tomllib
    except ModuleNotFoundError:
        import tomli as tomllib
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    with open(os.path.join(repo, "configs", "model",
                           f"{args.model}.toml"), "rb") as f:
        cfg = ProGenConfig.from_dict(tomllib.load(f))
    torch.manual_seed(1234)  # identical full weights on every rank
    model = ProGenBase(cfg)
    if world > 1:
        tp.init_tensor_parallel(world)
        model = tp_model.tp_shard_(model)
    model = model.to(device=device, dtype=torch.bfloat16 if on_gpu else torch.float32)
    if on_gpu:
        model.rotary_sin = model.rotary_sin.float()
        model.rotary_cos = model.rotary_cos.float()
    optim = ProGenAdamW(model, lr=2e-4)
    if world > 1:
        optim.norm_sumsq_fn = tp_model.tp_grad_sumsq_fn(model)

    B, N = args.batch, cfg.seq_len
    g = torch.Generator().manual_seed(7)
    data = torch.randint(1, cfg.num_tokens, (B, N + 1), generator=g).to(device)
    data[:, 0] = 0

    def step():
        loss = compute_loss(model, data)
        loss.backward()
        if world > 1:
            tp_model.sync_replicated_grads(model)
        optim.step()
        optim.zero_grad()
        return loss

    def sync():
        if world > 1:
            dist.barrier()
        if on_gpu:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        step()
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    sync()
    elapsed = time.perf_counter() - t0
    if world > 1:
        t = torch.tensor([elapsed], device=device if on_gpu else None)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    if rank == 0:
        toks = args.steps * B * N  # TP: one global batch per step
        print(json.dumps({
            "metric": "train tokens/sec (whole node)",
            "value": toks / elapsed,
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": 1e3 * elapsed / args.steps,
            "higher_is_better": True,
            "scaling": "strong",
            "dtype": "bf16" if on_gpu else "fp32",
            "data": "synthetic",
            "config": {"model": args.model, "global_batch": B,
                       "seq_len": N, "parallelism": f"tp{world}"},
        }))
    if world > 1:
        dist.barrier()
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
