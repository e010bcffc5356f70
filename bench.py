"""Benchmark: train tokens/sec, whole node (BASELINE.json contract).

Flagship config: ProGen-1.2B (depth=36, dim=1536, heads=24, seq_len=1024,
window=256, global_mlp_depth=2), bf16 compute, synthetic Uniref50-shaped
token batches, random-init weights. One full training step = forward +
masked-CE loss + backward (+ bucketed RCCL grad all-reduce under DP) +
fused clip+AdamW update.

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W          # single GPU
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N ...      # DP over RCCL

Rank 0 prints ONE JSON line with the whole-job aggregate tokens/sec.
"""

import argparse
import json
import math
import os
import time

import torch

from progen_amd import ProGenBase, ProGenConfig
from progen_amd.optim import ProGenAdamW
from progen_amd.parallel import DistributedTrainer, init_distributed
from progen_amd.utils import compute_loss

CONFIGS = {
    "progen-1.2b": dict(num_tokens=256, dim=1536, depth=36, heads=24,
                        dim_head=64, window_size=256, seq_len=1024,
                        global_mlp_depth=2),
    "progen-small": dict(num_tokens=256, dim=512, depth=12, heads=8,
                         dim_head=64, window_size=256, seq_len=1024,
                         global_mlp_depth=2),
    "tiny-cpu": dict(num_tokens=256, dim=128, depth=2, heads=2, dim_head=64,
                     window_size=64, seq_len=256, global_mlp_depth=1),
    # BASELINE config #5 vehicle (TP=8 target; single-GPU bench runs it
    # data-parallel-1 at a small batch, optionally with --fp8)
    "progen-6b": dict(num_tokens=256, dim=4096, depth=24, heads=64,
                      dim_head=64, window_size=512, seq_len=2048,
                      ff_glu=True, global_mlp_depth=2),
}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--warmup", type=int, default=8)
    p.add_argument("--model", default="progen-1.2b", choices=list(CONFIGS))
    p.add_argument("--batch", type=int, default=64, help="per-GPU batch size")
    p.add_argument("--dtype", default="bf16", choices=["bf16", "fp32"])
    p.add_argument("--fp8", action="store_true",
                   help="route big projections through e4m3 hipBLASLt "
                        "GEMMs (fwd+dgrad; wgrad stays bf16) — the "
                        "BASELINE #5 fp8 mode; reported dtype becomes "
                        "bf16+fp8")
    args = p.parse_args()
    if args.fp8:
        from progen_amd.ops import fp8 as _fp8
        _fp8.ENABLED = True

    from progen_amd.tuning import enable_tuned_gemms
    enable_tuned_gemms()
    local_rank = init_distributed()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    on_gpu = torch.cuda.is_available()
    device = torch.device("cuda", local_rank) if on_gpu else torch.device("cpu")

    if not on_gpu and args.model == "progen-1.2b":
        args.model = "tiny-cpu"  # plumbing check only (BASELINE config #1)

    cfg = ProGenConfig(**CONFIGS[args.model])
    dtype = torch.bfloat16 if (args.dtype == "bf16" and on_gpu) else torch.float32

    torch.manual_seed(1234)  # identical init on every rank (DP replicas)
    module = ProGenBase(cfg).to(device=device, dtype=dtype)
    module.rotary_sin = module.rotary_sin.float()
    module.rotary_cos = module.rotary_cos.float()

    if world > 1 and os.environ.get("PROGEN_ZERO1") == "1":
        from progen_amd.parallel.zero1 import Zero1AdamW
        optim = Zero1AdamW(module, lr=2e-4, weight_decay=1e-3, max_grad_norm=0.5)
    else:
        optim = ProGenAdamW(module, lr=2e-4, weight_decay=1e-3, max_grad_norm=0.5)
    ddp = DistributedTrainer(optim.space)
    if world > 1:  # belt-and-braces: bitwise-identical replicas
        torch.distributed.broadcast(optim.space.flat, src=0)
        optim.resync_master()
    torch.manual_seed(1234 + rank)  # rank-local data stream

    B, N = args.batch, cfg.seq_len
    # synthetic Uniref50-shaped batch: byte tokens 1..256 with zero BOS
    # column and a zero-pad tail (shape (B, N+1) like the data pipeline)
    data = torch.randint(1, cfg.num_tokens, (B, N + 1), device=device)
    data[:, 0] = 0
    data[:, -8:] = 0  # pad tail so the EOS-mask path is exercised

    graphed = None
    # hipGraph step: OPT-IN via PROGEN_GRAPH=1 since the late-r2 findings
    # (profiles/r02_graphed_nan_investigation.md): the graph's measured
    # launch-overhead win at the flagship config is ~0% once the
    # trajectory is finite, multi-step replay has an open probabilistic
    # corruption issue (mitigated by the AdamW non-finite skip guard),
    # and the r1 "graph is 12% faster" delta turned out to be the
    # NaN-poisoned weights clocking the power-capped chip higher. Eager
    # is the trajectory-honest default. Under world>1 the opt-in also
    # captures the RCCL all-reduce (one fixed-size collective per
    # replay, symmetric by construction); capture failure on any rank
    # degrades every rank to eager via the MIN-reduce agreement below.
    want_graph = os.environ.get("PROGEN_GRAPH") == "1"
    if want_graph:
        os.environ.setdefault("PROGEN_GRAPH_DP", "1")
    if on_gpu and want_graph:
        from progen_amd.runtime import GraphedTrainStep
        try:
            graphed = GraphedTrainStep(module, optim, ddp, B, N, device)
            if rank == 0:
                import sys
                print("[bench] hipGraph step captured", file=sys.stderr)
        except Exception:  # noqa: BLE001 — eager fallback, report it
            import sys
            import traceback
            traceback.print_exc()
            print("[bench] hipGraph capture failed; eager fallback",
                  file=sys.stderr)
        if world > 1:
            # all ranks must agree on the mode: the graphed step's
            # collective sequence (one whole-buffer all-reduce) differs
            # from eager's bucketed one, so a mixed fleet would deadlock
            # at the first step. MIN-reduce a success flag; any failure
            # degrades every rank to eager (the
            # graphed-DP opt-in must work or degrade CLEANLY).
            flag = torch.tensor([1 if graphed is not None else 0],
                                device=device)
            torch.distributed.all_reduce(
                flag, op=torch.distributed.ReduceOp.MIN)
            if int(flag.item()) == 0 and graphed is not None:
                graphed = None
                import sys
                print("[bench] a peer rank failed capture; all ranks "
                      "running eager", file=sys.stderr)

    def step():
        if graphed is not None:
            return graphed.run(data)
        loss = compute_loss(module, data)
        loss.backward()
        ddp.finish_backward()
        optim.step()
        optim.zero_grad()
        return loss

    def barrier_sync():
        if world > 1:
            torch.distributed.barrier()
        if on_gpu:
            torch.cuda.synchronize()

    for _ in range(args.warmup):
        step()
    barrier_sync()

    t0 = time.perf_counter()
    for _ in range(args.steps):
        last_loss = step()
    barrier_sync()
    elapsed = time.perf_counter() - t0
    # read AFTER the timed region: a silent NaN trajectory would make the
    # throughput number meaningless as training evidence
    final_loss = float(ddp.all_reduce_scalar(last_loss.detach()).item())
    if rank == 0 and not math.isfinite(final_loss):
        import sys
        print(f"[bench] WARNING: final loss is {final_loss} — the step "
              f"work is still representative but the trajectory is not",
              file=sys.stderr)

    # MAX over ranks
    if world > 1:
        t = torch.tensor([elapsed], device=device if on_gpu else "cpu")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = t.item()

    tokens_per_step = B * N * world  # whole-job tokens per step
    toks_per_sec = tokens_per_step * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000

    if rank == 0:
        print(json.dumps({
            "metric": "train tokens/sec (whole node)",
            "value": toks_per_sec,
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": ("bf16+fp8" if args.fp8 else
                      "bf16" if dtype == torch.bfloat16 else "fp32"),
            "data": "synthetic",
            "final_loss": round(final_loss, 4),
            "config": {
                "model": "ProGen-1.2B" if args.model == "progen-1.2b" else args.model,
                "global_batch": B * world,
                "seq_len": N,
                "parallelism": f"dp{world}",
            },
        }))

    # synchronized teardown (see train.py): a fast rank exiting early can
    # SIGABRT a slower rank still inside process-group destruction; the
    # result JSON is already printed, so teardown failures must not turn
    # a measured run into a nonzero exit (world=8 first-shot hardening)
    if world > 1:
        try:
            torch.distributed.barrier()
            torch.distributed.destroy_process_group()
        except Exception:  # noqa: BLE001
            import sys
            import traceback
            traceback.print_exc()
            print("[bench] teardown error ignored (result already "
                  "reported)", file=sys.stderr)


if __name__ == "__main__":
    main()
