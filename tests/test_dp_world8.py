"""DP at the DRIVER'S world size: gloo world=8 on CPU. The 8-GPU scale
bench is a first-shot run (no 8-GPU node in the dev loop), so the exact
world-8 semantics — bucketed all-reduce, per-rank sharding, ZeRO-1
gather geometry, synchronized teardown — are pinned here against the
single-process full-batch trajectory."""

import multiprocessing as mp
import os
import socket

import pytest
import torch


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _cfg():
    from progen_amd.config import ProGenConfig
    return ProGenConfig(num_tokens=64, dim=16, depth=2, dim_head=4,
                        heads=2, window_size=8, seq_len=32, ff_glu=True,
                        global_mlp_depth=1)


def _batches(steps, per_rank=1, world=8):
    torch.manual_seed(99)
    out = []
    for _ in range(steps):
        d = torch.randint(1, 64, (per_rank * world, 33))
        d[:, 0] = 0
        out.append(d)
    return out


def _losses_single(steps=2):
    from progen_amd.models.progen import ProGenBase
    from progen_amd.optim import ProGenAdamW
    from progen_amd.utils import compute_loss
    torch.manual_seed(77)
    model = ProGenBase(_cfg()).double()
    optim = ProGenAdamW(model, lr=1e-3)
    losses = []
    for data in _batches(steps):
        loss = compute_loss(model, data)
        loss.backward()
        optim.step()
        optim.zero_grad()
        losses.append(loss.item())
    return losses


def _worker(rank, world, port, q, zero1, steps=2):
    import torch.distributed as dist
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from progen_amd.models.progen import ProGenBase
    from progen_amd.optim import ProGenAdamW
    from progen_amd.parallel.ddp import DistributedTrainer
    from progen_amd.utils import compute_loss
    try:
        torch.manual_seed(77)
        model = ProGenBase(_cfg()).double()
        if zero1:
            from progen_amd.parallel.zero1 import Zero1AdamW
            optim = Zero1AdamW(model, lr=1e-3)
        else:
            optim = ProGenAdamW(model, lr=1e-3)
        ddp = DistributedTrainer(optim.space)
        losses = []
        for data in _batches(steps):
            my = data[rank:rank + 1]
            loss = compute_loss(model, my)
            loss.backward()
            ddp.finish_backward()
            optim.step()
            optim.zero_grad()
            losses.append(ddp.all_reduce_scalar(loss).item())
        q.put((rank, losses))
    except Exception as e:  # noqa: BLE001
        import traceback
        q.put((rank, f"fail: {e}\n{traceback.format_exc()[-900:]}"))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.parametrize("zero1", [False, True])
@pytest.mark.timeout(420)
def test_dp_world8_matches_single(zero1):
    want = _losses_single()
    world = 8
    port = _free_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    ps = [ctx.Process(target=_worker, args=(r, world, port, q, zero1))
          for r in range(world)]
    for p in ps:
        p.start()
    results = dict(q.get(timeout=360) for _ in range(world))
    for p in ps:
        p.join(timeout=60)
    for rank, got in results.items():
        assert isinstance(got, list), got
        for a, b in zip(got, want):
            assert abs(a - b) < 1e-9, (rank, got, want)
