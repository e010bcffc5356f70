"""TP x DP 2D mesh (gloo world_size=4, tp=2 x dp=2): TP-sharded
replicas + bucketed DP gradient averaging + replicated-param TP sync
must reproduce the single-process full-batch trajectory exactly."""

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
    return ProGenConfig(num_tokens=64, dim=16, depth=3, dim_head=4,
                        heads=4, window_size=8, seq_len=32, ff_glu=True,
                        global_mlp_depth=1)


def _batches(steps):
    torch.manual_seed(66)
    out = []
    for _ in range(steps):
        d = torch.randint(1, 64, (4, 33))
        d[:, 0] = 0
        out.append(d)
    return out


def _losses_single(steps=3):
    from progen_amd.models.progen import ProGenBase
    from progen_amd.optim import ProGenAdamW
    from progen_amd.utils import compute_loss
    torch.manual_seed(51)
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


def _worker(rank, world, port, q, steps=3, zero1=False):
    import torch.distributed as dist
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from progen_amd.models.progen import ProGenBase
    from progen_amd.optim import ProGenAdamW
    from progen_amd.parallel import tp, tp_model
    from progen_amd.parallel.ddp import DistributedTrainer
    from progen_amd.utils import compute_loss
    try:
        TP = 2
        tp.init_tensor_parallel(TP)
        dp_rank = rank // TP  # contiguous TP groups -> replica index
        torch.manual_seed(51)  # identical full init everywhere
        model = tp_model.tp_shard_(ProGenBase(_cfg()).double())
        if zero1:
            from progen_amd.parallel.zero1 import Zero1AdamW
            optim = Zero1AdamW(model, lr=1e-3, group=tp.dp_group())
            assert optim.world == 2  # shards over replicas only
        else:
            optim = ProGenAdamW(model, lr=1e-3)
        optim.norm_sumsq_fn = tp_model.tp_grad_sumsq_fn(model)
        ddp = DistributedTrainer(optim.space, group=tp.dp_group())
        assert ddp.world == 2  # reduces over replicas, not the world
        losses = []
        for data in _batches(steps):
            my = data[dp_rank * 2:(dp_rank + 1) * 2]  # replica batch shard
            loss = compute_loss(model, my)
            loss.backward()
            ddp.finish_backward()               # DP mean over replicas
            tp_model.sync_replicated_grads(model)  # TP sum (spatial W/b)
            optim.step()
            optim.zero_grad()
            losses.append(ddp.all_reduce_scalar(loss).item())
        q.put((rank, losses))
    except Exception as e:
        import traceback
        q.put((rank, f"fail: {e}\n{traceback.format_exc()[-1200:]}"))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(240)
@pytest.mark.parametrize("zero1", [False, True])
def test_mesh_tp2_dp2_matches_single(zero1):
    # zero1=True additionally shards the optimizer state over the DP
    # axis of the mesh (TP=2 x DP=2 x ZeRO-1)
    want = _losses_single()
    assert want[0] != want[-1]

    world = 4
    port = _free_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    ps = [ctx.Process(target=_worker, args=(r, world, port, q, 3, zero1))
          for r in range(world)]
    for p in ps:
        p.start()
    results = dict(q.get(timeout=200) for _ in range(world))
    for p in ps:
        p.join(timeout=60)
    for rank, got in results.items():
        assert isinstance(got, list), got
        for a, b in zip(got, want):
            assert abs(a - b) < 1e-9, (rank, got, want)
