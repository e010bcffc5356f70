"""CP x DP 2D mesh (gloo world_size=4, cp=2 x dp=2): sequence-sharded
replicas + DP gradient averaging must reproduce the single-process
full-batch trajectory exactly."""

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


def _batches(steps):
    torch.manual_seed(72)
    out = []
    for _ in range(steps):
        d = torch.randint(1, 64, (4, 33))
        d[:, 0] = 0
        d[1, 20:] = 0  # a pad tail crossing the CP shard boundary
        out.append(d)
    return out


def _losses_single(steps=3):
    from progen_amd.models.progen import ProGenBase
    from progen_amd.optim import ProGenAdamW
    from progen_amd.utils import compute_loss
    torch.manual_seed(61)
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


def _worker(rank, world, port, q, steps=3):
    import torch.distributed as dist
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from progen_amd.models.progen import ProGenBase
    from progen_amd.optim import ProGenAdamW
    from progen_amd.parallel import cp
    from progen_amd.parallel.ddp import DistributedTrainer
    try:
        CP = 2
        cp.init_context_parallel(CP)
        dp_rank = rank // CP
        torch.manual_seed(61)
        model = ProGenBase(_cfg()).double()
        optim = ProGenAdamW(model, lr=1e-3)
        ddp = DistributedTrainer(optim.space, group=cp.dp_group())
        assert ddp.world == 2
        losses = []
        for data in _batches(steps):
            my = data[dp_rank * 2:(dp_rank + 1) * 2]  # replica batch shard
            with ddp.no_sync():   # CP grads are partial until cp_sync;
                loss = cp.cp_loss(model, my)          # reduce manually
                loss.backward()
            cp.cp_sync_grads(model)                   # sum over sequence
            ddp.finish_backward()                     # mean over replicas
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
def test_mesh_cp2_dp2_matches_single():
    want = _losses_single()
    assert want[0] != want[-1]

    world = 4
    port = _free_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    ps = [ctx.Process(target=_worker, args=(r, world, port, q))
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
