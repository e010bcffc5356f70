"""End-to-end TP training under gloo world_size=2: the sharded model +
flat-space ProGenAdamW + replicated-grad sync must reproduce the
unsharded single-process loss trajectory exactly (fp64)."""

import multiprocessing as mp
import os
import socket

import pytest
import torch


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _losses_full(steps=3):
    from progen_amd.config import ProGenConfig
    from progen_amd.models.progen import ProGenBase
    from progen_amd.optim import ProGenAdamW
    from progen_amd.utils import compute_loss
    torch.manual_seed(31)
    cfg = ProGenConfig(num_tokens=64, dim=16, depth=2, dim_head=4,
                       heads=4, window_size=8, seq_len=32, ff_glu=True,
                       global_mlp_depth=1)
    model = ProGenBase(cfg).double()
    optim = ProGenAdamW(model, lr=1e-3)
    torch.manual_seed(77)
    losses = []
    for _ in range(steps):
        data = torch.randint(1, 64, (2, 33))
        data[:, 0] = 0
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
    from progen_amd.config import ProGenConfig
    from progen_amd.models.progen import ProGenBase
    from progen_amd.optim import ProGenAdamW
    from progen_amd.parallel import tp, tp_model
    from progen_amd.utils import compute_loss
    try:
        tp.init_tensor_parallel(world)
        torch.manual_seed(31)  # identical full init on both ranks
        cfg = ProGenConfig(num_tokens=64, dim=16, depth=2, dim_head=4,
                           heads=4, window_size=8, seq_len=32, ff_glu=True,
                           global_mlp_depth=1)
        model = tp_model.tp_shard_(ProGenBase(cfg).double())
        optim = ProGenAdamW(model, lr=1e-3)
        optim.norm_sumsq_fn = tp_model.tp_grad_sumsq_fn(model)
        torch.manual_seed(77)  # identical batches (TP ranks share data)
        losses = []
        for _ in range(steps):
            data = torch.randint(1, 64, (2, 33))
            data[:, 0] = 0
            loss = compute_loss(model, data)
            loss.backward()
            tp_model.sync_replicated_grads(model)
            optim.step()
            optim.zero_grad()
            losses.append(loss.item())
        q.put((rank, losses))
    except Exception as e:
        import traceback
        q.put((rank, f"fail: {e}\n{traceback.format_exc()[-1200:]}"))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_tp_training_matches_full():
    want = _losses_full()
    # sanity: training actually moves
    assert want[0] != want[-1]

    world = 2
    port = _free_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    ps = [ctx.Process(target=_worker, args=(r, world, port, q))
          for r in range(world)]
    for p in ps:
        p.start()
    results = dict(q.get(timeout=150) for _ in range(world))
    for p in ps:
        p.join(timeout=60)
    for rank, got in results.items():
        assert isinstance(got, list), got
        for a, b in zip(got, want):
            assert abs(a - b) < 1e-9, (rank, got, want)
