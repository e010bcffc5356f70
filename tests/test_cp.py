"""Context parallelism (parallel/cp.py) under gloo world_size=2:
sharded-sequence forward/loss/grads must match the unsharded model."""

import multiprocessing as mp
import os
import socket

import pytest
import torch


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _worker(rank, world, port, q):  # noqa: C901
    import torch.distributed as dist
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from progen_amd.config import ProGenConfig
    from progen_amd.models.progen import ProGenBase
    from progen_amd.parallel import cp
    from progen_amd.utils import compute_loss
    try:
        cp.init_context_parallel(world)
        torch.manual_seed(13)
        cfg = ProGenConfig(num_tokens=64, dim=16, depth=3, dim_head=4,
                           heads=2, window_size=8, seq_len=32, ff_glu=True,
                           global_mlp_depth=1)
        model = ProGenBase(cfg).double()
        torch.manual_seed(91)
        data = torch.randint(1, 64, (2, 33))
        data[:, 0] = 0
        data[0, 28:] = 0  # pad tail crossing nothing; EOS on rank 1's shard
        data[1, 12:] = 0  # EOS on rank 0's shard, pads continue into rank 1

        # forward parity on this rank's rows
        ids = data[:, :-1]
        L = ids.shape[1] // world
        with torch.no_grad():
            full_logits = model(ids)
        my_logits = cp.cp_forward(model, ids[:, rank * L:(rank + 1) * L])
        torch.testing.assert_close(
            my_logits, full_logits[:, rank * L:(rank + 1) * L],
            rtol=1e-9, atol=1e-9)

        # loss parity (incl. cross-shard first-pad-as-EOS masking)
        loss_cp = cp.cp_loss(model, data)
        loss_full = compute_loss(model, data)
        torch.testing.assert_close(loss_cp, loss_full, rtol=1e-9, atol=1e-9)

        # grad parity after the CP all-reduce
        model.zero_grad()
        loss_full2 = compute_loss(model, data)
        loss_full2.backward()
        want = {n: p.grad.detach().clone()
                for n, p in model.named_parameters()}
        model.zero_grad()
        cp.cp_loss(model, data).backward()
        cp.cp_sync_grads(model)
        for n, p in model.named_parameters():
            torch.testing.assert_close(p.grad, want[n], rtol=1e-7,
                                       atol=1e-9, msg=lambda m, _n=n: f"{_n}: {m}")
        q.put((rank, "ok"))
    except Exception as e:
        import traceback
        q.put((rank, f"fail: {e}\n{traceback.format_exc()[-1500:]}"))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(180)
@pytest.mark.parametrize("world", [2, 4])
def test_cp_parity(world):
    # world=4 exercises MIDDLE ranks (simultaneous halo send+recv),
    # which world=2 cannot
    port = _free_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    ps = [ctx.Process(target=_worker, args=(r, world, port, q))
          for r in range(world)]
    for p in ps:
        p.start()
    results = [q.get(timeout=150) for _ in range(world)]
    for p in ps:
        p.join(timeout=60)
    assert all(msg == "ok" for _, msg in results), results
