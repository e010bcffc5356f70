"""Tensor-parallel linears (progen_amd/parallel/tp.py) under gloo
world_size=2: forward/backward parity against the unsharded layer."""

import multiprocessing as mp
import os
import socket

import pytest
import torch


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _worker(rank, world, port, q, case):
    import torch.distributed as dist
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from progen_amd.parallel import tp
    tp.init_tensor_parallel(world)
    torch.manual_seed(123)  # identical on all ranks

    B, N, D, H = 2, 4, 16, 24
    x = torch.randn(B, N, D, dtype=torch.float64).requires_grad_()
    if case == "column":
        full = torch.nn.Linear(D, H, bias=True).double()
        torch.manual_seed(5)
        col = tp.ColumnParallelLinear(D, H, bias=True, gather_output=True)
        col = col.double()
        col.shard_from(full.weight.detach(), full.bias.detach())
        y = col(x)
        want = full(x.detach().clone().requires_grad_())
        torch.testing.assert_close(y, want)
        # backward: dx must equal the unsharded layer's
        g = torch.randn_like(y)
        y.backward(g)
        xw = x.detach().clone().requires_grad_()
        full(xw).backward(g)
        torch.testing.assert_close(x.grad, xw.grad)
        # weight grad parity on this rank's rows
        rows = col._rows()
        torch.testing.assert_close(col.weight.grad,
                                   _full_wgrad(full, x.detach(), g)[rows])
    elif case == "row":
        full = torch.nn.Linear(H, D, bias=True).double()
        row = tp.RowParallelLinear(H, D, bias=True).double()
        row.shard_from(full.weight.detach(), full.bias.detach())
        xin = torch.randn(B, N, H, dtype=torch.float64)
        # row-parallel input is feature-sharded
        cols = row._cols()
        xl = xin[..., cols].clone().requires_grad_()
        y = row(xl)
        torch.testing.assert_close(y, full(xin))
        g = torch.randn_like(y)
        y.backward(g)
        xw = xin.clone().requires_grad_()
        full(xw).backward(g)
        torch.testing.assert_close(xl.grad, xw.grad[..., cols])
    elif case == "glu_pairing":
        # GLU-aware column shard: chunk(2) of the local output must pair
        # the SAME channels the full layer's chunk(2) pairs
        full = torch.nn.Linear(D, H, bias=False).double()
        col = tp.ColumnParallelLinear(D, H, bias=False, shard_glu=True).double()
        col.shard_from(full.weight.detach())
        y_local = col(x.detach())                      # (B, N, H/world)
        a_l, g_l = y_local.chunk(2, dim=-1)
        a_f, g_f = full(x.detach()).chunk(2, dim=-1)
        per = (H // 2) // world
        sl = slice(rank * per, (rank + 1) * per)
        torch.testing.assert_close(a_l, a_f[..., sl])
        torch.testing.assert_close(g_l, g_f[..., sl])
    q.put((rank, "ok"))
    dist.barrier()
    dist.destroy_process_group()


def _full_wgrad(full, x, g):
    return torch.einsum("bno,bni->oi", g, x)


@pytest.mark.parametrize("case", ["column", "row", "glu_pairing"])
@pytest.mark.timeout(120)
def test_tp_linear_parity(case):
    world = 2
    port = _free_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    ps = [ctx.Process(target=_worker, args=(r, world, port, q, case))
          for r in range(world)]
    for p in ps:
        p.start()
    results = [q.get(timeout=90) for _ in range(world)]
    for p in ps:
        p.join(timeout=60)
    assert all(msg == "ok" for _, msg in results), results
    assert all(p.exitcode == 0 for p in ps)
