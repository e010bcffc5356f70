"""ZeRO-1 sharded optimizer (parallel/zero1.py) under gloo world_size=2:
DP training with sharded optimizer state must reproduce the single-
process full-batch trajectory exactly (fp64)."""

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
    torch.manual_seed(55)
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
    torch.manual_seed(41)
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
    from progen_amd.parallel.ddp import DistributedTrainer
    from progen_amd.parallel.zero1 import Zero1AdamW
    from progen_amd.utils import compute_loss
    try:
        torch.manual_seed(41)  # identical init across ranks
        model = ProGenBase(_cfg()).double()
        optim = Zero1AdamW(model, lr=1e-3)
        ddp = DistributedTrainer(optim.space)
        losses = []
        for data in _batches(steps):
            my = data[rank * 2:(rank + 1) * 2]  # per-rank shard of the batch
            loss = compute_loss(model, my)
            loss.backward()
            ddp.finish_backward()
            optim.step()
            optim.zero_grad()
            losses.append(ddp.all_reduce_scalar(loss).item())
        # every rank must hold identical full params after the gathers
        csum = optim.space.flat.sum()
        sums = [torch.zeros_like(csum) for _ in range(world)]
        dist.all_gather(sums, csum)
        assert torch.equal(sums[0], sums[1]), sums
        q.put((rank, losses))
    except Exception as e:
        import traceback
        q.put((rank, f"fail: {e}\n{traceback.format_exc()[-1200:]}"))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_zero1_matches_single_process():
    want = _losses_single()
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
            # per-rank loss is the half-batch loss; the all-reduced mean
            # equals the full-batch loss only when both halves weigh
            # equally — ProGen's CE is per-sequence mean then batch mean,
            # so mean-of-half-means == full mean for equal halves
            assert abs(a - b) < 1e-9, (rank, got, want)


def _odd_worker(rank, world, port, q):
    import torch.distributed as dist
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from progen_amd.optim import ProGenAdamW
    from progen_amd.parallel.zero1 import Zero1AdamW
    try:
        torch.manual_seed(17)

        class M(torch.nn.Module):
            def __init__(self):
                super().__init__()
                self.lin = torch.nn.Linear(3, 4)      # 16 params
                self.scale = torch.nn.Parameter(torch.ones(1))  # -> 17 (odd)

            def forward(self, x):
                return self.lin(x) * self.scale

        def run(cls, **kw):
            torch.manual_seed(17)
            m = M().double()
            opt = cls(m, lr=1e-2, **kw)
            torch.manual_seed(5)
            losses = []
            for _ in range(3):
                x = torch.randn(8, 3, dtype=torch.float64)
                loss = (m(x) ** 2).mean()
                loss.backward()
                opt.step()
                opt.zero_grad()
                losses.append(loss.item())
            return losses, opt.space.flat.detach().clone()

        want, want_flat = run(ProGenAdamW)
        got, got_flat = run(Zero1AdamW)
        assert got_flat.numel() == 17  # odd: last shard is smaller
        for a, b in zip(got, want):
            assert abs(a - b) < 1e-12, (got, want)
        torch.testing.assert_close(got_flat, want_flat)
        q.put((rank, "ok"))
    except Exception as e:
        import traceback
        q.put((rank, f"fail: {e}\n{traceback.format_exc()[-1000:]}"))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_zero1_odd_numel_padding():
    world = 2
    port = _free_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    ps = [ctx.Process(target=_odd_worker, args=(r, world, port, q))
          for r in range(world)]
    for p in ps:
        p.start()
    results = [q.get(timeout=90) for _ in range(world)]
    for p in ps:
        p.join(timeout=60)
    assert all(msg == "ok" for _, msg in results), results


def _resume_worker(rank, world, port, q):
    import torch.distributed as dist
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from progen_amd.models.progen import ProGenBase
    from progen_amd.parallel.zero1 import Zero1AdamW
    from progen_amd.utils import compute_loss
    try:
        torch.manual_seed(41)
        model = ProGenBase(_cfg()).double()
        opt = Zero1AdamW(model, lr=1e-3)
        data = _batches(1)[0]
        loss = compute_loss(model, data[rank * 2:(rank + 1) * 2])
        loss.backward()
        opt.step()
        opt.zero_grad()

        # state_dict() is a COLLECTIVE all-gather into the replicated
        # format (the old per-rank format only saved rank 0's
        # shard and could never resume)
        assert opt.state_dict_is_collective
        sd = {k: (v.clone() if torch.is_tensor(v) else v)
              for k, v in opt.state_dict().items()}
        assert sd["master"].numel() == opt.space.numel  # full, not shard
        opt2 = Zero1AdamW(model, lr=1e-3)
        opt2.load_state_dict(sd)
        assert opt2.step_count == 1
        torch.testing.assert_close(opt2.master, opt.master)
        torch.testing.assert_close(opt2.exp_avg, opt.exp_avg)

        # the replicated format also resumes WITHOUT ZeRO-1 (world-size
        # independent): full master must match the gathered shards
        from progen_amd.optim import ProGenAdamW
        opt3 = ProGenAdamW(model, lr=1e-3)
        opt3.load_state_dict(sd)
        torch.testing.assert_close(
            opt3.master[opt.lo:opt.hi], opt.master)

        # the removed round-1 per-rank format is rejected with a clear error
        bad = dict(sd)
        bad["zero1"] = {"world": world, "rank": rank}
        try:
            opt2.load_state_dict(bad)
            q.put((rank, "fail: no error raised"))
            return
        except ValueError as e:
            assert "ZeRO-1" in str(e)
        q.put((rank, "ok"))
    except Exception as e:
        import traceback
        q.put((rank, f"fail: {e}\n{traceback.format_exc()[-1000:]}"))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_zero1_state_roundtrip_and_guard():
    world = 2
    port = _free_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    ps = [ctx.Process(target=_resume_worker, args=(r, world, port, q))
          for r in range(world)]
    for p in ps:
        p.start()
    results = [q.get(timeout=90) for _ in range(world)]
    for p in ps:
        p.join(timeout=60)
    assert all(msg == "ok" for _, msg in results), results


def _skip_worker(rank, world, port, q):
    import torch.distributed as dist
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from progen_amd.parallel.zero1 import Zero1AdamW
    try:
        torch.manual_seed(17)
        m = torch.nn.Linear(3, 4).double()
        opt = Zero1AdamW(m, lr=1e-2, max_grad_norm=0.5)
        x = torch.randn(8, 3, dtype=torch.float64,
                        generator=torch.Generator().manual_seed(5))
        (m(x) ** 2).mean().backward()
        opt.space.flat_grad[1] = float("inf")
        before = opt.space.flat.detach().clone()
        opt.step()  # must skip on every rank together (full-grad norm)
        assert torch.equal(opt.space.flat, before)
        assert opt.exp_avg.abs().sum().item() == 0.0
        # healthy step afterwards applies
        opt.zero_grad()
        (m(x) ** 2).mean().backward()
        opt.step()
        assert not torch.equal(opt.space.flat, before)
        assert torch.isfinite(opt.space.flat).all()
        q.put((rank, "ok"))
    except Exception as e:
        import traceback
        q.put((rank, f"fail: {e}\n{traceback.format_exc()[-1000:]}"))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_zero1_nonfinite_grad_step_is_skipped():
    """Parity with the base optimizer's GradScaler-style skip: inf grads
    leave params/moments untouched on every rank (the clip coef comes
    from the full-grad norm, so the skip decision is rank-consistent)."""
    world = 2
    port = _free_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    ps = [ctx.Process(target=_skip_worker, args=(r, world, port, q))
          for r in range(world)]
    for p in ps:
        p.start()
    results = [q.get(timeout=90) for _ in range(world)]
    for p in ps:
        p.join(timeout=60)
    assert all(msg == "ok" for _, msg in results), results
