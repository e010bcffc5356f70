"""TP-sharded ProGen (parallel/tp_model.py) under gloo world_size=2:
forward logits and backward gradients must match the unsharded model."""

import multiprocessing as mp
import os
import socket

import pytest
import torch


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _worker(rank, world, port, q):
    import copy

    import torch.distributed as dist
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from progen_amd.config import ProGenConfig
    from progen_amd.models.progen import ProGenBase
    from progen_amd.parallel import tp, tp_model
    from progen_amd.utils import compute_loss
    try:
        tp.init_tensor_parallel(world)
        torch.manual_seed(21)  # identical full weights on both ranks
        cfg = ProGenConfig(num_tokens=64, dim=16, depth=3, dim_head=4,
                           heads=4, window_size=8, seq_len=32, ff_glu=True,
                           global_mlp_depth=1)
        full = ProGenBase(cfg).double()
        sharded = tp_model.tp_shard_(copy.deepcopy(full))

        torch.manual_seed(99)
        data = torch.randint(1, 64, (2, 33))
        data[:, 0] = 0

        # forward parity
        x = data[:, :-1]
        logits_full = full(x)
        logits_tp = sharded(x)
        torch.testing.assert_close(logits_tp, logits_full,
                                   rtol=1e-9, atol=1e-9)

        # backward parity
        compute_loss(full, data).backward()
        compute_loss(sharded, data).backward()
        tp_model.sync_replicated_grads(sharded)

        # replicated params: embed, logits head, attn LN, SGU spatial
        torch.testing.assert_close(sharded.embed.weight.grad,
                                   full.embed.weight.grad,
                                   rtol=1e-8, atol=1e-10)
        torch.testing.assert_close(sharded.to_logits.weight.grad,
                                   full.to_logits.weight.grad,
                                   rtol=1e-8, atol=1e-10)
        a_tp, f_tp = sharded.layers[0]
        a_f, f_f = full.layers[0]
        torch.testing.assert_close(a_tp.norm_weight.grad,
                                   a_f.norm_weight.grad,
                                   rtol=1e-8, atol=1e-10)
        sgu_tp = sharded.layers[-1][1].sgu
        sgu_f = full.layers[-1][1].sgu
        torch.testing.assert_close(sgu_tp.spatial_weights.grad,
                                   sgu_f.spatial_weights.grad,
                                   rtol=1e-8, atol=1e-10)

        # sharded params: this rank's rows/cols of the full gradient
        rows = a_tp.to_qkv._rows()
        torch.testing.assert_close(a_tp.to_qkv.weight.grad,
                                   a_f.to_qkv.weight.grad[rows],
                                   rtol=1e-8, atol=1e-10)
        cols = a_tp.to_out._cols()
        torch.testing.assert_close(a_tp.to_out.weight.grad,
                                   a_f.to_out.weight.grad[:, cols],
                                   rtol=1e-8, atol=1e-10)
        q.put((rank, "ok"))
    except Exception as e:  # surface assertion details to the parent
        import traceback
        q.put((rank, f"fail: {e}\n{traceback.format_exc()[-1500:]}"))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_tp_model_parity_world2():
    world = 2
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
