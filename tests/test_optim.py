"""Optimizer tests: flat-space AdamW vs torch.optim.AdamW oracle, clip
math, apply_every parity semantics."""

import copy

import numpy as np
import pytest
import torch

from progen_amd import ProGenBase, ProGenConfig
from progen_amd.optim import ProGenAdamW
from progen_amd.utils import compute_loss

TINY = dict(num_tokens=32, dim=16, seq_len=32, depth=2, window_size=8,
            global_mlp_depth=1, heads=2, dim_head=8)


def _models():
    torch.manual_seed(0)
    m1 = ProGenBase(ProGenConfig(**TINY))
    m2 = copy.deepcopy(m1)
    return m1, m2


def test_flat_space_preserves_params_and_grads():
    m1, m2 = _models()
    opt = ProGenAdamW(m1, max_grad_norm=None)
    # re-homing must not change values
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        np.testing.assert_array_equal(p1.detach().numpy(), p2.detach().numpy())
    data = torch.randint(0, 32, (2, 33))
    compute_loss(m1, data).backward()
    compute_loss(m2, data).backward()
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        np.testing.assert_allclose(p1.grad.numpy(), p2.grad.numpy(),
                                   atol=1e-6)


def test_adamw_matches_torch_oracle():
    m1, m2 = _models()
    lr, wd = 1e-3, 1e-2
    opt = ProGenAdamW(m1, lr=lr, weight_decay=wd, max_grad_norm=None)
    decay = [p for p in m2.parameters() if p.dim() > 1]
    nodecay = [p for p in m2.parameters() if p.dim() <= 1]
    oracle = torch.optim.AdamW([
        {"params": decay, "weight_decay": wd},
        {"params": nodecay, "weight_decay": 0.0},
    ], lr=lr, betas=(0.9, 0.999), eps=1e-8)

    for i in range(3):
        torch.manual_seed(100 + i)
        data = torch.randint(0, 32, (2, 33))
        opt.zero_grad()
        compute_loss(m1, data).backward()
        opt.step()

        oracle.zero_grad()
        compute_loss(m2, data).backward()
        oracle.step()

    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        np.testing.assert_allclose(p1.detach().numpy(), p2.detach().numpy(),
                                   atol=1e-6)


def test_clip_by_global_norm_optax_semantics():
    m1, _ = _models()
    opt = ProGenAdamW(m1, max_grad_norm=0.5)
    opt.space.flat_grad.normal_(0, 10.0)
    g32 = opt.space.flat_grad.float()
    coef = opt._clip_coef(g32).item()
    norm = g32.norm().item()
    assert coef == pytest.approx(0.5 / norm, rel=1e-6)
    # below the threshold: no scaling
    opt.space.flat_grad.zero_()
    opt.space.flat_grad[0] = 0.1
    coef = opt._clip_coef(opt.space.flat_grad.float()).item()
    assert coef == pytest.approx(1.0)


def test_accum_sum_equals_mean_of_grads():
    """k micro-batches with accum_mode=sum == one step on mean gradient."""
    m1, m2 = _models()
    opt1 = ProGenAdamW(m1, max_grad_norm=None, grad_accum_every=2)
    opt2 = ProGenAdamW(m2, max_grad_norm=None, grad_accum_every=1)

    torch.manual_seed(1)
    d1 = torch.randint(0, 32, (2, 33))
    d2 = torch.randint(0, 32, (2, 33))

    compute_loss(m1, d1).backward()
    assert not opt1.micro_step()
    compute_loss(m1, d2).backward()
    assert opt1.micro_step()

    # oracle: mean grad, single step
    (0.5 * (compute_loss(m2, d1) + compute_loss(m2, d2))).backward()
    opt2.micro_step()

    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        np.testing.assert_allclose(p1.detach().numpy(), p2.detach().numpy(),
                                   atol=1e-6)


def test_apply_every_mode_advances_moments_per_micro():
    """Reference quirk (train.py:117-121): Adam moments advance per
    micro-batch; params move only every k micro-batches."""
    m1, _ = _models()
    opt = ProGenAdamW(m1, max_grad_norm=None, grad_accum_every=2,
                      accum_mode="apply_every")
    p0 = opt.master.clone()

    compute_loss(m1, torch.randint(0, 32, (2, 33))).backward()
    applied = opt.micro_step()
    assert not applied
    assert torch.equal(opt.master, p0)          # params unchanged
    assert opt.exp_avg.abs().sum() > 0          # moments advanced
    assert opt.step_count == 1

    compute_loss(m1, torch.randint(0, 32, (2, 33))).backward()
    assert opt.micro_step()
    assert not torch.equal(opt.master, p0)
    assert opt.step_count == 2


def test_state_dict_roundtrip():
    m1, _ = _models()
    opt = ProGenAdamW(m1)
    compute_loss(m1, torch.randint(0, 32, (2, 33))).backward()
    opt.micro_step()
    sd = {k: (v.clone() if torch.is_tensor(v) else v)
          for k, v in opt.state_dict().items()}

    m2 = ProGenBase(ProGenConfig(**TINY))
    opt2 = ProGenAdamW(m2)
    opt2.load_state_dict(sd)
    np.testing.assert_allclose(opt2.master.numpy(), opt.master.numpy())
    np.testing.assert_allclose(opt2.exp_avg.numpy(), opt.exp_avg.numpy())
    assert opt2.step_count == opt.step_count


def test_nonfinite_grad_step_is_skipped():
    """GradScaler semantics: inf/NaN grads -> the whole update is skipped
    (no master/moment/param change, no step-count advance) instead of
    poisoning the weights with inf*0 = NaN. Parity with the fused
    kernel's step_ok guard (ops/hip/adamw.hip)."""
    m1, _ = _models()
    opt = ProGenAdamW(m1, lr=1e-3, max_grad_norm=0.5)
    data = torch.randint(0, 32, (2, 33))
    compute_loss(m1, data).backward()
    opt.space.flat_grad[7] = float("inf")
    before = opt.master.clone()
    opt.step()
    assert opt.step_count == 0
    np.testing.assert_array_equal(opt.master.numpy(), before.numpy())
    assert torch.isfinite(opt.master).all()
    assert opt.exp_avg.abs().sum().item() == 0.0
    # NaN grads skip too
    opt.space.flat_grad[7] = float("nan")
    opt.step()
    np.testing.assert_array_equal(opt.master.numpy(), before.numpy())
    # and a healthy step afterwards applies normally
    opt.zero_grad()
    compute_loss(m1, data).backward()
    opt.step()
    assert opt.step_count == 1
    assert not torch.equal(opt.master, before)
    assert torch.isfinite(opt.master).all()
