"""Model-level tests: shapes, causality, layer schedule, parity API."""

import numpy as np
import pytest
import torch

from progen_amd import ProGen, ProGenBase, ProGenConfig

TINY = dict(num_tokens=32, dim=16, seq_len=32, depth=3, window_size=8,
            global_mlp_depth=1, heads=2, dim_head=8, ff_mult=2)


def make_tiny():
    return ProGenBase(ProGenConfig(**TINY))


def test_forward_shapes_batched_and_unbatched():
    m = make_tiny()
    x = torch.randint(0, 32, (2, 32))
    out = m(x)
    assert out.shape == (2, 32, 32)
    out1 = m(x[0])
    assert out1.shape == (1, 32, 32)
    np.testing.assert_allclose(out1[0].detach().numpy(),
                               out[0].detach().numpy(), atol=1e-5)


def test_model_is_causal():
    """logits[i] must not depend on tokens at positions > i."""
    m = make_tiny().double()
    x = torch.randint(1, 32, (1, 32))
    base = m(x).detach()
    p = 13
    x2 = x.clone()
    x2[0, p] = (x[0, p] + 5) % 32
    out = m(x2).detach()
    np.testing.assert_allclose(base[0, :p].numpy(), out[0, :p].numpy(), atol=1e-12)
    assert not np.allclose(base[0, p:].numpy(), out[0, p:].numpy())


def test_layer_schedule_sgu_last_n():
    """Last global_mlp_depth layers use SGU, others GLU
    (reference: progen.py:211-212)."""
    cfg = ProGenConfig(**{**TINY, "depth": 4, "global_mlp_depth": 2})
    m = ProGenBase(cfg)
    kinds = [(ff.sgu is not None, ff.glu) for _, ff in m.layers]
    assert kinds == [(False, True), (False, True), (True, False), (True, False)]


def test_sgu_ff_hidden_dims():
    cfg = ProGenConfig(**TINY)
    m = ProGenBase(cfg)
    # GLU layer: proj_in doubles hidden (progen.py:119-120)
    ff_glu = m.layers[0][1]
    assert ff_glu.proj_in.out_features == cfg.dim * cfg.ff_mult * 2
    assert ff_glu.proj_out.in_features == cfg.dim * cfg.ff_mult
    # SGU layer: hidden not doubled; SGU halves it
    ff_sgu = m.layers[-1][1]
    assert ff_sgu.proj_in.out_features == cfg.dim * cfg.ff_mult
    assert ff_sgu.sgu.proj_out.in_features == cfg.dim * cfg.ff_mult // 2


def test_sgu_param_init():
    cfg = ProGenConfig(**TINY)
    m = ProGenBase(cfg)
    sgu = m.layers[-1][1].sgu
    n = cfg.seq_len
    eps = 1e-3
    assert sgu.spatial_weights.shape == (n, n)
    assert sgu.spatial_weights.abs().max().item() <= eps / n + 1e-9
    assert torch.all(sgu.spatial_biases == 1.0)


def test_progen_factory_init_apply_parity():
    """Reference API: model = ProGen(...); params = model.init(rng, seq);
    logits = model.apply(params, rng, seq) (reference: progen.py:235-243,
    README.md:29-51)."""
    model = ProGen(**TINY)
    seq = np.random.randint(0, 32, (32,))
    params = model.init(42, seq)
    logits = model.apply(params, 42, seq)
    assert logits.shape == (32, 32)
    # apply is deterministic given params
    logits2 = model.apply(params, 7, seq)
    np.testing.assert_allclose(logits.numpy(), logits2.numpy(), atol=1e-6)
    # dead kwargs accepted (progen.py:201-202)
    ProGen(**TINY, attn_dim=64, clamp_gate=True, mixed_precision=True,
           mixed_precision_policy=dict(params="float32", compute="float16",
                                       output="float32"))


def test_seq_len_window_divisibility_assert():
    with pytest.raises(ValueError):
        ProGenConfig(**{**TINY, "seq_len": 33})


def test_num_params_plausible():
    m = make_tiny()
    n = m.num_params()
    assert n == sum(p.numel() for p in m.parameters())
    assert n > 0


def test_loss_backward_runs():
    from progen_amd.utils import compute_loss
    m = make_tiny()
    data = torch.randint(0, 32, (2, 33))
    loss = compute_loss(m, data)
    loss.backward()
    grads = [p.grad for p in m.parameters()]
    assert all(g is not None for g in grads)
    assert all(torch.isfinite(g).all() for g in grads)
