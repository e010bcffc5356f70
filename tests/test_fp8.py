"""fp8 (e4m3) quantization plumbing (ops/fp8.py): round-trip error
bounds, scale mapping, simulated-GEMM error envelope."""

import pytest
import torch

from progen_amd.ops import fp8


def test_roundtrip_error_bound():
    torch.manual_seed(0)
    t = torch.randn(1000) * 3.0
    q, s = fp8.quantize_e4m3(t)
    back = fp8.dequantize(q, s)
    rel = (back - t).abs() / t.abs().clamp_min(1e-6)
    # e4m3 normals: <= 2^-4 relative rounding; subnormals (tiny values
    # relative to amax) can be worse — bound the bulk, not the tail
    assert rel.median() < 0.03
    assert (rel < 0.0625).float().mean() > 0.95


def test_amax_maps_to_max_normal():
    t = torch.tensor([0.5, -7.0, 3.0])
    q, s = fp8.quantize_e4m3(t)
    assert torch.isclose(s, torch.tensor(7.0 / fp8.E4M3_MAX))
    # the amax element hits exactly the top of the representable range
    assert fp8.dequantize(q, s)[1].item() == pytest.approx(-7.0, rel=1e-6)


def test_zero_input():
    q, s = fp8.quantize_e4m3(torch.zeros(8))
    assert s.item() == 1.0
    assert (fp8.dequantize(q, s) == 0).all()


def test_sim_gemm_error_envelope():
    torch.manual_seed(1)
    a = torch.randn(64, 128)
    b = torch.randn(128, 32)
    want = a @ b
    got = fp8.matmul_sim_fp8(a, b)
    rel = (got - want).norm() / want.norm()
    # two e4m3 operands, fp32 accumulate: a few percent end-to-end
    assert rel < 0.06, rel.item()


def test_sim_gemm_scale_invariance():
    # per-tensor scaling must make the quantization error independent of
    # the operands' absolute magnitude
    torch.manual_seed(2)
    a = torch.randn(32, 64)
    b = torch.randn(64, 16)
    r1 = fp8.matmul_sim_fp8(a, b)
    r2 = fp8.matmul_sim_fp8(a * 1000, b * 0.001)
    torch.testing.assert_close(r1, r2, rtol=1e-5, atol=1e-5)


def test_cached_weight_q_version_invalidation():
    """The quantized-weight cache returns the same objects while the
    weight is untouched (grad-accum micros / inference) and requantizes
    after any in-place update (optimizer step through the flat view)."""
    import torch

    from progen_amd.ops.fp8 import _cached_weight_q, dequantize

    w = torch.nn.Parameter(torch.randn(32, 16))
    q1, s1 = _cached_weight_q(w, transposed=False)
    q2, s2 = _cached_weight_q(w, transposed=False)
    assert q1 is q2 and s1 is s2  # cache hit
    qt1, _ = _cached_weight_q(w, transposed=True)
    qt2, _ = _cached_weight_q(w, transposed=True)
    assert qt1 is qt2
    # in-place update through a VIEW (how the flat optimizer writes)
    with torch.no_grad():
        w.view(-1).mul_(2.0)
    q3, s3 = _cached_weight_q(w, transposed=False)
    assert q3 is not q1
    err = (dequantize(q3, s3) - w.detach()).abs().max()
    assert err < 0.07 * w.detach().abs().max()  # e4m3 envelope
