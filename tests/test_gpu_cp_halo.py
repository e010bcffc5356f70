"""GPU: context-parallel halo path of the fused attention kernels
(VERDICT r1 item 8). Two 'virtual ranks' in one process: the sequence is
split in half, the second half's window-0 lookback comes from the first
half's last window of ROTATED [k|v] via the kernel's halo argument, and
the composite must reproduce the single full-sequence kernel run exactly
— forward rows AND the gradient that flows through the halo back into
the first half (chain rule through dhalo)."""

import pytest
import torch

pytestmark = [pytest.mark.gpu,
              pytest.mark.skipif(not torch.cuda.is_available(),
                                 reason="needs MI355X")]

B, N, H, DH, WSZ = 2, 512, 4, 64, 256


def _mk_qkv(seed):
    torch.manual_seed(seed)
    return (torch.randn(B, N, 3 * H * DH, device="cuda") * 0.25) \
        .to(torch.bfloat16)


def _tables():
    from progen_amd.ops import reference as R
    sin, cos = R.fixed_pos_embedding(N, DH, device="cuda")
    return sin, cos


def _rotated_kv_halo(qkv_rows, sin_rows, cos_rows):
    """Differentiably build the (B, wsz, 2*H*DH) rotated [k|v] band from
    the LAST wsz rows of a (B, L, 3*H*DH) qkv slice — what a CP rank
    sends its successor (torch ops, so autograd carries dhalo back)."""
    from progen_amd.ops import reference as R
    L = qkv_rows.shape[1]
    tail = qkv_rows[:, L - WSZ:]
    k = tail[..., H * DH:2 * H * DH].view(B, WSZ, H, DH)
    v = tail[..., 2 * H * DH:].view(B, WSZ, H, DH)
    s = sin_rows[L - WSZ:].view(WSZ, 1, DH)
    c = cos_rows[L - WSZ:].view(WSZ, 1, DH)
    k = (k.float() * c + R.rotate_every_two(k.float()) * s)
    v = (v.float() * c + R.rotate_every_two(v.float()) * s)
    return torch.cat((k, v), dim=2).reshape(B, WSZ, 2 * H * DH) \
        .to(qkv_rows.dtype)


def test_halo_matches_full_run_fwd_bwd():
    from progen_amd.ops import functional as OF
    sin, cos = _tables()
    qkv0 = _mk_qkv(5)

    # full-sequence reference run (kernel path, no halo)
    qkv_full = qkv0.clone().requires_grad_(True)
    out_full = OF.local_attention(qkv_full, sin, cos, H, WSZ)
    gout = torch.randn_like(out_full) * 0.1
    out_full.backward(gout)

    # two virtual CP ranks
    qkv_cp = qkv0.clone().requires_grad_(True)
    first, second = qkv_cp[:, :N // 2], qkv_cp[:, N // 2:]
    out_a = OF.local_attention(first.contiguous(), sin[:N // 2],
                               cos[:N // 2], H, WSZ)
    halo = _rotated_kv_halo(first, sin[:N // 2], cos[:N // 2])
    out_b = OF.local_attention(second.contiguous(), sin[N // 2:],
                               cos[N // 2:], H, WSZ, halo=halo.contiguous())
    out_cp = torch.cat((out_a, out_b), dim=1)
    out_cp.backward(gout)

    # the halo is rotated by torch mul/add while the kernel's rope may
    # contract to fma: one-ULP bf16 differences are possible, so the
    # comparison is tight-but-not-bitwise
    torch.testing.assert_close(out_cp.float(), out_full.float(),
                               rtol=2e-2, atol=3e-2)
    # gradient equality: the first half's grad includes the dhalo path
    # (its last window's k/v feed the second half's window 0)
    torch.testing.assert_close(qkv_cp.grad.float(), qkv_full.grad.float(),
                               rtol=1e-2, atol=1e-2)
    # the halo-fed rows specifically must carry gradient
    gk = qkv_cp.grad[:, N // 2 - WSZ:N // 2, H * DH:]
    assert gk.float().abs().sum() > 0


def test_no_halo_still_zero_quirk():
    """halo=None keeps the reference's window-0 zero-pad quirk."""
    from progen_amd.ops import functional as OF
    sin, cos = _tables()
    qkv = _mk_qkv(6)
    out = OF.local_attention(qkv, sin, cos, H, WSZ)
    zero_halo = torch.zeros(B, WSZ, 2 * H * DH, device="cuda",
                            dtype=torch.bfloat16)
    out2 = OF.local_attention(qkv, sin, cos, H, WSZ, halo=zero_halo)
    torch.testing.assert_close(out, out2, rtol=0, atol=0)
