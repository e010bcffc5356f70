"""fp8 (OCP e4m3) quantization groundwork for the ProGen-6B TP=8 config
(BASELINE.json config #5; plan: docs/tp_design.md "fp8 path").

MI355X doubles MFMA throughput at fp8 (≈5 PFLOP/s dense vs ≈2.5 bf16),
so the 6B projections move to e4m3 GEMMs with per-tensor scales and fp32
accumulation. This module holds both halves: the dtype plumbing —
amax-based scaling, quantize/dequantize, a simulated-fp8 matmul the CPU
tests pin the error envelope against — and (round 2) the REAL GEMM path:
hipBLASLt e4m3 via torch._scaled_mm with fused quantize kernels
(ops/hip/fp8_quant.hip), measured 1.77-2.06x bf16 on the 1.2B/6B
projection shapes (profiles/r02_fp8_and_decode.md; end-to-end wiring is
PROGEN_FP8=1, currently off by default — measured negative, see TODO.md).

E4M3 facts used here (OCP FP8, the `torch.float8_e4m3fn` variant):
max normal 448, no inf (S.1111.111 is NaN), ~3-bit mantissa → worst-case
relative rounding error 2^-4 ≈ 6.25% for normals.
"""

from __future__ import annotations

from typing import Tuple

import torch

E4M3_MAX = 448.0


def amax_scale(t: torch.Tensor, margin: float = 1.0) -> torch.Tensor:
    """Per-tensor scale s such that (t / s) fits e4m3: s = amax / (448 /
    margin). Returns a 0-dim fp32 tensor; 1.0 for an all-zero input."""
    amax = t.detach().abs().amax().float()
    s = amax * (margin / E4M3_MAX)
    return torch.where(amax > 0, s, torch.ones_like(s))


def _ext_or_none():
    try:
        from progen_amd import _C  # noqa: WPS433
        return _C
    except Exception:  # noqa: BLE001
        return None


def quantize_e4m3(t: torch.Tensor,
                  scale: torch.Tensor = None) -> Tuple[torch.Tensor, torch.Tensor]:
    """t -> (e4m3 tensor, fp32 scale) with t ≈ fp8 * scale. On GPU the
    scale+cast is ONE fused kernel pass (ops/hip/fp8_quant.hip —
    v_cvt_pk_fp8_f32 saturates, so no separate clamp; the eager chain's
    fp32 intermediate copy is gone)."""
    if scale is None:
        scale = amax_scale(t)
    C = _ext_or_none() if (t.is_cuda and t.dtype == torch.bfloat16
                           and t.numel() % 8 == 0) else None
    if C is not None:
        return C.fp8_quantize(t.contiguous(), scale.reshape(1)), scale
    q = (t.float() / scale).clamp(-E4M3_MAX, E4M3_MAX)
    return q.to(torch.float8_e4m3fn), scale


def dequantize(q: torch.Tensor, scale: torch.Tensor,
               dtype: torch.dtype = torch.float32) -> torch.Tensor:
    return (q.float() * scale).to(dtype)


def matmul_sim_fp8(a: torch.Tensor, b: torch.Tensor,
                   out_dtype: torch.dtype = torch.float32) -> torch.Tensor:
    """Simulated fp8 GEMM: quantize both operands to e4m3 (per-tensor
    amax scales), multiply in fp32, rescale. Numerically equivalent to a
    hardware fp8 GEMM with fp32 accumulate (the hardware accumulator IS
    fp32), so CPU tests against this pin the real path's error envelope."""
    qa, sa = quantize_e4m3(a)
    qb, sb = quantize_e4m3(b)
    acc = qa.float() @ qb.float()
    return (acc * (sa * sb)).to(out_dtype)


# ---------------------------------------------------------------------------
# real fp8 GEMM path (round 2): hipBLASLt e4m3 via torch._scaled_mm
# ---------------------------------------------------------------------------
#
# MI355X fp8 MFMA is 2x the bf16 rate (~5 PF/s dense). The projections
# route through torch._scaled_mm, which on ROCm lowers to hipBLASLt
# fp8 GEMMs with fp32 accumulation:
#   forward: y = x8 @ W8^T        (x row-major, W (N,K) row-major, so
#                                  W.t() is naturally column-major as
#                                  _scaled_mm requires)
#   dgrad:   dx = dy8 @ W8        (needs a column-major W copy)
#   wgrad:   bf16 (dW feeds the fp32 master update; keeping it bf16
#            avoids the e5m2-vs-e4m3 gradient-range question — measured
#            fwd+dgrad are where the 2x rate pays; TODO: fp8 wgrad)
# Scaling is per-tensor dynamic amax (device-side, graph-capturable).
# Enable with PROGEN_FP8=1 (train.py/bench.py); only GEMMs with every
# dim >= FP8_MIN_DIM route (the V=256 head and tiny test models stay
# bf16).

import os

FP8_MIN_DIM = 1024
ENABLED = os.environ.get("PROGEN_FP8", "0") == "1"


def scaled_mm(a: torch.Tensor, b_colmajor: torch.Tensor,
              out_dtype: torch.dtype = torch.bfloat16,
              bias: torch.Tensor = None) -> torch.Tensor:
    """(M,K) row-major @ (K,N) column-major in e4m3; fp32 accumulate.

    b is quantized through its row-major transpose view (the fused
    quantize kernel contiguous()-izes its input, which would silently
    relayout a column-major tensor) and transposed back."""
    qa, sa = quantize_e4m3(a)
    qbt, sb = quantize_e4m3(b_colmajor.t().contiguous())
    return torch._scaled_mm(qa, qbt.t(), scale_a=sa, scale_b=sb, bias=bias,
                            out_dtype=out_dtype)


def _cached_weight_q(weight: torch.Tensor, transposed: bool):
    """Quantized-weight cache keyed on the tensor's in-place version
    counter: micro-batches under grad accumulation (and every inference
    call) reuse one quantization instead of re-casting 2.4 GB of
    weights per projection call. The optimizer updates params in place
    through the flat buffer, and views share the version counter, so
    any real update invalidates the cache. Bypassed during hipGraph
    capture (a baked cache would freeze the weights into the graph).
    Memory: +1 byte/param per cached form (normal + transposed = +2
    bytes/param while PROGEN_FP8=1 — vs the 2-byte bf16 weights; fine
    in 288 GB)."""
    if weight.is_cuda and torch.cuda.is_current_stream_capturing():
        ent = None
    else:
        ent = getattr(weight, "_fp8_qcache", None)
    v = weight._version
    if ent is not None and ent.get("v") == v and transposed in ent:
        return ent[transposed]
    if transposed:
        C = _ext_or_none() if weight.is_cuda else None
        if C is not None and weight.dtype == torch.bfloat16:
            s = amax_scale(weight)
            q = C.fp8_quantize_t(weight.contiguous(), s.reshape(1))
        else:
            q, s = quantize_e4m3(weight.t().contiguous())
    else:
        q, s = quantize_e4m3(weight)
    if ent is None or ent.get("v") != v:
        ent = {"v": v}
        try:
            weight._fp8_qcache = ent
        except Exception:  # noqa: BLE001 — non-Parameter tensors may refuse
            pass
    ent[transposed] = (q, s)
    return q, s


class _Fp8LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias):
        x2 = x.reshape(-1, x.shape[-1])
        ctx.save_for_backward(x2, weight)
        ctx.has_bias = bias is not None
        ctx.x_shape = x.shape
        # W (N,K) row-major -> W.t() is (K,N) column-major
        qx, sx = quantize_e4m3(x2)
        qw, sw = _cached_weight_q(weight, transposed=False)
        y = torch._scaled_mm(qx, qw.t(), scale_a=sx, scale_b=sw,
                             bias=bias.to(torch.bfloat16) if bias is not None else None,
                             out_dtype=x.dtype)
        return y.reshape(*x.shape[:-1], weight.shape[0])

    @staticmethod
    def backward(ctx, dy):
        x2, weight = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1]).contiguous()
        # dgrad in fp8: dy (M,N) @ W (N,K); _scaled_mm wants the second
        # operand column-major — the fused transpose-quantize kernel
        # emits (K,N) e4m3 directly from the (N,K) bf16 weight (one
        # LDS-tiled pass instead of a bf16 transpose copy + 4-pass cast)
        qdy, sdy = quantize_e4m3(dy2)
        qwt, swt = _cached_weight_q(weight, transposed=True)
        dx = torch._scaled_mm(qdy, qwt.t(), scale_a=sdy, scale_b=swt,
                              out_dtype=dy.dtype)
        # wgrad in bf16 (library GEMM)
        dw = dy2.transpose(0, 1).matmul(x2)
        db = dy2.sum(dim=0) if ctx.has_bias else None
        return dx.reshape(ctx.x_shape), dw, db


def fp8_linear(x: torch.Tensor, weight: torch.Tensor,
               bias: torch.Tensor = None) -> torch.Tensor:
    return _Fp8LinearFn.apply(x, weight, bias)


def fp8_eligible(x: torch.Tensor, weight: torch.Tensor) -> bool:
    if not (ENABLED and x.is_cuda and x.dtype == torch.bfloat16):
        return False
    m = x.numel() // x.shape[-1]
    n, k = weight.shape
    # _scaled_mm needs 16-divisible dims; route only big projections
    return (min(m, n, k) >= FP8_MIN_DIM and m % 16 == 0 and n % 16 == 0
            and k % 16 == 0)
