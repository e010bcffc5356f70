"""fp8 (OCP e4m3) quantization groundwork for the ProGen-6B TP=8 config
(BASELINE.json config #5; plan: docs/tp_design.md "fp8 path").

MI355X doubles MFMA throughput at fp8 (≈5 PFLOP/s dense vs ≈2.5 bf16),
so the 6B projections move to e4m3 GEMMs with per-tensor scales and fp32
accumulation. This module is the dtype plumbing — amax-based scaling,
quantize/dequantize, and a simulated-fp8 matmul used by the CPU tests to
pin the quantization error envelope. The GPU GEMM itself (hipBLASLt fp8
or `mfma_f32_16x16x32_fp8_fp8` tiles in the hand-written kernels) is
round-2 work and is gated on measurement (TODO.md).

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


def quantize_e4m3(t: torch.Tensor,
                  scale: torch.Tensor = None) -> Tuple[torch.Tensor, torch.Tensor]:
    """t -> (e4m3 tensor, fp32 scale) with t ≈ fp8 * scale."""
    if scale is None:
        scale = amax_scale(t)
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
