"""Tensor-parallel ProGen: shards an existing ``ProGenBase`` in place
across the TP group (docs/tp_design.md; the reference is DP-only —
utils.py:70 — so TP is a new scale-out capability for the 6B config).

Sharding map per transformer block (two all-reduces per branch forward,
the classic Megatron column->row pairing, over RCCL on xGMI):

  - ``to_qkv``: column-parallel over HEADS — rank r takes its head block
    in EACH of the q/k/v sections, so the fused attention kernel runs
    unchanged on H/tp local heads (it is per-head: blockIdx.y).
  - ``to_out``: row-parallel (input = local heads' merged context;
    contiguous head blocks => contiguous input columns).
  - GLU FF: ``proj_in`` column-parallel with GLU pairing (matching
    slices of the value and gate halves), ``proj_out`` row-parallel.
  - SGU FF: ``proj_in`` column-parallel with the same half-pairing
    (SGU chunks its input into (x, gate) halves, progen.py:166);
    spatial (n, n) weights REPLICATED (channels are independent,
    progen.py:179 — replication is ~16 MB at n=2048); ``sgu.norm_weight``
    sharded with the gate channels; ``sgu.proj_out`` row-parallel;
    the outer ``proj_out`` stays replicated (its input is the reduced
    full hidden/2).
  - embeddings / logits / LayerNorms / residual stream: replicated
    (V=256 makes vocab-parallel pointless; LNs are cheap).

Replicated params with PARTIAL per-rank gradients (only the SGU spatial
weights/biases — every other replicated param sees replicated
activations and grads) must be summed across the TP group after
backward: ``sync_replicated_grads``.
"""

from __future__ import annotations

from typing import List

import torch
import torch.distributed as dist

from ..models.progen import FeedForward, LocalAttention, ProGenBase, SGU
from . import tp


class TPSGU(torch.nn.Module):
    """Channel-sharded SGU. The gate LN normalizes over the FULL d2
    channel axis (reference: progen.py:168 — hk.LayerNorm over the gate),
    so the per-row statistics are computed distributedly: an all-reduce
    of (sum, sumsq) — a (B, N, 2)-sized collective, negligible next to
    the branch all-reduces. Everything else is channel-local; the spatial
    (n, n) weights stay replicated and proj_out is row-parallel."""

    def __init__(self, full: SGU, eps: float = 1e-5):
        super().__init__()
        tpsz, r = tp.tp_size(), tp.tp_rank()
        d2 = full.norm_weight.shape[0]
        assert d2 % tpsz == 0
        self.d2_full = d2
        self.eps = eps
        sl = slice(r * (d2 // tpsz), (r + 1) * (d2 // tpsz))
        self.norm_weight = torch.nn.Parameter(
            full.norm_weight.detach()[sl].clone())
        self.spatial_weights = torch.nn.Parameter(
            full.spatial_weights.detach().clone())
        self.spatial_biases = torch.nn.Parameter(
            full.spatial_biases.detach().clone())
        self.proj_out = _row(full.proj_out)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        xa, gate = x.chunk(2, dim=-1)
        orig = gate.dtype
        if orig in (torch.bfloat16, torch.float16):
            gate = gate.float()
        # sum_both_tp, not reduce_from_tp: the reduced stats normalize
        # rank-LOCAL channels, so their gradient has a per-rank part that
        # must be all-reduced in backward too
        s = tp.sum_both_tp(gate.sum(-1, keepdim=True))
        ss = tp.sum_both_tp((gate * gate).sum(-1, keepdim=True))
        mu = s / self.d2_full
        var = torch.clamp(ss / self.d2_full - mu * mu, min=0.0)
        gate_ln = ((gate - mu) * torch.rsqrt(var + self.eps) *
                   self.norm_weight.to(gate.dtype)).to(orig)
        n = x.shape[1]
        w = self.spatial_weights[:n, :n].tril().to(gate_ln.dtype)
        gate_out = torch.einsum("bnd,mn->bmd", gate_ln, w) + \
            self.spatial_biases[:n].to(gate_ln.dtype)
        return self.proj_out(xa * gate_out)


def _qkv_head_rows(heads: int, dim_head: int) -> torch.Tensor:
    """Rows of the (3*H*DH, dim) QKV weight owned by this rank: the
    contiguous head block [r*H/tp, (r+1)*H/tp) inside each section."""
    tpsz, r = tp.tp_size(), tp.tp_rank()
    hl = heads // tpsz
    sec = heads * dim_head
    block = torch.arange(r * hl * dim_head, (r + 1) * hl * dim_head)
    return torch.cat([block, sec + block, 2 * sec + block])


def _col(full: torch.nn.Linear, **kw) -> tp.ColumnParallelLinear:
    m = tp.ColumnParallelLinear(full.in_features, full.out_features,
                                bias=full.bias is not None, **kw)
    m = m.to(device=full.weight.device, dtype=full.weight.dtype)
    m.shard_from(full.weight.detach(),
                 full.bias.detach() if full.bias is not None else None)
    return m


def _row(full: torch.nn.Linear) -> tp.RowParallelLinear:
    m = tp.RowParallelLinear(full.in_features, full.out_features,
                             bias=full.bias is not None)
    m = m.to(device=full.weight.device, dtype=full.weight.dtype)
    m.shard_from(full.weight.detach(),
                 full.bias.detach() if full.bias is not None else None)
    return m


def tp_shard_(model: ProGenBase) -> ProGenBase:
    """Shard ``model`` in place across the current TP group. All ranks
    must hold identical full weights on entry (same init seed or a
    broadcast). Forward/backward semantics are unchanged; activations on
    the residual stream stay replicated."""
    tpsz = tp.tp_size()
    if tpsz == 1:
        return model
    cfg = model.cfg
    assert cfg.heads % tpsz == 0, (cfg.heads, tpsz)
    for attn, ff in model.layers:
        assert isinstance(attn, LocalAttention) and isinstance(ff, FeedForward)
        rows = _qkv_head_rows(cfg.heads, cfg.dim_head)
        attn.to_qkv = _col(attn.to_qkv, rows=rows)
        attn.to_out = _row(attn.to_out)
        attn.heads = cfg.heads // tpsz

        if ff.sgu is not None:
            ff.proj_in = _col(ff.proj_in, shard_glu=True)  # (x, gate) pairing
            ff.sgu = TPSGU(ff.sgu)
            # outer proj_out stays replicated: input is the reduced full
            # hidden/2 coming out of sgu.proj_out
        elif ff.glu:
            ff.proj_in = _col(ff.proj_in, shard_glu=True)
            ff.proj_out = _row(ff.proj_out)
        else:
            ff.proj_in = _col(ff.proj_in)
            ff.proj_out = _row(ff.proj_out)
    return model


def sharded_params(model: ProGenBase) -> List[torch.nn.Parameter]:
    """Params whose flat-buffer content is rank-LOCAL (a shard of the
    full model): column/row linear weights (and column biases) and the
    channel-sharded SGU gate-LN scale. Everything else is replicated."""
    out = []
    for mod in model.modules():
        if isinstance(mod, tp.ColumnParallelLinear):
            out.append(mod.weight)
            if mod.bias is not None:
                out.append(mod.bias)
        elif isinstance(mod, tp.RowParallelLinear):
            out.append(mod.weight)  # bias is replicated (added post-reduce)
        elif isinstance(mod, TPSGU):
            out.append(mod.norm_weight)
    return out


def tp_grad_sumsq_fn(model: ProGenBase):
    """Returns a callable computing the GLOBAL grad sum-of-squares for
    clip-by-global-norm: sharded params' local sumsq is all-reduced
    across the TP group; replicated params (identical grads on every
    rank) are counted once. Attach as ``optim.norm_sumsq_fn``."""
    shard_ids = {id(p) for p in sharded_params(model)}

    def fn() -> torch.Tensor:
        s_sh = None
        s_rep = None
        for p in model.parameters():
            if p.grad is None:
                continue
            v = (p.grad.float() ** 2).sum()
            if id(p) in shard_ids:
                s_sh = v if s_sh is None else s_sh + v
            else:
                s_rep = v if s_rep is None else s_rep + v
        dev = next(model.parameters()).device
        s_sh = s_sh if s_sh is not None else torch.zeros((), device=dev)
        s_rep = s_rep if s_rep is not None else torch.zeros((), device=dev)
        s_sh = s_sh.contiguous().clone()
        dist.all_reduce(s_sh, group=tp.tp_group())
        return s_sh + s_rep

    return fn


def replicated_partial_grad_params(model: ProGenBase) -> List[torch.nn.Parameter]:
    """Replicated params whose per-rank gradient is PARTIAL (local
    channels only): the SGU spatial weights/biases."""
    out = []
    for _attn, ff in model.layers:
        if ff.sgu is not None:
            out.append(ff.sgu.spatial_weights)
            out.append(ff.sgu.spatial_biases)
    return out


def sync_replicated_grads(model: ProGenBase) -> None:
    """All-reduce (sum) the partial gradients of replicated params across
    the TP group. Call after backward, before the optimizer step."""
    if tp.tp_size() == 1:
        return
    for p in replicated_partial_grad_params(model):
        if p.grad is not None:
            dist.all_reduce(p.grad, group=tp.tp_group())
