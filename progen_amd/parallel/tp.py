"""Tensor parallelism: Megatron-style column/row-parallel linears over
RCCL (docs/tp_design.md; the reference has no TP — its pmap at
utils.py:70 is DP-only — so this is a new scale-out capability sized for
the MI355X node: TP group = the 8 xGMI-meshed GPUs of one node).

The classic pairing per transformer branch:

  ColumnParallelLinear (input replicated, output sharded; backward
  all-reduces dx) -> local nonlinearity / attention on the shard ->
  RowParallelLinear (input sharded, partial outputs all-reduced in
  forward).

Exactly 2 activation all-reduces per branch per direction, on
(B, N, dim) bf16 tensors — per-link-bound ring collectives on the
7x153 GB/s xGMI mesh.

Sharding rules for the ProGen block (docs/tp_design.md):
  - to_qkv: column-parallel over HEADS (the fused attention kernel is
    per-head — blockIdx.y — so it runs unchanged on H/tp local heads);
  - FF proj_in: column-parallel with GLU-aware sharding (each rank gets
    matching slices of the value and gate halves; shard_glu=True);
  - to_out / proj_out: row-parallel;
  - SGU: channel-sharded (spatial (n, n) weights replicated);
  - embeddings / logits / LNs: replicated (V=256 makes vocab-parallel
    pointless).
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist
import torch.nn.functional as F

from ..models.progen import _haiku_linear_init_

_TP_GROUP: Optional[dist.ProcessGroup] = None
_DP_GROUP: Optional[dist.ProcessGroup] = None


def init_tensor_parallel(tp_size: int) -> None:
    """Split WORLD into contiguous TP groups of ``tp_size`` ranks (ranks
    [0..tp-1], [tp..2tp-1], ... — contiguous ranks share a node's xGMI
    mesh under torchrun's rank assignment)."""
    global _TP_GROUP, _DP_GROUP
    world = dist.get_world_size()
    assert world % tp_size == 0, (world, tp_size)
    rank = dist.get_rank()
    for start in range(0, world, tp_size):
        ranks = list(range(start, start + tp_size))
        group = dist.new_group(ranks)
        if rank in ranks:
            _TP_GROUP = group
    # the orthogonal DP axis: ranks sharing a TP position across
    # replicas ({r, r+tp, r+2tp, ...}) — every rank must create every
    # group (new_group is collective)
    for pos in range(tp_size):
        ranks = list(range(pos, world, tp_size))
        group = dist.new_group(ranks)
        if rank in ranks:
            _DP_GROUP = group


def tp_group() -> Optional[dist.ProcessGroup]:
    return _TP_GROUP


def dp_group() -> Optional[dist.ProcessGroup]:
    """The orthogonal data-parallel group of a TP x DP mesh (ranks with
    the same TP position across replicas)."""
    return _DP_GROUP


def tp_size() -> int:
    return dist.get_world_size(_TP_GROUP) if _TP_GROUP is not None else 1


def tp_rank() -> int:
    return dist.get_rank(_TP_GROUP) if _TP_GROUP is not None else 0


class _CopyToTP(torch.autograd.Function):
    """Identity forward; all-reduce gradient (input side of a
    column-parallel linear)."""

    @staticmethod
    def forward(ctx, x):
        return x

    @staticmethod
    def backward(ctx, dx):
        dx = dx.contiguous()
        dist.all_reduce(dx, group=_TP_GROUP)
        return dx


class _ReduceFromTP(torch.autograd.Function):
    """All-reduce forward; identity gradient (output side of a
    row-parallel linear)."""

    @staticmethod
    def forward(ctx, x):
        x = x.contiguous()
        dist.all_reduce(x, group=_TP_GROUP)
        return x

    @staticmethod
    def backward(ctx, dx):
        return dx


class _SumBothTP(torch.autograd.Function):
    """All-reduce forward AND backward. Needed when the reduced value
    feeds DIFFERENT computations on each rank (e.g. distributed LN
    statistics normalizing rank-local channel shards): dL/dS then has a
    distinct per-rank part that must itself be summed. (_ReduceFromTP's
    identity backward is only correct when the consumers of the reduced
    value are replicated, as in a row-parallel linear.)"""

    @staticmethod
    def forward(ctx, x):
        x = x.contiguous().clone()
        dist.all_reduce(x, group=_TP_GROUP)
        return x

    @staticmethod
    def backward(ctx, dy):
        dy = dy.contiguous().clone()
        dist.all_reduce(dy, group=_TP_GROUP)
        return dy


class _GatherFromTP(torch.autograd.Function):
    """All-gather shards along the last dim; backward slices out this
    rank's piece (feature-dim gather has no cross-rank grad mixing)."""

    @staticmethod
    def forward(ctx, x):
        parts = [torch.empty_like(x) for _ in range(tp_size())]
        dist.all_gather(parts, x.contiguous(), group=_TP_GROUP)
        return torch.cat(parts, dim=-1)

    @staticmethod
    def backward(ctx, dy):
        n = dy.shape[-1] // tp_size()
        r = tp_rank()
        return dy[..., r * n:(r + 1) * n].contiguous()


def copy_to_tp(x: torch.Tensor) -> torch.Tensor:
    return _CopyToTP.apply(x) if tp_size() > 1 else x


def reduce_from_tp(x: torch.Tensor) -> torch.Tensor:
    return _ReduceFromTP.apply(x) if tp_size() > 1 else x


def sum_both_tp(x: torch.Tensor) -> torch.Tensor:
    return _SumBothTP.apply(x) if tp_size() > 1 else x


def _glu_shard_rows(out_features: int, tp: int, rank: int) -> torch.Tensor:
    """Row indices for a GLU-paired column shard: matching slices of the
    value half and the gate half, so chunk(2) of the sharded output pairs
    the right channels (progen.py:139-141 splits the doubled hidden)."""
    half = out_features // 2
    per = half // tp
    idx = torch.arange(rank * per, (rank + 1) * per)
    return torch.cat([idx, idx + half])


class ColumnParallelLinear(torch.nn.Module):
    """Y = X W^T sharded over output features; input replicated.

    ``shard_glu`` pairs the shard across the two GLU halves. With
    ``gather_output`` the full Y is all-gathered (used only at parity
    boundaries; inside a block the sharded output feeds the local
    nonlinearity directly)."""

    def __init__(self, in_features: int, out_features: int, bias: bool = True,
                 shard_glu: bool = False, gather_output: bool = False,
                 rows: Optional[torch.Tensor] = None):
        super().__init__()
        tp, rank = tp_size(), tp_rank()
        assert out_features % tp == 0, (out_features, tp)
        if shard_glu:
            assert (out_features // 2) % tp == 0
        self.in_features = in_features
        self.out_features = out_features
        self.local_out = out_features // tp
        self.shard_glu = shard_glu
        self.gather_output = gather_output
        # custom shard rows (e.g. the QKV head shard takes this rank's
        # head block in EACH of the q/k/v sections)
        self.custom_rows = rows
        if rows is not None:
            assert rows.numel() == self.local_out, (rows.numel(), self.local_out)
        self.weight = torch.nn.Parameter(
            torch.empty(self.local_out, in_features))
        self.bias = torch.nn.Parameter(
            torch.zeros(self.local_out)) if bias else None
        self.reset_parameters()

    def reset_parameters(self) -> None:
        # init the FULL matrix with the haiku default, then keep this
        # rank's rows: identical RNG state on all TP ranks => sharding a
        # single-rank model's weights gives bitwise-equal shards
        full_w = torch.empty(self.out_features, self.in_features)
        full_b = torch.zeros(self.out_features)
        _haiku_linear_init_(full_w, full_b)
        with torch.no_grad():
            self.weight.copy_(full_w[self._rows()])
            if self.bias is not None:
                self.bias.copy_(full_b[self._rows()])

    def _rows(self) -> torch.Tensor:
        tp, rank = tp_size(), tp_rank()
        if self.custom_rows is not None:
            return self.custom_rows
        if self.shard_glu:
            return _glu_shard_rows(self.out_features, tp, rank)
        per = self.out_features // tp
        return torch.arange(rank * per, (rank + 1) * per)

    def shard_from(self, full_weight: torch.Tensor,
                   full_bias: Optional[torch.Tensor] = None) -> None:
        """Load this rank's shard from a full (out, in) matrix."""
        rows = self._rows().to(full_weight.device)
        with torch.no_grad():
            self.weight.copy_(full_weight[rows])
            if self.bias is not None and full_bias is not None:
                self.bias.copy_(full_bias[rows])

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = F.linear(copy_to_tp(x), self.weight, self.bias)
        if self.gather_output and tp_size() > 1:
            y = _GatherFromTP.apply(y)
        return y


class RowParallelLinear(torch.nn.Module):
    """Y = X W^T with input features sharded; partial products summed by
    an all-reduce in forward. The bias is added AFTER the reduce (rank 0
    holds it) so it is counted once."""

    def __init__(self, in_features: int, out_features: int, bias: bool = True):
        super().__init__()
        tp = tp_size()
        assert in_features % tp == 0, (in_features, tp)
        self.in_features = in_features
        self.out_features = out_features
        self.local_in = in_features // tp
        self.weight = torch.nn.Parameter(
            torch.empty(out_features, self.local_in))
        self.bias = torch.nn.Parameter(
            torch.zeros(out_features)) if bias else None
        self.reset_parameters()

    def reset_parameters(self) -> None:
        full_w = torch.empty(self.out_features, self.in_features)
        full_b = torch.zeros(self.out_features)
        # fan_in of the FULL layer so the init distribution matches the
        # unsharded model
        _haiku_linear_init_(full_w, full_b)
        with torch.no_grad():
            self.weight.copy_(full_w[:, self._cols()])
            if self.bias is not None:
                self.bias.copy_(full_b)

    def _cols(self) -> torch.Tensor:
        rank = tp_rank()
        return torch.arange(rank * self.local_in, (rank + 1) * self.local_in)

    def shard_from(self, full_weight: torch.Tensor,
                   full_bias: Optional[torch.Tensor] = None) -> None:
        cols = self._cols().to(full_weight.device)
        with torch.no_grad():
            self.weight.copy_(full_weight[:, cols])
            if self.bias is not None and full_bias is not None:
                self.bias.copy_(full_bias)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        y = F.linear(x, self.weight)
        y = reduce_from_tp(y)
        if self.bias is not None:
            y = y + self.bias
        return y
