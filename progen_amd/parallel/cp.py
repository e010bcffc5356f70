"""Context (sequence) parallelism — the halo-exchange design of
docs/cp_design.md, implemented reference-path first (composed torch ops,
gloo-testable). The reference has no sequence sharding;
ProGen's window locality makes it nearly free:

  - rank r owns contiguous rows [r*L, (r+1)*L) of every sequence,
    L = N/P a multiple of window_size so window boundaries align with
    shard boundaries;
  - token shift: 1-row halo of the LN output (progen.py:43-46);
  - windowed attention: the previous rank's LAST window of rotated
    (k, v) is the lookback band for local window 0 — rank 0's halo is
    zeros, which IS the reference's window-0 zero-pad quirk
    (progen.py:90-96), so the quirk generalizes instead of being a
    special case;
  - SGU: the one global op — the LN'd gate is all-gathered along the
    sequence (option 1 of docs/cp_design.md; SGU layers are only the
    last ``global_mlp_depth``);
  - CE: per-sequence masked sums are all-reduced, the first-pad-as-EOS
    cumsum offset comes from a pad-count prefix exchange.

Parameter gradients are partial (local rows only) — ``cp_sync_grads``
all-reduces them, exactly like DP. Round 2 swaps the composed ops for
the HIP kernels (attention_fwd already consumes a [prev ‖ own] band, so
the kernel runs unchanged on an extended local KV buffer) and overlaps
the halo send/recv with the previous layer's compute.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist
import torch.nn.functional as F

from ..models.progen import ProGenBase
from ..ops import reference as R

_CP_GROUP: Optional[dist.ProcessGroup] = None
_DP_GROUP: Optional[dist.ProcessGroup] = None


def init_context_parallel(cp_size: int) -> None:
    global _CP_GROUP, _DP_GROUP
    world = dist.get_world_size()
    assert world % cp_size == 0
    rank = dist.get_rank()
    for start in range(0, world, cp_size):
        ranks = list(range(start, start + cp_size))
        group = dist.new_group(ranks)
        if rank in ranks:
            _CP_GROUP = group
    # orthogonal DP axis (ranks at the same sequence position across
    # replicas); every rank must create every group (collective)
    for pos in range(cp_size):
        ranks = list(range(pos, world, cp_size))
        group = dist.new_group(ranks)
        if rank in ranks:
            _DP_GROUP = group


def dp_group() -> Optional[dist.ProcessGroup]:
    """The orthogonal data-parallel group of a CP x DP mesh."""
    return _DP_GROUP


def cp_size() -> int:
    return dist.get_world_size(_CP_GROUP) if _CP_GROUP is not None else 1


def cp_rank() -> int:
    return dist.get_rank(_CP_GROUP) if _CP_GROUP is not None else 0


def _nbr(offset: int) -> int:
    """Global rank of the CP neighbor at +offset, or -1 at the boundary."""
    r = cp_rank() + offset
    if r < 0 or r >= cp_size():
        return -1
    ranks = dist.get_process_group_ranks(_CP_GROUP)
    return ranks[r]


class _HaloFromPrev(torch.autograd.Function):
    """Receive the LAST ``rows`` rows of the previous rank's tensor
    (zeros on rank 0); send ours to the next rank. Backward routes the
    received halo's grad back to the owner, who ADDS it to its own last
    rows' grad. x: (B, L, D) -> halo (B, rows, D)."""

    @staticmethod
    def forward(ctx, x, rows: int):
        ctx.rows = rows
        ctx.shape = x.shape
        prev, nxt = _nbr(-1), _nbr(+1)
        halo = torch.zeros(x.shape[0], rows, x.shape[2], dtype=x.dtype,
                           device=x.device)
        ops = []
        if nxt >= 0:
            ops.append(dist.P2POp(dist.isend, x[:, -rows:].contiguous(), nxt,
                                  group=_CP_GROUP))
        if prev >= 0:
            ops.append(dist.P2POp(dist.irecv, halo, prev, group=_CP_GROUP))
        if ops:
            for w in dist.batch_isend_irecv(ops):
                w.wait()
        return halo

    @staticmethod
    def backward(ctx, dhalo):
        prev, nxt = _nbr(-1), _nbr(+1)
        dx = torch.zeros(ctx.shape, dtype=dhalo.dtype, device=dhalo.device)
        recv = torch.zeros_like(dhalo)
        ops = []
        if prev >= 0:  # return the halo's grad to its owner
            ops.append(dist.P2POp(dist.isend, dhalo.contiguous(), prev,
                                  group=_CP_GROUP))
        if nxt >= 0:   # collect the grad of the rows we sent forward
            ops.append(dist.P2POp(dist.irecv, recv, nxt, group=_CP_GROUP))
        if ops:
            for w in dist.batch_isend_irecv(ops):
                w.wait()
        if nxt >= 0:
            dx[:, -ctx.rows:] = recv
        return dx, None


class _GatherSeq(torch.autograd.Function):
    """All-gather along the sequence dim. Each rank consumes the full
    tensor DIFFERENTLY (its own spatial-matmul rows), so backward
    all-reduces the full grad and slices this rank's rows."""

    @staticmethod
    def forward(ctx, x):
        P = cp_size()
        parts = [torch.empty_like(x) for _ in range(P)]
        dist.all_gather(parts, x.contiguous(), group=_CP_GROUP)
        return torch.cat(parts, dim=1)

    @staticmethod
    def backward(ctx, dy):
        dy = dy.contiguous().clone()
        dist.all_reduce(dy, group=_CP_GROUP)
        L = dy.shape[1] // cp_size()
        r = cp_rank()
        return dy[:, r * L:(r + 1) * L].contiguous()


def _shift_cp(y: torch.Tensor) -> torch.Tensor:
    """Token shift across the shard boundary: local row 0's shifted half
    comes from the previous rank's last LN'd row (progen.py:43-46)."""
    d = y.shape[-1]
    split = -(-d // 2)
    halo = _HaloFromPrev.apply(y, 1)           # (B, 1, D)
    shifted = torch.cat((halo[..., :split], y[:, :-1, :split]), dim=1)
    return torch.cat((shifted, y[..., split:]), dim=-1)


class _PatchRow0(torch.autograd.Function):
    """Overwrite local row 0's shifted (first-half) channels with the
    cross-rank halo row; backward routes that slice's grad to the halo
    (and zeroes it locally, since the fused kernel put zeros there)."""

    @staticmethod
    def forward(ctx, y, halo_head):
        ctx.split = halo_head.shape[-1]
        out = y.clone()
        out[:, 0, :ctx.split] = halo_head
        return out

    @staticmethod
    def backward(ctx, dy):
        dhead = dy[:, 0, :ctx.split].contiguous()
        dy = dy.clone()
        dy[:, 0, :ctx.split] = 0
        return dy, dhead


def _ln_shift_cp(x, weight, shift: bool) -> torch.Tensor:
    from ..ops import dispatch
    from ..ops import functional as OF
    if dispatch.use_hip(x):
        # kernel path: the fused ln_shift kernel
        # zero-fills row 0's shifted half; the cross-rank halo row is
        # patched in afterwards. The raw LN of the LAST row (what the
        # next rank needs) is recomputed eagerly — one row.
        y = OF.ln_shift(x, weight, shift=shift)
        if not shift:
            return y
        d = y.shape[-1]
        split = -(-d // 2)
        ln_last = R.layernorm_nobias(x[:, -1:], weight)
        halo_row = _HaloFromPrev.apply(ln_last, 1)     # (B, 1, D)
        return _PatchRow0.apply(y, halo_row[:, 0, :split])
    y = R.layernorm_nobias(x, weight)
    return _shift_cp(y) if shift else y


def _attn_cp(attn, x, sin_l, cos_l) -> torch.Tensor:
    """Local windowed attention with a one-window KV halo. Bands are
    [prev-window ‖ own-window]; rank 0's first band uses the zero halo,
    reproducing the reference's unmasked zero-key quirk exactly.

    On GPU this runs the fused HIP kernels: the local shard goes through
    rope_qkv + attn_fwd with the kernels' ``halo`` argument carrying the
    previous rank's last window of ROTATED [k|v] (the rank rotates its
    own tail in torch — differentiably, so attn_bwd's dhalo flows back
    through the exchange; tests/test_gpu_cp_halo.py pins the mechanics
    against a full-sequence kernel run)."""
    from ..ops import dispatch
    from ..ops import functional as OF
    B, L, _ = x.shape
    h = attn.heads
    wsz = attn.window_size
    y = _ln_shift_cp(x, attn.norm_weight, attn.shift_tokens)
    if dispatch.use_hip(x):
        qkv = attn.to_qkv(y)
        dh = qkv.shape[-1] // (3 * h)
        # rotate this rank's last window of (k ‖ v) at absolute
        # positions to send forward
        tail = qkv[:, L - wsz:]
        kt = tail[..., h * dh:2 * h * dh].view(B, wsz, h, dh)
        vt = tail[..., 2 * h * dh:].view(B, wsz, h, dh)
        st = sin_l[L - wsz:].to(kt.dtype).view(wsz, 1, dh)
        ct = cos_l[L - wsz:].to(kt.dtype).view(wsz, 1, dh)
        kt = kt * ct + R.rotate_every_two(kt) * st
        vt = vt * ct + R.rotate_every_two(vt) * st
        kv_tail = torch.cat((kt, vt), dim=2).reshape(B, wsz, 2 * h * dh)
        halo = _HaloFromPrev.apply(kv_tail, wsz)
        out = OF.local_attention(qkv, sin_l, cos_l, h, wsz,
                                 halo=halo.contiguous())
        return attn.to_out(out)
    qkv = F.linear(y, attn.to_qkv.weight)
    dh = qkv.shape[-1] // (3 * h)
    q, k, v = qkv.chunk(3, dim=-1)

    def heads(t):
        return t.view(B, L, h, dh).transpose(1, 2)  # (B, h, L, dh)

    q, k, v = map(heads, (q, k, v))
    sin_l = sin_l.to(q.dtype)
    cos_l = cos_l.to(q.dtype)
    # rotary at ABSOLUTE positions (sin_l/cos_l are the local slice);
    # applied to q, k AND v (progen.py:87)
    q, k, v = (R.apply_rotary_pos_emb(t, sin_l, cos_l) for t in (q, k, v))

    # exchange the last window of rotated (k ‖ v) in one message
    kv = torch.cat((k, v), dim=-1)                     # (B, h, L, 2dh)
    kv_flat = kv.transpose(1, 2).reshape(B, L, h * 2 * dh)
    halo = _HaloFromPrev.apply(kv_flat, wsz)           # (B, wsz, h*2dh)
    halo = halo.view(B, wsz, h, 2 * dh).transpose(1, 2)
    kv_ext = torch.cat((halo, kv), dim=2)              # (B, h, wsz+L, 2dh)
    k_ext, v_ext = kv_ext.split(dh, dim=-1)

    w = L // wsz
    scale = dh ** -0.5
    qw = q.view(B, h, w, wsz, dh)
    # band j = kv_ext rows [j*wsz, j*wsz + 2*wsz)
    kb = torch.stack([k_ext[:, :, j * wsz:(j + 2) * wsz] for j in range(w)], dim=2)
    vb = torch.stack([v_ext[:, :, j * wsz:(j + 2) * wsz] for j in range(w)], dim=2)
    sim = torch.einsum("bhwid,bhwjd->bhwij", qw, kb) * scale
    mask = torch.ones(wsz, 2 * wsz, dtype=torch.bool, device=x.device).tril(wsz)
    sim = torch.where(mask, sim, torch.tensor(R.ATTN_MASK_VALUE,
                                              dtype=sim.dtype, device=sim.device))
    sim = sim - sim.amax(dim=-1, keepdim=True).detach()
    attn_w = sim.softmax(dim=-1)
    out = torch.einsum("bhwij,bhwjd->bhwid", attn_w, vb)
    out = out.permute(0, 2, 3, 1, 4).reshape(B, L, h * dh)
    return F.linear(out, attn.to_out.weight, attn.to_out.bias)


def _ff_cp(ff, x, row0: int) -> torch.Tensor:
    y = _ln_shift_cp(x, ff.norm_weight, ff.shift_tokens)
    hdn = F.linear(y, ff.proj_in.weight, ff.proj_in.bias)
    if ff.glu:
        a, g = hdn.chunk(2, dim=-1)
        hdn = a * F.gelu(g, approximate="tanh")
    else:
        hdn = F.gelu(hdn, approximate="tanh")
    if ff.sgu is not None:
        sgu = ff.sgu
        xa, gate = hdn.chunk(2, dim=-1)
        gate_ln = R.layernorm_nobias(gate, sgu.norm_weight)
        gate_full = _GatherSeq.apply(gate_ln)          # (B, N, d2)
        n = gate_full.shape[1]
        wmat = sgu.spatial_weights[:n, :n].tril().to(gate_full.dtype)
        # only this rank's output rows of the spatial matmul
        wrows = wmat[row0:row0 + xa.shape[1]]
        gate_out = torch.einsum("bnd,mn->bmd", gate_full, wrows) + \
            sgu.spatial_biases[row0:row0 + xa.shape[1]].to(gate_full.dtype)
        hdn = xa * gate_out
        hdn = F.linear(hdn, sgu.proj_out.weight, sgu.proj_out.bias)
    return F.linear(hdn, ff.proj_out.weight, ff.proj_out.bias)


def cp_forward(model: ProGenBase, x_local: torch.Tensor) -> torch.Tensor:
    """Forward this rank's sequence shard. x_local: (B, L) int tokens,
    rows [r*L, (r+1)*L) of the global sequence; returns local logits."""
    cfg = model.cfg
    L = x_local.shape[1]
    assert L % cfg.window_size == 0, "shard must align to window boundaries"
    row0 = cp_rank() * L
    sin = model.rotary_sin[row0:row0 + L]
    cos = model.rotary_cos[row0:row0 + L]
    h = model.embed(x_local.long())
    for attn, ff in model.layers:
        h = h + _attn_cp(attn, h, sin, cos)
        h = h + _ff_cp(ff, h, row0)
    h = R.layernorm_nobias(h, model.final_norm_weight)
    return F.linear(h, model.to_logits.weight, model.to_logits.bias)


def cp_loss(model: ProGenBase, data: torch.Tensor) -> torch.Tensor:
    """Full-batch loss from sequence shards: every rank passes the SAME
    (B, N+1) batch; ids/labels are sharded internally. Matches
    utils.compute_loss on one rank exactly (per-seq masked mean with
    first-pad-as-EOS, then batch mean — reference utils.py:45-76)."""
    P, r = cp_size(), cp_rank()
    ids, labels = data[:, :-1], data[:, 1:]
    N = ids.shape[1]
    L = N // P
    my_ids = ids[:, r * L:(r + 1) * L]
    my_labels = labels[:, r * L:(r + 1) * L].long()

    logits = cp_forward(model, my_ids)
    if logits.dtype in (torch.bfloat16, torch.float16):
        logits = logits.float()
    lp = torch.log_softmax(logits, dim=-1)
    nll = -lp.gather(-1, my_labels.unsqueeze(-1)).squeeze(-1)

    # first-pad-as-EOS needs the GLOBAL pad cumsum: offset by the number
    # of pads on earlier ranks (host ints, non-differentiable)
    local_pads = (my_labels == 0).long().sum(dim=-1)
    all_pads = [torch.zeros_like(local_pads) for _ in range(P)]
    dist.all_gather(all_pads, local_pads, group=_CP_GROUP)
    prefix = sum(all_pads[:r]) if r > 0 else torch.zeros_like(local_pads)

    mask = my_labels != 0
    cum = (~mask).long().cumsum(dim=-1) + prefix.unsqueeze(-1)
    eos_mask = (~mask) & (cum == 1)
    m = (mask | eos_mask).to(nll.dtype)

    # per-sequence masked sums, reduced across the sequence shards
    num = (nll * m).sum(dim=-1)
    den = m.sum(dim=-1)
    packed = torch.stack((num, den), dim=0).contiguous()
    dist.all_reduce(packed, group=_CP_GROUP)  # consumers replicated:
    num_g = num + (packed[0] - num).detach()  # keep autograd through the
    den_g = packed[1].detach()                # local contribution only
    return (num_g / den_g).mean()


def cp_sync_grads(model: ProGenBase) -> None:
    """All-reduce (sum) every parameter gradient across the CP group —
    each rank's grads cover its sequence rows only (analogous to DP)."""
    if cp_size() == 1:
        return
    for p in model.parameters():
        if p.grad is not None:
            dist.all_reduce(p.grad, group=_CP_GROUP)
