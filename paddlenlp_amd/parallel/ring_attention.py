"""Ring (context-parallel) flash attention over xGMI p2p.

Reference behavior: paddlenlp/transformers/ring_flash_attention.py —
RingFlashAttention PyLayer :306, fwd :97 (per-step attention on rotated K/V
blocks + online-softmax merge update_out_and_lse :69-84), bwd :192 (rotates
K/V one way and accumulates dK/dV via a second ring), p2p via
batch_isend_irecv :56-66.

MI355X design: the ring neighbors are adjacent ranks in the sep group =
adjacent GPUs on the xGMI fabric, so each step's K/V rotation is one
point-to-point hop.  GPU blocks run the gfx950 flash kernels via ops._C
(which return/consume the LSE); the CPU path uses fp32 torch math (the same
formulas) so gloo tests validate the ring logic end-to-end.

Two sharding modes:
- contiguous: rank r owns chunk r.  Per ring step, kv from an earlier chunk
  is fully visible (causal=False), the own chunk is causal, later chunks are
  skipped — simple, but causal work is unbalanced (rank 0 does 1 block,
  rank w-1 does w blocks).
- balanced (zigzag, `balanced=True` + `zigzag_split`): the sequence is cut
  into 2w chunks and rank r owns chunks (r, 2w-1-r), so every rank does the
  same causal work per step (reference's load-balanced chunking).  Per step
  against the kv of source rank s: s == r is plain causal on the local
  concat; s < r attends all local q to the kv's FIRST chunk (full
  visibility); s > r attends the local SECOND chunk of q to all kv.
"""
from __future__ import annotations

import math

import torch
import torch.distributed as dist

from .topology import get_topology


def _ring_send_recv(tensors, group):
    """Rotate tensors one hop around the ring: send to next, recv from prev."""
    rank = dist.get_rank(group)
    world = dist.get_world_size(group)
    ranks = dist.get_process_group_ranks(group)
    nxt = ranks[(rank + 1) % world]
    prv = ranks[(rank - 1) % world]
    recvs = [torch.empty_like(t) for t in tensors]
    ops = []
    for t, r in zip(tensors, recvs):
        ops.append(dist.P2POp(dist.isend, t.contiguous(), nxt, group))
        ops.append(dist.P2POp(dist.irecv, r, prv, group))
    for work in dist.batch_isend_irecv(ops):
        work.wait()
    return recvs


def _merge(out_acc, lse_acc, out_new, lse_new):
    """Online-softmax merge of two partial attention results.

    out: [B, S, H, D] fp32; lse: [B, H, S] fp32 (-inf where no keys seen)."""
    if out_acc is None:
        return out_new, lse_new
    m = torch.maximum(lse_acc, lse_new)
    # avoid inf-inf
    m = torch.where(torch.isinf(m) & (m < 0), torch.zeros_like(m), m)
    w_acc = torch.exp(lse_acc - m)
    w_new = torch.exp(lse_new - m)
    denom = w_acc + w_new
    lse_out = m + torch.log(denom)
    wa = (w_acc / denom).permute(0, 2, 1).unsqueeze(-1)  # [B,S,H,1]
    wn = (w_new / denom).permute(0, 2, 1).unsqueeze(-1)
    return out_acc * wa + out_new * wn, lse_out


def _block_fwd(q, k, v, causal):
    """(out fp32 [B,S,H,D], lse fp32 [B,H,S]) for one kv block."""
    if q.is_cuda:
        from ..ops.functional import _load_extension

        C = _load_extension()
        o, lse = C.flash_attn_fwd(q, k, v, causal)
        return o.float(), lse
    from ..ops import reference

    o, lse = reference.flash_attention(q.float(), k.float(), v.float(),
                                       causal=causal, return_lse=True)
    return o, lse


def _block_bwd(dout, q, k, v, o, lse, causal):
    """Per-block FA2 backward with the GLOBAL (o, lse)."""
    if q.is_cuda:
        from ..ops.functional import _load_extension

        C = _load_extension()
        return C.flash_attn_bwd(dout, q, k, v, o, lse, causal)
    # fp32 reference math
    B, Sq, Hq, D = q.shape
    Hk = k.shape[2]
    rep = Hq // Hk
    kf = k.float().repeat_interleave(rep, dim=2).permute(0, 2, 1, 3)  # [B,H,Skv,D]
    vf = v.float().repeat_interleave(rep, dim=2).permute(0, 2, 1, 3)
    qf = q.float().permute(0, 2, 1, 3)          # [B,H,Sq,D]
    dof = dout.float().permute(0, 2, 1, 3)
    of = o.float().permute(0, 2, 1, 3)
    scale = 1.0 / math.sqrt(D)
    S = qf @ kf.transpose(-1, -2) * scale       # [B,H,Sq,Skv]
    if causal:
        Skv = kf.shape[-2]
        mask = torch.ones(Sq, Skv, dtype=torch.bool).tril(Skv - Sq)
        S = S.masked_fill(~mask, float("-inf"))
    P = torch.exp(S - lse.unsqueeze(-1))
    P = torch.nan_to_num(P, nan=0.0, posinf=0.0)
    delta = (dof * of).sum(-1, keepdim=True)    # [B,H,Sq,1]
    dV = P.transpose(-1, -2) @ dof
    dP = dof @ vf.transpose(-1, -2)
    dS = P * (dP - delta) * scale
    dQ = dS @ kf
    dK = dS.transpose(-1, -2) @ qf
    dq = dQ.permute(0, 2, 1, 3).to(q.dtype)
    # sum GQA groups
    Skv = k.shape[1]
    dk = dK.permute(0, 2, 1, 3).reshape(B, Skv, Hk, rep, D).sum(3).to(k.dtype)
    dv = dV.permute(0, 2, 1, 3).reshape(B, Skv, Hk, rep, D).sum(3).to(v.dtype)
    return dq, dk, dv


class RingFlashAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, group, causal):
        rank = dist.get_rank(group)
        world = dist.get_world_size(group)
        q, k, v = q.contiguous(), k.contiguous(), v.contiguous()

        out, lse = None, None
        k_cur, v_cur = k, v
        for step in range(world):
            src = (rank - step) % world
            # rotate early so comm overlaps compute in the GPU path
            if step < world - 1:
                nxt = _ring_send_recv([k_cur, v_cur], group)
            if (not causal) or src <= rank:
                block_causal = causal and (src == rank)
                o_s, lse_s = _block_fwd(q, k_cur, v_cur, block_causal)
                out, lse = _merge(out, lse, o_s, lse_s)
            if step < world - 1:
                k_cur, v_cur = nxt
        out = out.to(q.dtype)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.group = group
        ctx.causal = causal
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse = ctx.saved_tensors
        group, causal = ctx.group, ctx.causal
        rank = dist.get_rank(group)
        world = dist.get_world_size(group)
        dout = dout.contiguous()

        dq_acc = torch.zeros_like(q)
        # (k, v, dk, dv) travel the ring together; after `world` hops each
        # block is back home with all contributions accumulated
        k_cur, v_cur = k, v
        dk_cur = torch.zeros_like(k)
        dv_cur = torch.zeros_like(v)
        for step in range(world):
            src = (rank - step) % world
            if (not causal) or src <= rank:
                block_causal = causal and (src == rank)
                dq_s, dk_s, dv_s = _block_bwd(dout, q, k_cur, v_cur, out, lse, block_causal)
                dq_acc += dq_s
                dk_cur = dk_cur + dk_s
                dv_cur = dv_cur + dv_s
            if world > 1:
                k_cur, v_cur, dk_cur, dv_cur = _ring_send_recv(
                    [k_cur, v_cur, dk_cur, dv_cur], group)
        # after world rotations, dk_cur/dv_cur hold the fully-reduced grads
        # for OUR block (it traveled all ranks and returned)
        return dq_acc, dk_cur, dv_cur, None, None


def zigzag_split(t: torch.Tensor, world: int, rank: int, dim: int = 1) -> torch.Tensor:
    """Load-balanced context-parallel shard: cut `dim` into 2*world chunks and
    return cat(chunk[rank], chunk[2*world-1-rank]).  Inverse: zigzag_gather."""
    chunks = t.chunk(2 * world, dim=dim)
    return torch.cat([chunks[rank], chunks[2 * world - 1 - rank]], dim=dim).contiguous()


def zigzag_gather(locals_per_rank, dim: int = 1) -> torch.Tensor:
    """Reassemble the full sequence from every rank's zigzag shard (test /
    logging helper; locals_per_rank[r] = that rank's [.., S/w, ..] shard)."""
    world = len(locals_per_rank)
    slots = [None] * (2 * world)
    for r, t in enumerate(locals_per_rank):
        first, second = t.chunk(2, dim=dim)
        slots[r] = first
        slots[2 * world - 1 - r] = second
    return torch.cat(slots, dim=dim)


class ZigzagRingFlashAttention(torch.autograd.Function):
    """Balanced causal ring attention over zigzag shards (see module doc)."""

    @staticmethod
    def forward(ctx, q, k, v, group):
        rank = dist.get_rank(group)
        world = dist.get_world_size(group)
        q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        B, S2, Hq, D = q.shape  # S2 = 2 chunks
        C = S2 // 2

        out = torch.zeros(B, S2, Hq, D, dtype=torch.float32, device=q.device)
        lse = torch.full((B, Hq, S2), float("-inf"),
                         dtype=torch.float32, device=q.device)
        k_cur, v_cur = k, v
        for step in range(world):
            src = (rank - step) % world
            if step < world - 1:
                nxt = _ring_send_recv([k_cur, v_cur], group)
            if src == rank:
                o_s, lse_s = _block_fwd(q, k_cur, v_cur, True)
                out, lse = _merge(out, lse, o_s, lse_s)
            elif src < rank:
                # kv chunks (src, 2w-1-src) straddle ours: only the FIRST is
                # earlier than all local q -> full visibility, second skipped
                o_s, lse_s = _block_fwd(q, k_cur[:, :C].contiguous(), v_cur[:, :C].contiguous(), False)
                out, lse = _merge(out, lse, o_s, lse_s)
            else:
                # only our SECOND q chunk (2w-1-rank) is after both kv chunks
                o_s, lse_s = _block_fwd(q[:, C:].contiguous(), k_cur, v_cur, False)
                o_m, lse_m = _merge(out[:, C:], lse[:, :, C:], o_s, lse_s)
                out = torch.cat([out[:, :C], o_m], dim=1)
                lse = torch.cat([lse[:, :, :C], lse_m], dim=2)
            if step < world - 1:
                k_cur, v_cur = nxt
        out = out.to(q.dtype)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.group = group
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse = ctx.saved_tensors
        group = ctx.group
        rank = dist.get_rank(group)
        world = dist.get_world_size(group)
        dout = dout.contiguous()
        C = q.shape[1] // 2

        dq_acc = torch.zeros_like(q)
        k_cur, v_cur = k, v
        dk_cur = torch.zeros_like(k)
        dv_cur = torch.zeros_like(v)
        for step in range(world):
            src = (rank - step) % world
            if src == rank:
                dq_s, dk_s, dv_s = _block_bwd(dout, q, k_cur, v_cur, out, lse, True)
                dq_acc += dq_s
                dk_cur = dk_cur + dk_s
                dv_cur = dv_cur + dv_s
            elif src < rank:
                dq_s, dk_s, dv_s = _block_bwd(
                    dout, q, k_cur[:, :C].contiguous(), v_cur[:, :C].contiguous(), out, lse, False)
                dq_acc += dq_s
                dk_cur = torch.cat([dk_cur[:, :C] + dk_s, dk_cur[:, C:]], dim=1)
                dv_cur = torch.cat([dv_cur[:, :C] + dv_s, dv_cur[:, C:]], dim=1)
            else:
                dq_s, dk_s, dv_s = _block_bwd(
                    dout[:, C:].contiguous(), q[:, C:].contiguous(), k_cur, v_cur,
                    out[:, C:].contiguous(), lse[:, :, C:].contiguous(), False)
                dq_acc[:, C:] += dq_s
                dk_cur = dk_cur + dk_s
                dv_cur = dv_cur + dv_s
            if world > 1:
                k_cur, v_cur, dk_cur, dv_cur = _ring_send_recv(
                    [k_cur, v_cur, dk_cur, dv_cur], group)
        return dq_acc, dk_cur, dv_cur, None


def ring_flash_attention(q, k, v, group=None, causal: bool = True,
                         balanced: bool = False):
    """q/k/v: this rank's sequence chunk [B, S/cp, H, D].  With
    balanced=True the chunk must be a `zigzag_split` shard (causal only)."""
    group = group if group is not None else get_topology().sep_parallel_group
    if group is None:
        from .. import ops

        return ops.flash_attention(q, k, v, causal=causal)
    if balanced:
        assert causal, "balanced (zigzag) sharding only applies to causal attention"
        return ZigzagRingFlashAttention.apply(q, k, v, group)
    return RingFlashAttention.apply(q, k, v, group, causal)
