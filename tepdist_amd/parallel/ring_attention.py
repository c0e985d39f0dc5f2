"""Ring (context-parallel) attention over P2P — beyond-parity, MI355X-first.

The reference has NO sequence/context parallelism (any
sequence split would go through generic resharding; ring attention is
explicitly out of its scope). On MI355X the ring layout is a natural fit:
each step exchanges K/V blocks with ring NEIGHBORS only — one xGMI
point-to-point link per step, the topology's sweet spot (7 links x
~153 GB/s).

Math (flash-combine): the sequence is split contiguously across the
group; each rank holds Q/K/V for its block. Forward rotates (K, V)
around the ring; per visible block the backend flash kernel produces a
NORMALIZED partial (o_j, lse_j), merged online:

    lse   <- logaddexp(lse, lse_j)
    o     <- o * exp(lse_prev - lse) + o_j * exp(lse_j - lse)

Causality with contiguous blocks: block j contributes to q-rank r iff
j < r (full, causal=False) or j == r (the aligned diagonal,
causal=True); j > r is skipped (the rotation still carries the tensors).
Backward makes a second ring pass with the GLOBAL (out, lse): calling
the block backward with the global lse yields exactly that block's
partial gradients, so dQ accumulates locally while (dK, dV) accumulate
in buffers that travel WITH their K/V around the ring and arrive home
after n steps.

Causal load balance: contiguous blocks make rank r compute r+1 blocks
(the last rank bounds wall-clock). `zigzag=True` instead gives each rank
chunks (r, 2W-1-r) of 2W global chunks; every step then computes exactly
2 chunk-pair equivalents of work on every rank (the j==rank step is the
two diagonals plus one full pair; j<rank is both q chunks against the
LOW kv chunk; j>rank is the HIGH q chunk against both kv chunks), so
causal work is identical across ranks. `zigzag_shard`/`zigzag_unshard`
convert between a full sequence and the zigzag layout."""

from __future__ import annotations

import math

import torch
import torch.distributed as dist

from tepdist_amd.ops.interface import _backend


def _ring_sendrecv(tensors, src, dst, group):
    """Exchange a list of tensors with ring neighbors: send to `dst`,
    receive same-shaped tensors from `src`. Returns the received list."""
    recvs = [torch.empty_like(t) for t in tensors]
    ops = []
    for t, r in zip(tensors, recvs):
        ops.append(dist.P2POp(dist.isend, t.contiguous(), dst, group))
        ops.append(dist.P2POp(dist.irecv, r, src, group))
    for w in dist.batch_isend_irecv(ops):
        w.wait()
    return recvs


def _blk_fwd(q, k, v, causal):
    """One block's normalized partial (o, lse) via the backend (flash on
    GPU, fp32 composed on CPU)."""
    if q.is_cuda:
        be = _backend(q)
        out, res = be.attention_fwd(q, k, v, causal=causal)
        if len(res) == 2:               # flash path: (out, lse)
            return out, res[1].reshape(q.shape[0], q.shape[1], q.shape[2])
    scale = 1.0 / math.sqrt(q.shape[-1])
    s = (q.float() @ k.float().transpose(-1, -2)) * scale
    if causal:
        S = s.shape[-1]
        mask = torch.ones(S, s.shape[-1], dtype=torch.bool,
                          device=s.device).triu(1)
        s = s.masked_fill(mask, float("-inf"))
    lse = torch.logsumexp(s, dim=-1)
    p = torch.exp(s - lse.unsqueeze(-1))
    return (p @ v.float()).to(q.dtype), lse


def _blk_bwd(dout, q, k, v, out_g, lse_g, delta, causal):
    """One block's partial grads given the GLOBAL (out, lse): p recomputed
    against lse_g is exactly this block's share of the global softmax."""
    if q.is_cuda:
        be = _backend(q)
        return be.attention_bwd(dout, q, k, v, (out_g, lse_g.reshape(
            lse_g.shape[0] * lse_g.shape[1], -1)), causal=causal)
    scale = 1.0 / math.sqrt(q.shape[-1])
    s = (q.float() @ k.float().transpose(-1, -2)) * scale
    if causal:
        S = s.shape[-1]
        mask = torch.ones(S, s.shape[-1], dtype=torch.bool,
                          device=s.device).triu(1)
        s = s.masked_fill(mask, float("-inf"))
    p = torch.exp(s - lse_g.unsqueeze(-1).float())
    dp = dout.float() @ v.float().transpose(-1, -2)
    ds = p * (dp - delta.unsqueeze(-1)) * scale
    dq = (ds @ k.float()).to(q.dtype)
    dk = (ds.transpose(-1, -2) @ q.float()).to(q.dtype)
    dv = (p.transpose(-1, -2) @ dout.float()).to(q.dtype)
    return dq, dk, dv


def _zz_pairs(rank, j, world):
    """Visible (q_chunk, kv_chunk, causal) triples for the zigzag layout
    when rank `rank` holds the kv blocks of rank `j`. Local chunk 0 is
    global chunk `owner`, chunk 1 is global chunk 2*world-1-owner; chunk
    visibility follows global causal order (see module docstring)."""
    if j == rank:
        return ((0, 0, True), (1, 0, False), (1, 1, True))
    if j < rank:
        return ((0, 0, False), (1, 0, False))
    return ((1, 0, False), (1, 1, False))


def zigzag_shard(x, world, dim=2):
    """Full sequence -> list of per-rank zigzag shards: rank r gets
    chunks (r, 2*world-1-r) of 2*world, concatenated along `dim`."""
    chunks = x.chunk(2 * world, dim=dim)
    return [torch.cat([chunks[r], chunks[2 * world - 1 - r]], dim=dim)
            for r in range(world)]


def zigzag_unshard(shards, dim=2):
    """Inverse of zigzag_shard: per-rank shards -> full sequence."""
    world = len(shards)
    out = [None] * (2 * world)
    for r, s in enumerate(shards):
        a, b = s.chunk(2, dim=dim)
        out[r], out[2 * world - 1 - r] = a, b
    return torch.cat(out, dim=dim)


class _RingAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, group, causal):
        world = dist.get_world_size(group) if dist.is_initialized() else 1
        if world == 1:
            out, lse = _blk_fwd(q, k, v, causal)
            ctx.save_for_backward(q, k, v, out, lse)
            ctx.group, ctx.causal, ctx.world = group, causal, 1
            return out
        rank = dist.get_rank(group)
        ranks = dist.get_process_group_ranks(group) if group is not None \
            else list(range(world))
        nxt, prv = ranks[(rank + 1) % world], ranks[(rank - 1) % world]
        kj, vj = k, v
        out = None
        lse = None
        for step in range(world):
            j = (rank - step) % world          # kv block now resident
            if step + 1 < world:               # pre-rotate for next step
                nk, nv = _ring_sendrecv([kj, vj], prv, nxt, group)
            if not causal or j <= rank:
                o_j, lse_j = _blk_fwd(q, kj, vj,
                                      causal and j == rank)
                if out is None:
                    out, lse = o_j.float(), lse_j
                else:
                    lse_n = torch.logaddexp(lse, lse_j)
                    out = out * torch.exp(lse - lse_n).unsqueeze(-1) + \
                        o_j.float() * torch.exp(lse_j - lse_n).unsqueeze(-1)
                    lse = lse_n
            if step + 1 < world:
                kj, vj = nk, nv
        out = out.to(q.dtype)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.group, ctx.causal, ctx.world = group, causal, world
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse = ctx.saved_tensors
        group, causal, world = ctx.group, ctx.causal, ctx.world
        if world == 1:
            dq, dk, dv = _blk_bwd(dout.contiguous(), q, k, v, out, lse,
                                  (dout.float() * out.float()).sum(-1),
                                  causal)
            return dq, dk, dv, None, None
        rank = dist.get_rank(group)
        ranks = dist.get_process_group_ranks(group) if group is not None \
            else list(range(world))
        nxt, prv = ranks[(rank + 1) % world], ranks[(rank - 1) % world]
        delta = (dout.float() * out.float()).sum(-1)
        dout = dout.contiguous()
        dq = torch.zeros_like(q)
        # (k, v, dk_acc, dv_acc) travel together; after `world` rotations
        # each (dk, dv) is home with every rank's contribution summed
        kj, vj = k, v
        dk_acc = torch.zeros_like(k)
        dv_acc = torch.zeros_like(v)
        for step in range(world):
            j = (rank - step) % world
            if not causal or j <= rank:
                dq_j, dk_j, dv_j = _blk_bwd(dout, q, kj, vj, out, lse,
                                            delta, causal and j == rank)
                dq += dq_j
                dk_acc += dk_j
                dv_acc += dv_j
            kj, vj, dk_acc, dv_acc = _ring_sendrecv(
                [kj, vj, dk_acc, dv_acc], prv, nxt, group)
        # after world rotations the accumulators are back at their owner
        return dq, dk_acc, dv_acc, None, None


class _ZigzagRingAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, group):
        world = dist.get_world_size(group) if dist.is_initialized() else 1
        if world == 1:
            out, lse = _blk_fwd(q, k, v, True)
            ctx.save_for_backward(q, k, v, out, lse)
            ctx.group, ctx.world = group, 1
            return out
        rank = dist.get_rank(group)
        ranks = dist.get_process_group_ranks(group) if group is not None \
            else list(range(world))
        nxt, prv = ranks[(rank + 1) % world], ranks[(rank - 1) % world]
        c = q.shape[2] // 2
        out = torch.zeros(q.shape, dtype=torch.float32, device=q.device)
        lse = torch.full(q.shape[:3], float("-inf"),
                         dtype=torch.float32, device=q.device)
        kj, vj = k, v
        for step in range(world):
            j = (rank - step) % world
            if step + 1 < world:
                nk, nv = _ring_sendrecv([kj, vj], prv, nxt, group)
            for qa, kb, caus in _zz_pairs(rank, j, world):
                o_p, lse_p = _blk_fwd(q.narrow(2, qa * c, c),
                                      kj.narrow(2, kb * c, c),
                                      vj.narrow(2, kb * c, c), caus)
                sl = out.narrow(2, qa * c, c)
                ll = lse.narrow(2, qa * c, c)
                lse_n = torch.logaddexp(ll, lse_p)
                sl.mul_(torch.exp(ll - lse_n).unsqueeze(-1))
                sl.add_(o_p.float() *
                        torch.exp(lse_p - lse_n).unsqueeze(-1))
                ll.copy_(lse_n)
            if step + 1 < world:
                kj, vj = nk, nv
        out = out.to(q.dtype)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.group, ctx.world = group, world
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse = ctx.saved_tensors
        group, world = ctx.group, ctx.world
        if world == 1:
            dq, dk, dv = _blk_bwd(dout.contiguous(), q, k, v, out, lse,
                                  (dout.float() * out.float()).sum(-1),
                                  True)
            return dq, dk, dv, None
        rank = dist.get_rank(group)
        ranks = dist.get_process_group_ranks(group) if group is not None \
            else list(range(world))
        nxt, prv = ranks[(rank + 1) % world], ranks[(rank - 1) % world]
        c = q.shape[2] // 2
        delta = (dout.float() * out.float()).sum(-1)
        dout = dout.contiguous()
        dq = torch.zeros_like(q)
        kj, vj = k, v
        dk_acc = torch.zeros_like(k)
        dv_acc = torch.zeros_like(v)
        for step in range(world):
            j = (rank - step) % world
            for qa, kb, caus in _zz_pairs(rank, j, world):
                dq_p, dk_p, dv_p = _blk_bwd(
                    dout.narrow(2, qa * c, c).contiguous(),
                    q.narrow(2, qa * c, c).contiguous(),
                    kj.narrow(2, kb * c, c).contiguous(),
                    vj.narrow(2, kb * c, c).contiguous(),
                    out.narrow(2, qa * c, c).contiguous(),
                    lse.narrow(2, qa * c, c).contiguous(),
                    delta.narrow(2, qa * c, c).contiguous(), caus)
                dq.narrow(2, qa * c, c).add_(dq_p)
                dk_acc.narrow(2, kb * c, c).add_(dk_p)
                dv_acc.narrow(2, kb * c, c).add_(dv_p)
            kj, vj, dk_acc, dv_acc = _ring_sendrecv(
                [kj, vj, dk_acc, dv_acc], prv, nxt, group)
        return dq, dk_acc, dv_acc, None


def ring_attention(q, k, v, group=None, causal: bool = True,
                   zigzag: bool = False):
    """Context-parallel attention: q/k/v are this rank's sequence block
    [B, H, S_local, D]; returns this rank's output block. Exact (up to
    dtype rounding) vs full-sequence attention. zigzag=True uses the
    causal load-balanced chunk layout (shards from `zigzag_shard`)."""
    if zigzag:
        if not causal:
            raise ValueError("zigzag layout is causal-only")
        return _ZigzagRingAttention.apply(q, k, v, group)
    return _RingAttention.apply(q, k, v, group, causal)
