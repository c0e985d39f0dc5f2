"""GPT-MoE: GShard-style top-2 gated mixture-of-experts transformer.

Mirrors the reference example's model family
(/root/reference/examples/gpt_moe/: layers/moe_layers.py top-2 gating with
capacity factor, modeling_gpt_moe.py — 768 hidden, 8 layers, 8 experts,
seq 1024 in pretrain_moe.json). Expert parallelism is the planner's
all-to-all reshard executed over RCCL/xGMI (the reference's
kDAPPLEAllToAll, SURVEY.md §2.7 EP): tokens are permuted by expert,
exchanged with a variable-split all-to-all across the EP group, processed
by each rank's local experts through the fused linear kernels, and
returned. The gating weight stays replicated (the reference pins it with
xla_sharding.replicate, moe_layers.py:296)."""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.distributed as dist

from tepdist_amd import ops
from tepdist_amd.models.configs import MoEConfig
from tepdist_amd.models.gpt2 import GPT2Block
from tepdist_amd.models.configs import GPT2Config
from tepdist_amd.parallel.tp import ParallelEnv


class _AllToAllVar(torch.autograd.Function):
    """Variable-split all_to_all_single with autograd (backward runs the
    transposed exchange)."""

    @staticmethod
    def forward(ctx, x, out_splits, in_splits, group):
        ctx.group = group
        ctx.in_splits = in_splits
        ctx.out_splits = out_splits
        # group=None means the DEFAULT world group, not "undistributed"
        if not dist.is_initialized() or dist.get_world_size(group) == 1:
            return x
        out = x.new_empty((sum(out_splits),) + tuple(x.shape[1:]))
        dist.all_to_all_single(out, x.contiguous(),
                               output_split_sizes=out_splits,
                               input_split_sizes=in_splits, group=group)
        return out

    @staticmethod
    def backward(ctx, dy):
        if not dist.is_initialized() or \
                dist.get_world_size(ctx.group) == 1:
            return dy, None, None, None
        dx = dy.new_empty((sum(ctx.in_splits),) + tuple(dy.shape[1:]))
        dist.all_to_all_single(dx, dy.contiguous(),
                               output_split_sizes=ctx.in_splits,
                               input_split_sizes=ctx.out_splits,
                               group=ctx.group)
        return dx, None, None, None


class MoELayer(nn.Module):
    """Top-2 gated FFN experts with capacity dropping and EP all-to-all."""

    def __init__(self, d: int, num_experts: int, top_k: int = 2,
                 capacity_factor: float = 1.25,
                 env: Optional[ParallelEnv] = None, ep_group=None,
                 ep_size: int = 1, ep_rank: int = 0, dtype=torch.bfloat16):
        super().__init__()
        assert num_experts % ep_size == 0
        self.d = d
        self.E = num_experts
        self.k = top_k
        self.cf = capacity_factor
        self.ep_group = ep_group
        self.ep_size = ep_size
        self.ep_rank = ep_rank
        self.e_local = num_experts // ep_size
        self.e_start = ep_rank * self.e_local
        # gate weight replicated (reference moe_layers.py:296)
        self.w_gate = nn.Parameter(torch.empty(num_experts, d, dtype=dtype))
        # local experts' FFN weights
        self.w1 = nn.Parameter(torch.empty(self.e_local, 4 * d, d,
                                           dtype=dtype))
        self.b1 = nn.Parameter(torch.zeros(self.e_local, 4 * d, dtype=dtype))
        self.w2 = nn.Parameter(torch.empty(self.e_local, d, 4 * d,
                                           dtype=dtype))
        self.b2 = nn.Parameter(torch.zeros(self.e_local, d, dtype=dtype))
        self.aux_loss = torch.zeros(())

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        """STATIC-SHAPE GShard dispatch (hipGraph-capture-safe): every
        step exchanges exactly E x capacity padded slots — no .item() /
        .tolist() host syncs, no data-dependent tensor shapes. Dropped
        tokens (over capacity) contribute zero; padded slots compute
        garbage that is never read (their combine weights are zero)."""
        B, S, d = x.shape
        T = B * S
        xt = x.reshape(T, d)
        logits = ops.linear(xt, self.w_gate)             # [T, E]
        gates = ops.softmax(logits.unsqueeze(0)).squeeze(0)
        topv, topi = torch.topk(gates.float(), self.k, dim=-1)  # [T, k]
        denom = topv.sum(-1, keepdim=True).clamp_min(1e-9)
        topv = topv / denom

        # load-balancing aux loss (GShard): E * sum_e f_e * P_e
        with torch.no_grad():
            f = torch.zeros(self.E, device=x.device)
            f.scatter_add_(0, topi.reshape(-1),
                           torch.ones_like(topi.reshape(-1),
                                           dtype=torch.float32))
            f = f / (T * self.k)
        P = gates.float().mean(0)
        self.aux_loss = self.E * (f * P).sum()

        C = max(int(self.cf * T * self.k / self.E), 4)   # static capacity

        flat_e = topi.reshape(-1)                        # [T*k]
        flat_w = topv.reshape(-1)
        flat_t = torch.arange(T, device=x.device).repeat_interleave(self.k)
        # position within expert (stable sort -> rank - expert offset)
        order = torch.argsort(flat_e, stable=True)
        counts = torch.zeros(self.E, dtype=torch.long,
                             device=x.device).scatter_add_(
            0, flat_e, torch.ones_like(flat_e))
        offs = torch.cumsum(counts, 0) - counts
        r = torch.arange(flat_e.numel(), device=x.device)
        pos = torch.empty_like(r)
        pos[order] = r - offs[flat_e[order]]
        keep = pos < C
        slot = flat_e * C + pos
        slot_safe = torch.where(keep, slot, torch.zeros_like(slot))

        # dispatch: D[E*C, d]; dropped entries accumulate ZERO into slot 0
        contrib = xt[flat_t] * keep.unsqueeze(-1).to(xt.dtype)
        D = torch.zeros(self.E * C, d, dtype=xt.dtype, device=x.device)
        D = D.index_put((slot_safe,), contrib, accumulate=True)

        if self.ep_size > 1:
            from tepdist_amd.parallel.mappings import all_to_all
            Dx = all_to_all(D, self.ep_group)            # equal splits
            recv = Dx.reshape(self.ep_size, self.e_local, C, d)
        else:
            recv = D.reshape(1, self.E, C, d)[:, self.e_start:
                                              self.e_start + self.e_local]

        # local experts over padded batches [src_ranks * C, d]
        ys = []
        for e in range(self.e_local):
            xe = recv[:, e].reshape(-1, d).contiguous()
            ys.append(ops.mlp(xe, self.w1[e], self.b1[e],
                              self.w2[e], self.b2[e]))
        Y = torch.stack(ys, dim=0)                       # [e_local, src*C, d]
        Y = Y.reshape(self.e_local, -1, C, d).transpose(0, 1)  # [src, e_l, C, d]

        if self.ep_size > 1:
            from tepdist_amd.parallel.mappings import all_to_all
            back = all_to_all(Y.reshape(self.ep_size, -1, d).reshape(
                self.ep_size * self.e_local * C, d).contiguous(),
                self.ep_group)
        else:
            back = Y.reshape(self.E * C, d)

        # combine: gather each (token, slot)'s expert output, weight, sum
        gathered = back[slot_safe] *             (flat_w * keep.to(flat_w.dtype)).unsqueeze(-1).to(back.dtype)
        out = torch.zeros_like(xt).index_add(0, flat_t,
                                             gathered.to(xt.dtype))
        return out.reshape(B, S, d)


class GPTMoEBlock(nn.Module):
    """Transformer block whose MLP is a MoE layer."""

    def __init__(self, cfg: MoEConfig, moe: bool, env: ParallelEnv,
                 ep_group=None, ep_size=1, ep_rank=0, dtype=torch.bfloat16):
        super().__init__()
        gcfg = GPT2Config(name="moe-inner", n_layer=cfg.n_layer,
                          n_embd=cfg.n_embd, n_head=cfg.n_head,
                          n_ctx=cfg.n_ctx, vocab_size=cfg.vocab_size,
                          ln_eps=cfg.ln_eps)
        self.inner = GPT2Block(gcfg, dtype, env)
        self.moe = None
        if moe:
            d = cfg.n_embd
            self.ln_moe_g = nn.Parameter(torch.ones(d, dtype=dtype))
            self.ln_moe_b = nn.Parameter(torch.zeros(d, dtype=dtype))
            self.moe = MoELayer(d, cfg.num_experts, cfg.top_k,
                                cfg.capacity_factor, env, ep_group, ep_size,
                                ep_rank, dtype)

    def forward(self, x):
        x = self.inner(x)
        if self.moe is not None:
            h = ops.layernorm(x, self.ln_moe_g, self.ln_moe_b)
            x = x + self.moe(h)
        return x


class GPTMoE(nn.Module):
    def __init__(self, cfg: MoEConfig, dtype=torch.bfloat16,
                 env: Optional[ParallelEnv] = None, ep_group=None,
                 ep_size: int = 1, ep_rank: int = 0):
        super().__init__()
        self.cfg = cfg
        self.env = env or ParallelEnv.single()
        V, d = cfg.padded_vocab, cfg.n_embd
        self.wte = nn.Parameter(torch.empty(V, d, dtype=dtype))
        self.wpe = nn.Parameter(torch.empty(cfg.n_ctx, d, dtype=dtype))
        self.blocks = nn.ModuleList(
            GPTMoEBlock(cfg, (l + 1) % cfg.moe_every == 0, self.env,
                        ep_group, ep_size, ep_rank, dtype)
            for l in range(cfg.n_layer))
        self.lnf_g = nn.Parameter(torch.ones(d, dtype=dtype))
        self.lnf_b = nn.Parameter(torch.zeros(d, dtype=dtype))
        self.aux_weight = 0.01
        self.reset_parameters()

    @torch.no_grad()
    @torch.no_grad()
    def reset_parameters(self, seed: int = 1234):
        """Shard-aware (counter RNG): EP ranks draw DIFFERENT experts —
        their w1/w2 shards are slices of one global [E, ...] tensor (the
        per-rank-identical seeding would duplicate experts across the EP
        group, the ADVICE r1 bug class); TP shards follow the GPT-2
        convention via models.gpt2._draw."""
        from tepdist_amd.models.gpt2 import _draw
        cfg, env = self.cfg, self.env
        tp, r = env.tp_size, env.tp_rank
        V, d, H = cfg.padded_vocab, cfg.n_embd, cfg.n_head
        hd = d // H
        std = 0.02
        dt = self.wte.dtype
        self.wte.copy_(_draw("wte", (V, d), std, dt, seed))
        self.wpe.copy_(_draw("wpe", (cfg.n_ctx, d), std, dt, seed))
        for i, blk in enumerate(self.blocks):
            b = blk.inner
            qkv = _draw(f"h{i}.w_qkv", (3, H, hd * d), std, dt, seed,
                        1, r, tp).reshape(3 * (H // tp) * hd, d)
            proj = _draw(f"h{i}.w_proj", (d, d), std, dt, seed, 1, r, tp)
            fc = _draw(f"h{i}.w_fc", (4 * d, d), std, dt, seed, 0, r, tp)
            out = _draw(f"h{i}.w_out", (d, 4 * d), std, dt, seed, 1, r, tp)
            if tp == 1:
                b.w_qkv.copy_(qkv); b.w_proj.copy_(proj)
                b.w_fc.copy_(fc); b.w_out.copy_(out)
            else:
                b.qkv.weight.copy_(qkv); b.proj.weight.copy_(proj)
                b.fc.weight.copy_(fc); b.out.weight.copy_(out)
            m = blk.moe
            if m is not None:
                # gate replicated (reference pins it replicated); experts
                # sharded on the GLOBAL expert dim across the EP group
                m.w_gate.copy_(_draw(f"h{i}.moe.gate", (m.E, d), std, dt,
                                     seed))
                m.w1.copy_(_draw(f"h{i}.moe.w1", (m.E, 4 * d, d), std, dt,
                                 seed, 0, m.ep_rank, m.ep_size))
                m.w2.copy_(_draw(f"h{i}.moe.w2", (m.E, d, 4 * d), std, dt,
                                 seed, 0, m.ep_rank, m.ep_size))
        self.wte[self.cfg.vocab_size:].zero_()

    def forward(self, input_ids, labels=None):
        B, S = input_ids.shape
        pos = torch.arange(S, device=input_ids.device)
        x = ops.embedding(input_ids, self.wte) + ops.embedding(pos, self.wpe)
        for blk in self.blocks:
            x = blk(x)
        x = ops.layernorm(x, self.lnf_g, self.lnf_b, self.cfg.ln_eps)
        logits = ops.linear(x, self.wte)
        if labels is None:
            return logits
        loss = ops.cross_entropy(logits.reshape(-1, logits.shape[-1]),
                                 labels.reshape(-1), ignore_index=-1)
        aux = sum(blk.moe.aux_loss for blk in self.blocks
                  if blk.moe is not None)
        loss = loss + self.aux_weight * aux
        # drop the layers' live references to this step's autograd graph:
        # a retained aux_loss keeps last step's AccumulateGrad nodes alive
        # on the default stream, which breaks (segfaults) hipGraph capture
        # of the next step
        for blk in self.blocks:
            if blk.moe is not None:
                blk.moe.aux_loss = blk.moe.aux_loss.detach()
        return loss


# -- functional static-capacity dispatch/combine (the IR ops' semantics) ----

def static_dispatch(x: torch.Tensor, gates: torch.Tensor, k: int,
                    capacity: int):
    """GShard dispatch with BLOCKED capacity: x [T, d], gates [T, E] ->
    D [E, C, d]. Deterministic in (x, gates); autograd flows through the
    gather/scatter and the combine weights. Shared by MoELayer and the
    planner IR's moe_dispatch op (capacity is per token GROUP, so a
    token-sharded dispatch is exactly the capacity-dim narrow of the
    grouped tensor — what makes expert parallelism a planner reshard)."""
    T, d = x.shape
    E = gates.shape[1]
    topv, topi = torch.topk(gates.float(), k, dim=-1)
    topv = topv / topv.sum(-1, keepdim=True).clamp_min(1e-9)
    flat_e = topi.reshape(-1)
    flat_t = torch.arange(T, device=x.device).repeat_interleave(k)
    order = torch.argsort(flat_e, stable=True)
    counts = torch.zeros(E, dtype=torch.long, device=x.device).scatter_add_(
        0, flat_e, torch.ones_like(flat_e))
    offs = torch.cumsum(counts, 0) - counts
    r = torch.arange(flat_e.numel(), device=x.device)
    pos = torch.empty_like(r)
    pos[order] = r - offs[flat_e[order]]
    keep = pos < capacity
    slot = flat_e * capacity + pos
    slot_safe = torch.where(keep, slot, torch.zeros_like(slot))
    contrib = x[flat_t] * keep.unsqueeze(-1).to(x.dtype)
    D = torch.zeros(E * capacity, d, dtype=x.dtype, device=x.device)
    D = D.index_put((slot_safe,), contrib, accumulate=True)
    return D.reshape(E, capacity, d), (flat_t, slot_safe, keep,
                                       topv.reshape(-1))


def static_combine(y: torch.Tensor, x: torch.Tensor, gates: torch.Tensor,
                   k: int):
    """Inverse of static_dispatch: y [E, C, d] expert outputs -> [T, d]
    (recomputes the deterministic slot map from gates)."""
    E, C, d = y.shape
    _, (flat_t, slot_safe, keep, flat_w) = static_dispatch(x, gates, k, C)
    back = y.reshape(E * C, d)
    gathered = back[slot_safe] * \
        (flat_w * keep.to(flat_w.dtype)).unsqueeze(-1).to(back.dtype)
    return torch.zeros_like(x).index_add(0, flat_t, gathered.to(x.dtype))
