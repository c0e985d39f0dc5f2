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

        capacity = max(int(self.cf * T * self.k / self.E), 4)

        # flatten (token, slot) assignments, capacity-drop per expert
        flat_e = topi.reshape(-1)                        # [T*k]
        flat_w = topv.reshape(-1)
        flat_t = torch.arange(T, device=x.device).repeat_interleave(self.k)
        order = torch.argsort(flat_e, stable=True)
        fe, fw, ft = flat_e[order], flat_w[order], flat_t[order]
        # position within expert
        ones = torch.ones_like(fe)
        counts = torch.zeros(self.E, dtype=torch.long,
                             device=x.device).scatter_add_(0, fe, ones)
        offs = torch.cumsum(counts, 0) - counts
        pos = torch.arange(fe.numel(), device=x.device) - offs[fe]
        keep = pos < capacity
        fe, fw, ft = fe[keep], fw[keep], ft[keep]

        # gather tokens in expert order, exchange across EP ranks
        send = xt[ft]                                    # [n_send, d]
        kept_counts = torch.zeros(self.E, dtype=torch.long,
                                  device=x.device).scatter_add_(
            0, fe, torch.ones_like(fe))
        per_rank = kept_counts.reshape(self.ep_size, self.e_local).sum(-1)
        in_splits = per_rank.tolist()
        if self.ep_size > 1:
            ex = torch.tensor(in_splits, device=x.device)
            all_splits = [torch.zeros_like(ex) for _ in range(self.ep_size)]
            dist.all_gather(all_splits, ex, group=self.ep_group)
            out_splits = [int(s[self.ep_rank].item()) for s in all_splits]
            recv = _AllToAllVar.apply(send, out_splits, in_splits,
                                      self.ep_group)
            # exchange per-expert counts for exact segmentation
            pe = kept_counts.reshape(self.ep_size, self.e_local).contiguous()
            pe_all = torch.empty_like(pe)
            dist.all_to_all_single(pe_all, pe, group=self.ep_group)
            seg = pe_all  # [src_rank, local_expert]
        else:
            recv = send
            seg = kept_counts.reshape(1, self.E)[:, self.e_start:
                                                 self.e_start + self.e_local]

        # process: received tokens are grouped rank-major, expert-minor;
        # rebuild expert-contiguous batches (out-of-place index_add keeps
        # autograd through the expert FFNs)
        cursor = 0
        segs = []
        for r in range(seg.shape[0]):
            for e in range(self.e_local):
                c = int(seg[r, e].item())
                segs.append((e, cursor, c))
                cursor += c
        sels, ys = [], []
        for e in range(self.e_local):
            idxs = [torch.arange(st, st + c, device=x.device)
                    for (ee, st, c) in segs if ee == e and c > 0]
            if not idxs:
                continue
            sel = torch.cat(idxs)
            h = ops.linear(recv[sel], self.w1[e], self.b1[e], act="gelu")
            y = ops.linear(h, self.w2[e], self.b2[e])
            sels.append(sel)
            ys.append(y)
        if sels:
            outs = torch.zeros_like(recv).index_add(
                0, torch.cat(sels), torch.cat(ys).to(recv.dtype))
        else:
            outs = torch.zeros_like(recv)

        if self.ep_size > 1:
            back = _AllToAllVar.apply(outs, in_splits, out_splits,
                                      self.ep_group)
        else:
            back = outs

        # combine: scatter back to tokens with gate weights
        out = torch.zeros_like(xt).index_add(
            0, ft, back * fw.unsqueeze(-1).to(back.dtype))
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
    def reset_parameters(self, seed: int = 1234):
        g = torch.Generator().manual_seed(seed)
        for name, p in self.named_parameters():
            if p.dim() >= 2:
                p.copy_(torch.randn(p.shape, generator=g) * 0.02)
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
        return loss + self.aux_weight * aux
