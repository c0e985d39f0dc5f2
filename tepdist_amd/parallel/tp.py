"""Tensor-parallel layers: the execution form of the planner's tensor
sharding strategies (SURVEY.md §2.7 "TP / tensor sharding (Megatron-style)").

Weight shards follow the planner's DimStrategy output: a linear whose weight
is split on the OUTPUT dim is a ColumnParallelLinear (partial activations
concatenated or kept sharded); split on the INPUT dim is a RowParallelLinear
(partial sums all-reduced). Embedding and the LM head / cross entropy are
vocab-parallel so the largest GEMM and its gradient stay sharded end-to-end.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.distributed as dist

from tepdist_amd import ops
from tepdist_amd.parallel.mappings import (
    copy_to_group,
    reduce_from_group,
    gather_from_group,
)


@dataclass
class ParallelEnv:
    """Process-group context for one rank: dp x tp (x pp handled by the
    pipeline runtime)."""
    tp_group: object = None
    dp_group: object = None
    tp_rank: int = 0
    tp_size: int = 1
    dp_rank: int = 0
    dp_size: int = 1

    @staticmethod
    def single():
        return ParallelEnv()

    @staticmethod
    def create(tp_size: int = 1):
        """Builds dp x tp groups over the global world (tp fastest-varying,
        so TP stays inside a node over xGMI)."""
        if not dist.is_initialized():
            return ParallelEnv()
        world = dist.get_world_size()
        rank = dist.get_rank()
        assert world % tp_size == 0
        dp_size = world // tp_size
        tp_group = dp_group = None
        for d in range(dp_size):
            ranks = list(range(d * tp_size, (d + 1) * tp_size))
            g = dist.new_group(ranks)
            if rank in ranks:
                tp_group = g
        for t in range(tp_size):
            ranks = list(range(t, world, tp_size))
            g = dist.new_group(ranks)
            if rank in ranks:
                dp_group = g
        return ParallelEnv(tp_group=tp_group, dp_group=dp_group,
                           tp_rank=rank % tp_size, tp_size=tp_size,
                           dp_rank=rank // tp_size, dp_size=dp_size)


class ColumnParallelLinear(nn.Module):
    """y = x @ W^T + b with W split on the output dim across tp ranks."""

    def __init__(self, in_features: int, out_features: int, env: ParallelEnv,
                 bias: bool = True, act: str = "none",
                 gather_output: bool = False, dtype=torch.bfloat16):
        super().__init__()
        assert out_features % env.tp_size == 0
        self.env = env
        self.out_local = out_features // env.tp_size
        self.act = act
        self.gather_output = gather_output
        self.weight = nn.Parameter(torch.empty(self.out_local, in_features,
                                               dtype=dtype))
        self.bias = nn.Parameter(torch.zeros(self.out_local, dtype=dtype)) \
            if bias else None

    def forward(self, x):
        x = copy_to_group(x, self.env.tp_group)
        y = ops.linear(x, self.weight, self.bias, act=self.act)
        if self.gather_output:
            y = gather_from_group(y, self.env.tp_group, dim=-1)
        return y


class RowParallelLinear(nn.Module):
    """y = x_local @ W_local^T summed across tp ranks (input pre-sharded on
    the last dim, as produced by a ColumnParallelLinear)."""

    def __init__(self, in_features: int, out_features: int, env: ParallelEnv,
                 bias: bool = True, dtype=torch.bfloat16):
        super().__init__()
        assert in_features % env.tp_size == 0
        self.env = env
        self.in_local = in_features // env.tp_size
        self.weight = nn.Parameter(torch.empty(out_features, self.in_local,
                                               dtype=dtype))
        self.bias = nn.Parameter(torch.zeros(out_features, dtype=dtype)) \
            if bias else None

    def forward(self, x):
        y = ops.linear(x, self.weight, None)
        y = reduce_from_group(y, self.env.tp_group)
        if self.bias is not None:
            y = y + self.bias
        return y


class VocabParallelEmbedding(nn.Module):
    """Embedding table split on the vocab dim; out-of-shard ids contribute
    zero and the partials are all-reduced."""

    def __init__(self, vocab: int, dim: int, env: ParallelEnv,
                 dtype=torch.bfloat16):
        super().__init__()
        assert vocab % env.tp_size == 0
        self.env = env
        self.vocab_local = vocab // env.tp_size
        self.vocab_start = env.tp_rank * self.vocab_local
        self.weight = nn.Parameter(torch.empty(self.vocab_local, dim,
                                               dtype=dtype))

    def forward(self, ids):
        if self.env.tp_size == 1:
            return ops.embedding(ids, self.weight)
        local = ids - self.vocab_start
        in_shard = (local >= 0) & (local < self.vocab_local)
        local = local.clamp(0, self.vocab_local - 1)
        y = ops.embedding(local, self.weight)
        y = y * in_shard.unsqueeze(-1).to(y.dtype)
        return reduce_from_group(y, self.env.tp_group)


class _VocabParallelCE(torch.autograd.Function):
    """Cross entropy over vocab-sharded logits without gathering them:
    combine per-shard logsumexp across the group, pick the target logit from
    its owning shard."""

    @staticmethod
    def forward(ctx, logits_local, targets, vocab_start, vocab_local, group,
                ignore_index):
        from tepdist_amd.ops.interface import _backend
        be = _backend(logits_local)
        # local lse with no targets (every row "no-target": nll unused)
        no_tgt = torch.full_like(targets, -2)
        _, lse_local = be.cross_entropy_fwd(logits_local, no_tgt, -1)
        world = dist.get_world_size(group) if dist.is_initialized() and group is not None else 1
        if world > 1:
            lse_all = [torch.empty_like(lse_local) for _ in range(world)]
            dist.all_gather(lse_all, lse_local.contiguous(), group=group)
            lse_g = torch.logsumexp(torch.stack(lse_all), dim=0)
        else:
            lse_g = lse_local
        valid = targets != ignore_index
        local_t = targets - vocab_start
        owned = (local_t >= 0) & (local_t < vocab_local) & valid
        lt = local_t.clamp(0, vocab_local - 1)
        tl = logits_local.float().gather(
            -1, lt.unsqueeze(-1)).squeeze(-1) * owned.float()
        if world > 1:
            tl = tl.contiguous()
            dist.all_reduce(tl, group=group)
        n = valid.sum().clamp_min(1)
        loss = ((lse_g - tl) * valid.float()).sum() / n
        shard_t = torch.where(owned, lt, torch.full_like(lt, -2))
        shard_t = torch.where(valid, shard_t, torch.full_like(lt, ignore_index))
        ctx.save_for_backward(logits_local, shard_t, lse_g, n)
        ctx.ignore_index = ignore_index
        return loss

    @staticmethod
    def backward(ctx, dloss):
        logits_local, shard_t, lse_g, n = ctx.saved_tensors
        from tepdist_amd.ops.interface import _backend
        be = _backend(logits_local)
        dl = be.cross_entropy_bwd(dloss, logits_local, shard_t, lse_g,
                                  ctx.ignore_index)
        return dl, None, None, None, None, None


def vocab_parallel_cross_entropy(logits_local, targets, vocab_start,
                                 vocab_local, group, ignore_index=-1):
    return _VocabParallelCE.apply(logits_local, targets, vocab_start,
                                  vocab_local, group, ignore_index)
