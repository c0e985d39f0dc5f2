"""Autograd-transparent collectives for tensor parallelism.

These are the runtime form of the planner's reshard ops (SURVEY.md §2.4:
kCustomCollective lowered to all-reduce / all-gather / all-to-all /
dynamic-slice). On ROCm, torch.distributed backend "nccl" is RCCL over xGMI;
on CPU tests the same code runs over gloo.
"""

from __future__ import annotations

import torch
import torch.distributed as dist


def _world(group):
    return dist.get_world_size(group) if dist.is_initialized() else 1


class _CopyToGroup(torch.autograd.Function):
    """Identity forward; all-reduce gradient (input broadcast to the group)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, dy):
        if _world(ctx.group) > 1:
            dy = dy.contiguous()
            dist.all_reduce(dy, group=ctx.group)
        return dy, None


class _ReduceFromGroup(torch.autograd.Function):
    """All-reduce forward (sum of partials); identity gradient."""

    @staticmethod
    def forward(ctx, x, group):
        if _world(group) > 1:
            x = x.contiguous()
            dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, dy):
        return dy, None


class _GatherFromGroup(torch.autograd.Function):
    """All-gather along `dim` forward; slice gradient back."""

    @staticmethod
    def forward(ctx, x, group, dim):
        ctx.group = group
        ctx.dim = dim
        world = _world(group)
        if world == 1:
            return x
        x = x.contiguous()
        parts = [torch.empty_like(x) for _ in range(world)]
        dist.all_gather(parts, x, group=group)
        return torch.cat(parts, dim=dim)

    @staticmethod
    def backward(ctx, dy):
        world = _world(ctx.group)
        if world == 1:
            return dy, None, None
        rank = dist.get_rank(ctx.group)
        n = dy.shape[ctx.dim] // world
        return dy.narrow(ctx.dim, rank * n, n).contiguous(), None, None


class _ScatterToGroup(torch.autograd.Function):
    """Slice along `dim` forward; all-gather gradient."""

    @staticmethod
    def forward(ctx, x, group, dim):
        ctx.group = group
        ctx.dim = dim
        world = _world(group)
        if world == 1:
            return x
        rank = dist.get_rank(group)
        n = x.shape[dim] // world
        return x.narrow(dim, rank * n, n).contiguous()

    @staticmethod
    def backward(ctx, dy):
        world = _world(ctx.group)
        if world == 1:
            return dy, None, None
        dy = dy.contiguous()
        parts = [torch.empty_like(dy) for _ in range(world)]
        dist.all_gather(parts, dy, group=ctx.group)
        return torch.cat(parts, dim=ctx.dim), None, None


class _AllToAllSingle(torch.autograd.Function):
    """all_to_all_single (MoE expert dispatch); inverse in backward."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        if _world(group) == 1:
            return x
        x = x.contiguous()
        out = torch.empty_like(x)
        dist.all_to_all_single(out, x, group=group)
        return out

    @staticmethod
    def backward(ctx, dy):
        if _world(ctx.group) == 1:
            return dy, None
        dy = dy.contiguous()
        out = torch.empty_like(dy)
        dist.all_to_all_single(out, dy, group=ctx.group)
        return out, None


def copy_to_group(x, group=None):
    return _CopyToGroup.apply(x, group)


def reduce_from_group(x, group=None):
    return _ReduceFromGroup.apply(x, group)


def gather_from_group(x, group=None, dim=-1):
    return _GatherFromGroup.apply(x, group, dim)


def scatter_to_group(x, group=None, dim=-1):
    return _ScatterToGroup.apply(x, group, dim)


def all_to_all(x, group=None):
    return _AllToAllSingle.apply(x, group)
