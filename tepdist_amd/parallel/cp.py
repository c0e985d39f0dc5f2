"""Context parallelism (sequence sharding) for GPT-2 — beyond parity.

The reference has no sequence/context parallelism; this
is the MI355X-native long-context mode built on ring attention
(`parallel/ring_attention.py`): the sequence is sharded across the CP
group and ONLY attention communicates — K/V blocks rotate between ring
neighbors (one xGMI point-to-point hop per step). Everything else in the
transformer is token-local (layernorm, MLP, embedding, cross entropy),
so it runs unchanged on the shard; parameters are replicated across CP
ranks, so gradients need a DP-style all-reduce (SUM — the loss below is
the global token mean, so local grads arrive pre-scaled; use
`GradReducer(average=False)` over the CP group).

Activation memory per rank falls by the CP degree, which is what buys
long context: S=128k on 8 GPUs holds 16k tokens per rank.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist

from tepdist_amd.models.gpt2 import GPT2, GPT2Config
from tepdist_amd.parallel.mappings import reduce_from_group
from tepdist_amd.parallel.ring_attention import ring_attention, zigzag_shard


def cp_shard(x: torch.Tensor, world: int, rank: int, dim: int = 1,
             zigzag: bool = False) -> torch.Tensor:
    """This rank's sequence shard of `x` along `dim` (contiguous block,
    or the zigzag chunk pair for causal load balance)."""
    if zigzag:
        return zigzag_shard(x, world, dim=dim)[rank].contiguous()
    n = x.shape[dim] // world
    return x.narrow(dim, rank * n, n).contiguous()


class ContextParallelGPT2(GPT2):
    """GPT-2 over a sequence-sharded batch: construct with the CP group,
    feed `cp_shard`-ed input_ids/labels plus the matching global `pos`
    (from `shard_inputs`). Returns the GLOBAL mean loss on every rank."""

    def __init__(self, cfg: GPT2Config, group=None, zigzag: bool = False,
                 dtype=torch.bfloat16):
        super().__init__(cfg, dtype=dtype)
        self.cp_group = group
        self.zigzag = zigzag
        for blk in self.blocks:
            blk.attn_impl = self._ring_attn(blk)

    def _ring_attn(self, blk):
        group, zigzag = self.cp_group, self.zigzag

        def attn(qkv: torch.Tensor) -> torch.Tensor:
            B, S, d3 = qkv.shape
            h = blk.n_head_local
            hd = d3 // 3 // h
            q, k, v = (t.contiguous() for t in
                       qkv.reshape(B, S, 3, h, hd).permute(2, 0, 3, 1, 4))
            o = ring_attention(q, k, v, group, causal=True, zigzag=zigzag)
            return o.transpose(1, 2).reshape(B, S, h * hd)
        return attn

    def shard_inputs(self, input_ids: torch.Tensor,
                     labels: Optional[torch.Tensor] = None):
        """Full-sequence batch -> (local ids, local labels, global pos)
        for this rank. Labels are next-token ids prepared by the caller
        on the FULL sequence, then sharded like the inputs — so chunk
        boundaries need no special handling."""
        world = dist.get_world_size(self.cp_group) \
            if dist.is_initialized() else 1
        rank = dist.get_rank(self.cp_group) if dist.is_initialized() else 0
        pos = torch.arange(input_ids.shape[1], device=input_ids.device)
        return (cp_shard(input_ids, world, rank, 1, self.zigzag),
                None if labels is None
                else cp_shard(labels, world, rank, 1, self.zigzag),
                cp_shard(pos, world, rank, 0, self.zigzag))

    def forward(self, input_ids, labels=None, pos=None):
        out = super().forward(input_ids, labels, pos=pos)
        if labels is None or not dist.is_initialized():
            return out
        # global token-mean loss: weight each rank's local mean by its
        # valid-token count, all-reduce (autograd-aware) the weighted sum.
        # A shard whose labels are ALL ignored contributes exactly 0 (its
        # local mean may be nan — mean over zero tokens).
        n = (labels.reshape(-1) != -1).sum().to(out.dtype)
        tot = n.clone()
        dist.all_reduce(tot, group=self.cp_group)
        local = torch.where(n > 0, out * (n / tot.clamp_min(1)),
                            torch.zeros_like(out))
        return reduce_from_group(local, self.cp_group)
