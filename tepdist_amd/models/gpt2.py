"""GPT-2 built on tepdist_amd.ops (the planner-sharded op set).

Semantics follow the reference example's model
(/root/reference/examples/GPT2/models/gpt2/ — pre-LN transformer, gelu MLP,
tied embedding / LM head, learned positional embeddings), re-implemented
natively on our op layer: every matmul / layernorm / softmax / embedding /
cross-entropy call dispatches to a hand-written CDNA4 HIP kernel on GPU.

Tensor parallelism: pass a ParallelEnv with tp_size>1 and the model builds
with Megatron-style sharded layers (qkv column-parallel split per head, proj
row-parallel, MLP column+row, vocab-parallel embedding and cross entropy) —
the execution form of the auto-planner's tensor sharding strategies.

Weights are bf16; optimizer keeps fp32 masters (see train/optim.py).
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn

from tepdist_amd import ops
from tepdist_amd.models.configs import GPT2Config
from tepdist_amd.parallel.tp import (
    ColumnParallelLinear,
    ParallelEnv,
    RowParallelLinear,
    VocabParallelEmbedding,
    vocab_parallel_cross_entropy,
)
from tepdist_amd.runtime.initializers import InitSpec, init_shard


def _draw(name: str, full_shape, std: float, dtype, seed: int,
          shard_dim: int = -1, shard_index: int = 0, num_shards: int = 1):
    """Shard-aware weight draw: rank r's shard is bit-identical to slicing
    the full tensor (counter-based global-index RNG, runtime/initializers).
    Under tensor parallelism each rank therefore gets a DIFFERENT slice of
    the same global tensor — heads/neurons are not duplicated across the
    TP group (identical per-rank seeding silently shrank the
    effective width by 1/tp)."""
    return init_shard(name, tuple(full_shape),
                      InitSpec("random_normal", std=std), global_seed=seed,
                      shard_dim=shard_dim, shard_index=shard_index,
                      num_shards=num_shards, dtype=dtype)


class GPT2Block(nn.Module):
    def __init__(self, cfg: GPT2Config, dtype=torch.bfloat16,
                 env: Optional[ParallelEnv] = None):
        super().__init__()
        env = env or ParallelEnv.single()
        d = cfg.n_embd
        tp = env.tp_size
        assert cfg.n_head % tp == 0, "n_head must divide tp_size"
        self.cfg = cfg
        self.env = env
        self.n_head_local = cfg.n_head // tp
        self.d_local = d // tp
        self.ln1_g = nn.Parameter(torch.ones(d, dtype=dtype))
        self.ln1_b = nn.Parameter(torch.zeros(d, dtype=dtype))
        self.ln2_g = nn.Parameter(torch.ones(d, dtype=dtype))
        self.ln2_b = nn.Parameter(torch.zeros(d, dtype=dtype))
        if tp == 1:
            # weights in [out, in] layout (ops.linear computes x @ w^T)
            self.w_qkv = nn.Parameter(torch.empty(3 * d, d, dtype=dtype))
            self.b_qkv = nn.Parameter(torch.zeros(3 * d, dtype=dtype))
            self.w_proj = nn.Parameter(torch.empty(d, d, dtype=dtype))
            self.b_proj = nn.Parameter(torch.zeros(d, dtype=dtype))
            self.w_fc = nn.Parameter(torch.empty(4 * d, d, dtype=dtype))
            self.b_fc = nn.Parameter(torch.zeros(4 * d, dtype=dtype))
            self.w_out = nn.Parameter(torch.empty(d, 4 * d, dtype=dtype))
            self.b_out = nn.Parameter(torch.zeros(d, dtype=dtype))
        else:
            # qkv shard is [3, H/tp, hd, d] flattened: q,k,v of THIS rank's
            # heads (see shard_qkv_weight for the master->shard mapping)
            self.qkv = ColumnParallelLinear(d, 3 * d, env, bias=True,
                                            dtype=dtype)
            self.proj = RowParallelLinear(d, d, env, bias=True, dtype=dtype)
            self.fc = ColumnParallelLinear(d, 4 * d, env, bias=True,
                                           act="gelu", dtype=dtype)
            self.out = RowParallelLinear(4 * d, d, env, bias=True, dtype=dtype)

    def _attn(self, qkv: torch.Tensor, B: int, S: int) -> torch.Tensor:
        if getattr(self, "attn_impl", None) is not None:
            # override hook (context parallelism swaps in ring attention)
            return self.attn_impl(qkv)
        # packed-qkv fused attention: no transpose copies on the GPU path
        return ops.attention_qkv(qkv, self.n_head_local, causal=True)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, S, d = x.shape
        h = ops.layernorm(x, self.ln1_g, self.ln1_b, self.cfg.ln_eps)
        if self.env.tp_size == 1:
            qkv = ops.linear(h, self.w_qkv, self.b_qkv)
            a = self._attn(qkv, B, S)
            # fused residual+ln (one kernel) and fused MLP on GPU
            x, h = ops.add_layernorm(
                x, ops.linear(a, self.w_proj, self.b_proj),
                self.ln2_g, self.ln2_b, self.cfg.ln_eps)
            x = x + ops.mlp(h, self.w_fc, self.b_fc, self.w_out, self.b_out)
        else:
            qkv = self.qkv(h)                     # [B,S,3*d/tp]
            a = self._attn(qkv, B, S)
            x = x + self.proj(a)
            h = ops.layernorm(x, self.ln2_g, self.ln2_b, self.cfg.ln_eps)
            x = x + self.out(self.fc(h))
        return x


class GPT2(nn.Module):
    def __init__(self, cfg: GPT2Config, dtype=torch.bfloat16,
                 env: Optional[ParallelEnv] = None):
        super().__init__()
        self.cfg = cfg
        self.env = env or ParallelEnv.single()
        V, d = cfg.padded_vocab, cfg.n_embd
        if self.env.tp_size == 1:
            self.wte = nn.Parameter(torch.empty(V, d, dtype=dtype))
        else:
            self.wte_mod = VocabParallelEmbedding(V, d, self.env, dtype=dtype)
        self.wpe = nn.Parameter(torch.empty(cfg.n_ctx, d, dtype=dtype))
        self.blocks = nn.ModuleList(
            GPT2Block(cfg, dtype, self.env) for _ in range(cfg.n_layer))
        self.lnf_g = nn.Parameter(torch.ones(d, dtype=dtype))
        self.lnf_b = nn.Parameter(torch.zeros(d, dtype=dtype))
        self.reset_parameters()

    @torch.no_grad()
    def reset_parameters(self, seed: int = 1234):
        cfg, env = self.cfg, self.env
        tp, r = env.tp_size, env.tp_rank
        V, d, H = cfg.padded_vocab, cfg.n_embd, cfg.n_head
        hd = d // H
        std = 0.02
        proj_std = std / math.sqrt(2 * cfg.n_layer)
        dt = self.wpe.dtype
        if tp == 1:
            self.wte.copy_(_draw("wte", (V, d), std, dt, seed))
        else:
            self.wte_mod.weight.copy_(_draw("wte", (V, d), std, dt, seed,
                                            0, r, tp))
        self.wpe.copy_(_draw("wpe", (cfg.n_ctx, d), std, dt, seed))
        for i, blk in enumerate(self.blocks):
            # qkv full tensor is [3d, d] == [3, H, hd*d] flattened; the TP
            # shard takes this rank's heads (shard_qkv_weight layout)
            qkv = _draw(f"h{i}.w_qkv", (3, H, hd * d), std, dt, seed,
                        1, r, tp).reshape(3 * (H // tp) * hd, d)
            proj = _draw(f"h{i}.w_proj", (d, d), proj_std, dt, seed, 1, r, tp)
            fc = _draw(f"h{i}.w_fc", (4 * d, d), std, dt, seed, 0, r, tp)
            out = _draw(f"h{i}.w_out", (d, 4 * d), proj_std, dt, seed,
                        1, r, tp)
            if tp == 1:
                blk.w_qkv.copy_(qkv); blk.w_proj.copy_(proj)
                blk.w_fc.copy_(fc); blk.w_out.copy_(out)
            else:
                blk.qkv.weight.copy_(qkv); blk.proj.weight.copy_(proj)
                blk.fc.weight.copy_(fc); blk.out.weight.copy_(out)
        if tp == 1:
            # zero the padded vocab rows so they never win the softmax
            self.wte[self.cfg.vocab_size:].zero_()
        else:
            emb = self.wte_mod
            lo = max(0, self.cfg.vocab_size - emb.vocab_start)
            if lo < emb.vocab_local:
                emb.weight[lo:].zero_()

    def forward(self, input_ids: torch.Tensor,
                labels: Optional[torch.Tensor] = None,
                pos: Optional[torch.Tensor] = None):
        """input_ids [B, S]; labels [B, S] (next-token ids, -1 = ignore).
        Returns loss (scalar f32) if labels given, else logits [B,S,V]
        ([B,S,V/tp] local shard under tensor parallelism). `pos` overrides
        position ids (context parallelism feeds the shard's GLOBAL
        positions)."""
        B, S = input_ids.shape
        if pos is None:
            pos = torch.arange(S, device=input_ids.device)
        if self.env.tp_size == 1:
            x = ops.embedding(input_ids, self.wte) + ops.embedding(pos, self.wpe)
        else:
            x = self.wte_mod(input_ids) + ops.embedding(pos, self.wpe)
        for blk in self.blocks:
            x = blk(x)
        x = ops.layernorm(x, self.lnf_g, self.lnf_b, self.cfg.ln_eps)
        if self.env.tp_size == 1:
            logits = ops.linear(x, self.wte)  # tied LM head: x @ wte^T
            if labels is None:
                return logits
            return ops.cross_entropy(
                logits.reshape(-1, logits.shape[-1]), labels.reshape(-1),
                ignore_index=-1)
        # vocab-parallel tied LM head + cross entropy
        from tepdist_amd.parallel.mappings import copy_to_group
        x = copy_to_group(x, self.env.tp_group)
        logits_local = ops.linear(x, self.wte_mod.weight)
        if labels is None:
            return logits_local
        return vocab_parallel_cross_entropy(
            logits_local.reshape(-1, logits_local.shape[-1]),
            labels.reshape(-1), self.wte_mod.vocab_start,
            self.wte_mod.vocab_local, self.env.tp_group, ignore_index=-1)

    def flops_per_token(self) -> float:
        """Approximate training FLOPs per token (fwd+bwd, 6N + attention)."""
        cfg = self.cfg
        n = sum(p.numel() for p in self.parameters())
        return 6 * n + 12 * cfg.n_layer * cfg.n_embd * cfg.n_ctx


def shard_qkv_weight(w_qkv: torch.Tensor, b_qkv: torch.Tensor, n_head: int,
                     tp_rank: int, tp_size: int):
    """Maps a full packed qkv weight [3d, d] / bias [3d] to rank tp_rank's
    ColumnParallel shard (contiguous q,k,v of that rank's heads)."""
    d3, d = w_qkv.shape
    hd = d // n_head
    hl = n_head // tp_size
    w = w_qkv.reshape(3, n_head, hd, d)[:, tp_rank * hl:(tp_rank + 1) * hl]
    b = b_qkv.reshape(3, n_head, hd)[:, tp_rank * hl:(tp_rank + 1) * hl]
    return w.reshape(3 * hl * hd, d).contiguous(), b.reshape(-1).contiguous()


class GPT2Stage(nn.Module):
    """One pipeline stage of GPT-2 (a contiguous layer range; the first
    stage owns the embeddings, the last owns the final LN + LM head and the
    loss). The head is untied from the embedding across stages (the tied
    form only exists when both live on one stage). Composable with tensor
    parallelism via env."""

    def __init__(self, cfg: GPT2Config, layer_start: int, layer_end: int,
                 is_first: bool, is_last: bool, dtype=torch.bfloat16,
                 env: Optional[ParallelEnv] = None):
        super().__init__()
        self.cfg = cfg
        self.env = env or ParallelEnv.single()
        self.is_first, self.is_last = is_first, is_last
        self.layer_start = layer_start
        V, d = cfg.padded_vocab, cfg.n_embd
        if is_first:
            self.wte = nn.Parameter(torch.empty(V, d, dtype=dtype))
            self.wpe = nn.Parameter(torch.empty(cfg.n_ctx, d, dtype=dtype))
        self.blocks = nn.ModuleList(
            GPT2Block(cfg, dtype, self.env)
            for _ in range(layer_start, layer_end))
        if is_last:
            self.lnf_g = nn.Parameter(torch.ones(d, dtype=dtype))
            self.lnf_b = nn.Parameter(torch.zeros(d, dtype=dtype))
            self.lm_head = nn.Parameter(torch.empty(V, d, dtype=dtype))
        self.reset_parameters()

    @torch.no_grad()
    def reset_parameters(self, seed: int = 1234):
        """Blocks are named by GLOBAL layer index, so a stage's weights are
        bit-identical to the corresponding layers of the unsplit model
        (shard-aware under TP, same counter RNG as GPT2)."""
        cfg, env = self.cfg, self.env
        tp, r = env.tp_size, env.tp_rank
        V, d, H = cfg.padded_vocab, cfg.n_embd, cfg.n_head
        hd = d // H
        std = 0.02
        proj_std = std / math.sqrt(2 * cfg.n_layer)
        params = list(self.parameters())
        if not params:   # empty middle stage (fewer layers than stages)
            return
        dt = params[0].dtype
        if self.is_first:
            self.wte.copy_(_draw("wte", (V, d), std, dt, seed))
            self.wpe.copy_(_draw("wpe", (cfg.n_ctx, d), std, dt, seed))
        for li, blk in enumerate(self.blocks):
            i = self.layer_start + li
            qkv = _draw(f"h{i}.w_qkv", (3, H, hd * d), std, dt, seed,
                        1, r, tp).reshape(3 * (H // tp) * hd, d)
            proj = _draw(f"h{i}.w_proj", (d, d), proj_std, dt, seed, 1, r, tp)
            fc = _draw(f"h{i}.w_fc", (4 * d, d), std, dt, seed, 0, r, tp)
            out = _draw(f"h{i}.w_out", (d, 4 * d), proj_std, dt, seed,
                        1, r, tp)
            if tp == 1:
                blk.w_qkv.copy_(qkv); blk.w_proj.copy_(proj)
                blk.w_fc.copy_(fc); blk.w_out.copy_(out)
            else:
                blk.qkv.weight.copy_(qkv); blk.proj.weight.copy_(proj)
                blk.fc.weight.copy_(fc); blk.out.weight.copy_(out)
        if self.is_first:
            self.wte[self.cfg.vocab_size:].zero_()
        if self.is_last:
            # untied head (tied form only exists when wte is on this stage):
            # initialized from the SAME stream as wte so a pipeline split of
            # the tied model starts from the tied value
            self.lm_head.copy_(_draw("wte", (V, d), std, dt, seed))
            self.lm_head[self.cfg.vocab_size:].zero_()

    def forward(self, x, labels=None):
        if self.is_first:
            input_ids = x
            B, S = input_ids.shape
            pos = torch.arange(S, device=input_ids.device)
            x = ops.embedding(input_ids, self.wte) + \
                ops.embedding(pos, self.wpe)
        for blk in self.blocks:
            x = blk(x)
        if not self.is_last:
            return x
        x = ops.layernorm(x, self.lnf_g, self.lnf_b, self.cfg.ln_eps)
        logits = ops.linear(x, self.lm_head)
        if labels is None:
            return logits
        return ops.cross_entropy(logits.reshape(-1, logits.shape[-1]),
                                 labels.reshape(-1), ignore_index=-1)


def layer_ranges(n_layer: int, num_stages: int,
                 layer_stage: Optional[list] = None):
    """Contiguous [start, end) layer range per stage (from a plan's
    layer->stage map or balanced)."""
    if layer_stage is None:
        per = (n_layer + num_stages - 1) // num_stages
        return [(s * per, min((s + 1) * per, n_layer))
                for s in range(num_stages)]
    ranges = []
    for s in range(num_stages):
        ls = [l for l, st in enumerate(layer_stage) if st == s]
        ranges.append((min(ls), max(ls) + 1) if ls else (0, 0))
    return ranges
