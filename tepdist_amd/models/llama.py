"""Llama-family decoder (beyond the reference's model set): RMSNorm,
rotary position embedding, SwiGLU MLP, no biases, untied head — exercises
the wide-head (D=128) flash-attention path and the llama op kernels
(ops/csrc/llama_ops.hip). Same training surface as models/gpt2.py
(forward(ids, labels) -> loss) so the Trainer / bench / parallel layers
apply unchanged."""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn

from tepdist_amd import ops
from tepdist_amd.parallel.tp import (ColumnParallelLinear, ParallelEnv,
                                     RowParallelLinear,
                                     VocabParallelEmbedding,
                                     vocab_parallel_cross_entropy)


@dataclass
class LlamaConfig:
    name: str = "llama-test"
    n_layer: int = 2
    n_embd: int = 256
    n_head: int = 4          # head_dim = n_embd / n_head in {64, 128}
    n_ctx: int = 512
    vocab_size: int = 512
    ffn_mult: int = 2        # intermediate = ffn_mult * n_embd (llama ~2.7)
    rope_theta: float = 10000.0
    rms_eps: float = 1e-6


LLAMA_CONFIGS = {
    "llama-test": LlamaConfig(),
    # ~1.1B-parameter class: 2048 hidden, 16 x 128 heads (wide-head path)
    "llama-1b": LlamaConfig(name="llama-1b", n_layer=22, n_embd=2048,
                            n_head=16, n_ctx=2048, vocab_size=32000,
                            ffn_mult=3),
}


def _param(*shape, std=0.02, dtype=torch.float32):
    return nn.Parameter(torch.randn(*shape, dtype=dtype) * std)


class LlamaBlock(nn.Module):
    def __init__(self, cfg: LlamaConfig, dtype=torch.float32, env=None):
        super().__init__()
        env = env or ParallelEnv.single()
        d = cfg.n_embd
        h = cfg.ffn_mult * d
        tp = env.tp_size
        assert cfg.n_head % tp == 0
        self.cfg = cfg
        self.env = env
        self.n_head_local = cfg.n_head // tp
        self.head_dim = d // cfg.n_head
        self.ln1_g = nn.Parameter(torch.ones(d, dtype=dtype))
        self.ln2_g = nn.Parameter(torch.ones(d, dtype=dtype))
        if tp == 1:
            self.w_qkv = _param(3 * d, d, dtype=dtype)
            self.w_o = _param(d, d, dtype=dtype)
            self.w_gate = _param(h, d, dtype=dtype)
            self.w_up = _param(h, d, dtype=dtype)
            self.w_down = _param(d, h, dtype=dtype)
        else:
            # Megatron layout: qkv/gate/up column-parallel, o/down
            # row-parallel (same shard convention as the GPT-2 block)
            self.qkv = ColumnParallelLinear(d, 3 * d, env, bias=False,
                                            dtype=dtype)
            self.o = RowParallelLinear(d, d, env, bias=False, dtype=dtype)
            self.gate = ColumnParallelLinear(d, h, env, bias=False,
                                             dtype=dtype)
            self.up = ColumnParallelLinear(d, h, env, bias=False,
                                           dtype=dtype)
            self.down = RowParallelLinear(h, d, env, bias=False, dtype=dtype)

    def _attn(self, qkv, seq_len: int):
        cfg = self.cfg
        T = qkv.shape[0]
        nh, hd = self.n_head_local, self.head_dim
        dloc = nh * hd
        q, k, v = qkv.split(dloc, dim=-1)
        q = ops.rope(q.reshape(T, nh, hd), seq_len, cfg.rope_theta)
        k = ops.rope(k.reshape(T, nh, hd), seq_len, cfg.rope_theta)
        b = T // seq_len

        def heads(t):
            return t.reshape(b, seq_len, nh, hd).transpose(1, 2).contiguous()
        o = ops.attention(heads(q), heads(k),
                          heads(v.reshape(T, nh, hd)), causal=True)
        return o.transpose(1, 2).reshape(T, dloc).contiguous()

    def forward(self, x, seq_len: int):
        cfg = self.cfg
        hnorm = ops.rmsnorm(x, self.ln1_g, cfg.rms_eps)
        if self.env.tp_size == 1:
            o = self._attn(ops.linear(hnorm, self.w_qkv), seq_len)
            x = x + ops.linear(o, self.w_o)
            hn = ops.rmsnorm(x, self.ln2_g, cfg.rms_eps)
            gate = ops.linear(hn, self.w_gate)
            up = ops.linear(hn, self.w_up)
            x = x + ops.linear(ops.swiglu(gate, up), self.w_down)
        else:
            o = self._attn(self.qkv(hnorm), seq_len)
            x = x + self.o(o)
            hn = ops.rmsnorm(x, self.ln2_g, cfg.rms_eps)
            x = x + self.down(ops.swiglu(self.gate(hn), self.up(hn)))
        return x


class Llama(nn.Module):
    def __init__(self, cfg: LlamaConfig, dtype=torch.float32, env=None):
        super().__init__()
        self.cfg = cfg
        self.env = env or ParallelEnv.single()
        if self.env.tp_size == 1:
            self.wte = _param(cfg.vocab_size, cfg.n_embd, dtype=dtype)
            self.lm_head = _param(cfg.vocab_size, cfg.n_embd, dtype=dtype)
        else:
            self.wte_mod = VocabParallelEmbedding(cfg.vocab_size,
                                                  cfg.n_embd, self.env,
                                                  dtype=dtype)
            self.head = ColumnParallelLinear(cfg.n_embd, cfg.vocab_size,
                                             self.env, bias=False,
                                             dtype=dtype)
        self.blocks = nn.ModuleList(
            LlamaBlock(cfg, dtype, self.env) for _ in range(cfg.n_layer))
        self.ln_f_g = nn.Parameter(torch.ones(cfg.n_embd, dtype=dtype))
        self.reset_parameters()

    @torch.no_grad()
    def reset_parameters(self, seed: int = 1234):
        """Shard-aware init (tp>1 builds left ColumnParallel /
        RowParallel weights uninitialized): every weight is drawn from the
        counter-based global-index RNG (runtime/initializers) so each TP
        rank holds exactly its slice of the same global tensor."""
        from tepdist_amd.models.gpt2 import _draw
        cfg, env = self.cfg, self.env
        tp, r = env.tp_size, env.tp_rank
        d, H = cfg.n_embd, cfg.n_head
        hd = d // H
        h = cfg.ffn_mult * d
        V = cfg.vocab_size
        std = 0.02
        dt = self.ln_f_g.dtype
        if tp == 1:
            self.wte.copy_(_draw("wte", (V, d), std, dt, seed))
            self.lm_head.copy_(_draw("lm_head", (V, d), std, dt, seed))
        else:
            self.wte_mod.weight.copy_(
                _draw("wte", (V, d), std, dt, seed, 0, r, tp))
            self.head.weight.copy_(
                _draw("lm_head", (V, d), std, dt, seed, 0, r, tp))
        for i, blk in enumerate(self.blocks):
            qkv = _draw(f"h{i}.w_qkv", (3, H, hd * d), std, dt, seed,
                        1, r, tp).reshape(3 * (H // tp) * hd, d)
            o = _draw(f"h{i}.w_o", (d, d), std, dt, seed, 1, r, tp)
            gate = _draw(f"h{i}.w_gate", (h, d), std, dt, seed, 0, r, tp)
            up = _draw(f"h{i}.w_up", (h, d), std, dt, seed, 0, r, tp)
            down = _draw(f"h{i}.w_down", (d, h), std, dt, seed, 1, r, tp)
            if tp == 1:
                blk.w_qkv.copy_(qkv); blk.w_o.copy_(o)
                blk.w_gate.copy_(gate); blk.w_up.copy_(up)
                blk.w_down.copy_(down)
            else:
                blk.qkv.weight.copy_(qkv); blk.o.weight.copy_(o)
                blk.gate.weight.copy_(gate); blk.up.weight.copy_(up)
                blk.down.weight.copy_(down)

    def forward(self, ids, labels=None):
        b, s = ids.shape
        if self.env.tp_size == 1:
            x = ops.embedding(ids.reshape(-1), self.wte)
        else:
            x = self.wte_mod(ids.reshape(-1))
        for blk in self.blocks:
            x = blk(x, s)
        x = ops.rmsnorm(x, self.ln_f_g, self.cfg.rms_eps)
        if self.env.tp_size == 1:
            logits = ops.linear(x, self.lm_head)
            if labels is None:
                return logits
            return ops.cross_entropy(logits, labels.reshape(-1))
        logits_local = self.head(x)
        if labels is None:
            return logits_local
        emb = self.env
        vs = self.head.out_local * emb.tp_rank
        return vocab_parallel_cross_entropy(
            logits_local, labels.reshape(-1), vs, self.head.out_local,
            emb.tp_group)
