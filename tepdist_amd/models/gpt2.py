"""GPT-2 built on tepdist_amd.ops (the planner-sharded op set).

Semantics follow the reference example's model
(/root/reference/examples/GPT2/models/gpt2/ — pre-LN transformer, gelu MLP,
tied embedding / LM head, learned positional embeddings), re-implemented
natively on our op layer: every matmul / layernorm / softmax / embedding /
cross-entropy call dispatches to a hand-written CDNA4 HIP kernel on GPU.

Weights are bf16; optimizer keeps fp32 masters (see train/optim.py).
"""

from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn

from tepdist_amd import ops
from tepdist_amd.models.configs import GPT2Config


class GPT2Block(nn.Module):
    def __init__(self, cfg: GPT2Config, dtype=torch.bfloat16):
        super().__init__()
        d = cfg.n_embd
        self.cfg = cfg
        self.n_head = cfg.n_head
        self.ln1_g = nn.Parameter(torch.ones(d, dtype=dtype))
        self.ln1_b = nn.Parameter(torch.zeros(d, dtype=dtype))
        self.ln2_g = nn.Parameter(torch.ones(d, dtype=dtype))
        self.ln2_b = nn.Parameter(torch.zeros(d, dtype=dtype))
        # weights in [out, in] layout (ops.linear computes x @ w^T)
        self.w_qkv = nn.Parameter(torch.empty(3 * d, d, dtype=dtype))
        self.b_qkv = nn.Parameter(torch.zeros(3 * d, dtype=dtype))
        self.w_proj = nn.Parameter(torch.empty(d, d, dtype=dtype))
        self.b_proj = nn.Parameter(torch.zeros(d, dtype=dtype))
        self.w_fc = nn.Parameter(torch.empty(4 * d, d, dtype=dtype))
        self.b_fc = nn.Parameter(torch.zeros(4 * d, dtype=dtype))
        self.w_out = nn.Parameter(torch.empty(d, 4 * d, dtype=dtype))
        self.b_out = nn.Parameter(torch.zeros(d, dtype=dtype))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        B, S, d = x.shape
        H = self.n_head
        hd = d // H

        h = ops.layernorm(x, self.ln1_g, self.ln1_b, self.cfg.ln_eps)
        qkv = ops.linear(h, self.w_qkv, self.b_qkv)          # [B,S,3d]
        q, k, v = qkv.split(d, dim=-1)
        q = q.reshape(B, S, H, hd).transpose(1, 2).contiguous()
        k = k.reshape(B, S, H, hd).transpose(1, 2).contiguous()
        v = v.reshape(B, S, H, hd).transpose(1, 2).contiguous()
        a = ops.attention(q, k, v, causal=True)              # [B,H,S,hd]
        a = a.transpose(1, 2).reshape(B, S, d).contiguous()
        x = x + ops.linear(a, self.w_proj, self.b_proj)

        h = ops.layernorm(x, self.ln2_g, self.ln2_b, self.cfg.ln_eps)
        h = ops.linear(h, self.w_fc, self.b_fc, act="gelu")  # fused bias+gelu
        x = x + ops.linear(h, self.w_out, self.b_out)
        return x


class GPT2(nn.Module):
    def __init__(self, cfg: GPT2Config, dtype=torch.bfloat16):
        super().__init__()
        self.cfg = cfg
        V, d = cfg.padded_vocab, cfg.n_embd
        self.wte = nn.Parameter(torch.empty(V, d, dtype=dtype))
        self.wpe = nn.Parameter(torch.empty(cfg.n_ctx, d, dtype=dtype))
        self.blocks = nn.ModuleList(GPT2Block(cfg, dtype) for _ in range(cfg.n_layer))
        self.lnf_g = nn.Parameter(torch.ones(d, dtype=dtype))
        self.lnf_b = nn.Parameter(torch.zeros(d, dtype=dtype))
        self.reset_parameters()

    @torch.no_grad()
    def reset_parameters(self, seed: int = 1234):
        g = torch.Generator().manual_seed(seed)
        std = 0.02
        proj_std = std / math.sqrt(2 * self.cfg.n_layer)
        for name, p in self.named_parameters():
            if p.dim() == 2:
                s = proj_std if ("w_proj" in name or "w_out" in name) else std
                p.copy_(torch.randn(p.shape, generator=g) * s)
            elif name.endswith("_b") or "b_" in name:
                pass  # biases stay zero
        # zero the padded vocab rows so they never win the softmax
        self.wte[self.cfg.vocab_size:].zero_()

    def forward(self, input_ids: torch.Tensor,
                labels: Optional[torch.Tensor] = None):
        """input_ids [B, S]; labels [B, S] (next-token ids, -1 = ignore).
        Returns loss (scalar f32) if labels given, else logits [B,S,V]."""
        B, S = input_ids.shape
        pos = torch.arange(S, device=input_ids.device)
        x = ops.embedding(input_ids, self.wte) + ops.embedding(pos, self.wpe)
        for blk in self.blocks:
            x = blk(x)
        x = ops.layernorm(x, self.lnf_g, self.lnf_b, self.cfg.ln_eps)
        logits = ops.linear(x, self.wte)  # tied LM head: x @ wte^T
        if labels is None:
            return logits
        loss = ops.cross_entropy(
            logits.reshape(-1, logits.shape[-1]), labels.reshape(-1),
            ignore_index=-1)
        return loss

    def flops_per_token(self) -> float:
        """Approximate training FLOPs per token (fwd+bwd, 6N + attention)."""
        cfg = self.cfg
        n = sum(p.numel() for p in self.parameters()) - cfg.padded_vocab * cfg.n_embd
        # 6*N matmul flops + attention 12*L*d*S
        return 6 * (n + cfg.padded_vocab * cfg.n_embd) + \
            12 * cfg.n_layer * cfg.n_embd * cfg.n_ctx
