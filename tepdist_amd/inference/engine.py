"""Autoregressive generation with a KV cache (serving path; beyond the
reference's training-only scope).

Prefill runs the flash kernels over the whole prompt while recording K/V;
decode steps run one token at a time against the cache with composed
batched matmuls (at q_len = 1 the GEMV-shaped attention is bandwidth
bound — the cache layout [B, H, S_max, D] keeps the K/V reads
contiguous). The engine reads the model's parameters directly (tp = 1
layout of models/gpt2.py), so no changes to the training forward."""

from __future__ import annotations

import math
from typing import Optional

import torch

from tepdist_amd import ops


class Generator:
    def __init__(self, model, max_seq: Optional[int] = None):
        assert model.env.tp_size == 1, "generation engine: tp=1 layout"
        self.m = model
        self.cfg = model.cfg
        self.max_seq = max_seq or self.cfg.n_ctx

    # -- one transformer stack pass over new tokens x [B, T, d], with
    # caches kc/vc [L][B, H, S_max, D] filled up to `pos` -----------------

    def _stack(self, x, kc, vc, pos):
        cfg = self.cfg
        B, T, d = x.shape
        H = cfg.n_head
        D = d // H
        scale = 1.0 / math.sqrt(D)
        for li, blk in enumerate(self.m.blocks):
            h = ops.layernorm(x, blk.ln1_g, blk.ln1_b, cfg.ln_eps)
            qkv = ops.linear(h, blk.w_qkv, blk.b_qkv)      # [B,T,3d]
            q, k, v = qkv.split(d, dim=-1)

            def heads(t):
                return t.reshape(B, T, H, D).transpose(1, 2)  # [B,H,T,D]
            qh, kh, vh = heads(q), heads(k), heads(v)
            kc[li][:, :, pos:pos + T] = kh
            vc[li][:, :, pos:pos + T] = vh
            if T > 1 and pos == 0:
                a = ops.attention(qh.contiguous(), kh.contiguous(),
                                  vh.contiguous(), causal=True)
            else:
                # decode: q over the whole cache (composed, GEMV-shaped)
                kall = kc[li][:, :, :pos + T]
                vall = vc[li][:, :, :pos + T]
                scores = ops.matmul(qh.contiguous(),
                                    kall.transpose(-1, -2)) * scale
                if T > 1:  # chunked prefill continuation: causal inside
                    qpos = torch.arange(pos, pos + T, device=x.device)
                    kpos = torch.arange(pos + T, device=x.device)
                    mask = kpos[None, :] > qpos[:, None]
                    scores = scores.masked_fill(mask, float("-inf"))
                p = torch.softmax(scores.float(), dim=-1).to(vall.dtype)
                a = ops.matmul(p, vall.contiguous())
            a = a.transpose(1, 2).reshape(B, T, d)
            x = x + ops.linear(a, blk.w_proj, blk.b_proj)
            h = ops.layernorm(x, blk.ln2_g, blk.ln2_b, cfg.ln_eps)
            h = ops.linear(h, blk.w_fc, blk.b_fc, act="gelu")
            x = x + ops.linear(h, blk.w_out, blk.b_out)
        return x

    def _logits(self, x):
        x = ops.layernorm(x, self.m.lnf_g, self.m.lnf_b, self.cfg.ln_eps)
        return ops.linear(x[:, -1:], self.m.wte)   # last position only

    @torch.no_grad()
    def generate(self, ids: torch.Tensor, max_new_tokens: int,
                 temperature: float = 0.0, top_k: int = 0,
                 seed: int = 0) -> torch.Tensor:
        cfg = self.cfg
        B, S0 = ids.shape
        H, d = cfg.n_head, cfg.n_embd
        D = d // H
        S_max = min(self.max_seq, S0 + max_new_tokens)
        dev = next(self.m.parameters()).device
        dt = next(self.m.parameters()).dtype
        kc = [torch.zeros(B, H, S_max, D, dtype=dt, device=dev)
              for _ in range(cfg.n_layer)]
        vc = [torch.zeros(B, H, S_max, D, dtype=dt, device=dev)
              for _ in range(cfg.n_layer)]
        gen = torch.Generator(device="cpu").manual_seed(seed)

        out = ids
        pos_ids = torch.arange(S0, device=dev)
        x = ops.embedding(ids, self.m.wte) + ops.embedding(pos_ids,
                                                           self.m.wpe)
        x = self._stack(x, kc, vc, 0)
        cur = S0
        for _ in range(max_new_tokens):
            if cur >= S_max:
                break
            logits = self._logits(x).float()[:, -1]       # [B, V]
            logits = logits[:, :cfg.vocab_size]            # drop pad rows
            if temperature > 0:
                logits = logits / temperature
                if top_k > 0:
                    kth = logits.topk(top_k, dim=-1).values[:, -1:]
                    logits = logits.masked_fill(logits < kth, float("-inf"))
                probs = torch.softmax(logits, dim=-1)
                nxt = torch.multinomial(probs.cpu(), 1,
                                        generator=gen).to(dev)
            else:
                nxt = logits.argmax(-1, keepdim=True)
            out = torch.cat([out, nxt], dim=1)
            pos = torch.full((1,), cur, device=dev, dtype=torch.long)
            x = ops.embedding(nxt, self.m.wte) + ops.embedding(pos,
                                                               self.m.wpe)
            x = self._stack(x, kc, vc, cur)
            cur += 1
        return out
