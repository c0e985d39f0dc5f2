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
import os
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

    # -- static-shape greedy decode step (hipGraph-capturable) -----------
    #
    # Eager per-token decode is LAUNCH-bound (~1000 kernels/token across
    # 24 layers). This step runs on fixed buffers only: embeds tok at the
    # device-side position cur_t, writes K/V into the cache at cur_t
    # (index_copy_, no host sync), attends over the FULL S_max cache with
    # a kpos > cur_t mask (exp(-inf) = 0 makes it bitwise equal to
    # slicing), argmaxes, feeds the next token back into tok, and
    # advances cur_t — so a captured graph replays autonomously.

    def _decode_step(self, tok, cur_t, kc, vc, kpos):
        cfg = self.cfg
        B = tok.shape[0]
        H, d = cfg.n_head, self.m.wte.shape[1]
        D = d // H
        scale = 1.0 / math.sqrt(D)
        x = ops.embedding(tok.reshape(-1), self.m.wte).reshape(B, 1, d) \
            + ops.embedding(cur_t, self.m.wpe)
        mask = (kpos > cur_t)                       # [S_max] device bool
        for li, blk in enumerate(self.m.blocks):
            h = ops.layernorm(x, blk.ln1_g, blk.ln1_b, cfg.ln_eps)
            qkv = ops.linear(h, blk.w_qkv, blk.b_qkv)
            q, k, v = qkv.split(d, dim=-1)
            qh = q.reshape(B, 1, H, D).transpose(1, 2)
            kc[li].index_copy_(2, cur_t,
                               k.reshape(B, 1, H, D).transpose(1, 2))
            vc[li].index_copy_(2, cur_t,
                               v.reshape(B, 1, H, D).transpose(1, 2))
            scores = ops.matmul(qh.contiguous(),
                                kc[li].transpose(-1, -2)) * scale
            scores = scores.masked_fill(mask, float("-inf"))
            p = torch.softmax(scores.float(), dim=-1).to(x.dtype)
            a = ops.matmul(p, vc[li])
            a = a.transpose(1, 2).reshape(B, 1, d)
            x = x + ops.linear(a, blk.w_proj, blk.b_proj)
            h = ops.layernorm(x, blk.ln2_g, blk.ln2_b, cfg.ln_eps)
            h = ops.linear(h, blk.w_fc, blk.b_fc, act="gelu")
            x = x + ops.linear(h, blk.w_out, blk.b_out)
        logits = self._logits(x).float()[:, -1, :cfg.vocab_size]
        tok.copy_(logits.argmax(-1, keepdim=True))
        cur_t.add_(1)

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

        # greedy decode on GPU: static-buffer steps, hipGraph-captured
        # after two eager warmups (the eager per-token loop is
        # launch-bound; replaying one recorded step removes ~1000
        # launches/token). TEPDIST_DECODE_GRAPH=0 forces eager.
        if (temperature == 0 and ids.is_cuda and S_max > S0
                and max_new_tokens >= 8
                and os.environ.get("TEPDIST_DECODE_GRAPH", "1") != "0"):
            logits = self._logits(x).float()[:, -1, :cfg.vocab_size]
            tok = logits.argmax(-1, keepdim=True)
            outs = [tok.clone()]
            cur_t = torch.full((1,), S0, device=dev, dtype=torch.long)
            kpos = torch.arange(S_max, device=dev)
            n_rest = min(max_new_tokens, S_max - S0) - 1
            graph = None
            done = 0
            try:
                side = torch.cuda.Stream()
                side.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(side):
                    for _ in range(min(2, n_rest)):
                        self._decode_step(tok, cur_t, kc, vc, kpos)
                        outs.append(tok.clone())
                        done += 1
                torch.cuda.current_stream().wait_stream(side)
                if done < n_rest:
                    graph = torch.cuda.CUDAGraph()
                    with torch.cuda.graph(graph):
                        self._decode_step(tok, cur_t, kc, vc, kpos)
            except Exception:
                graph = None           # capture failed: stay eager
            while done < n_rest:
                if graph is not None:
                    graph.replay()
                else:
                    self._decode_step(tok, cur_t, kc, vc, kpos)
                outs.append(tok.clone())
                done += 1
            return torch.cat([ids] + outs, dim=1)
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
