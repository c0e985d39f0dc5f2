"""Autograd-facing op layer with device dispatch.

Every op the planner shards goes through here. Dispatch:
  - CUDA (= ROCm/MI355X) tensors -> tepdist_amd.ops.hip (hand-written CDNA4
    HIP kernels; raises ImportError loudly if the extension is missing on a
    GPU machine — no silent eager fallback).
  - CPU tensors -> tepdist_amd.ops.reference (pure torch; used by CPU tests
    and as the numerics baseline).
"""

from __future__ import annotations

from typing import Optional

import torch

from tepdist_amd.ops import reference as ref


def _backend(t: torch.Tensor):
    if t.is_cuda:
        from tepdist_amd.ops import hip as be  # loud ImportError if .so missing
        return be
    return ref


# --------------------------------------------------------------------------


class LinearFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, bias, act: str):
        be = _backend(x)
        x2d = x.reshape(-1, x.shape[-1])
        y, pre_act = be.linear_fwd(x2d, w, bias, act)
        ctx.save_for_backward(x2d, w, pre_act if pre_act is not None else torch.empty(0))
        ctx.has_bias = bias is not None
        ctx.act = act
        ctx.in_shape = x.shape
        return y.reshape(*x.shape[:-1], w.shape[0])

    @staticmethod
    def backward(ctx, dy):
        x2d, w, pre_act = ctx.saved_tensors
        if ctx.act == "none":
            pre_act = None
        be = _backend(dy)
        dy2d = dy.reshape(-1, dy.shape[-1]).contiguous()
        dx, dw, db = be.linear_bwd(dy2d, x2d, w, ctx.has_bias, ctx.act, pre_act)
        return dx.reshape(ctx.in_shape), dw, db, None


def linear(x: torch.Tensor, w: torch.Tensor, bias: Optional[torch.Tensor] = None,
           act: str = "none") -> torch.Tensor:
    """y = act(x @ w^T + bias); w in [out, in] layout."""
    return LinearFn.apply(x, w, bias, act)


class MlpFn(torch.autograd.Function):
    """Fused transformer MLP: gelu epilogues ride the producing GEMMs
    (hip.mlp_fwd/mlp_bwd — hipBLASLt GELU_AUX_BIAS / DGELU_BGRAD, measured
    per shape against the composed path)."""

    @staticmethod
    def forward(ctx, x, w1, b1, w2, b2):
        be = _backend(x)
        x2d = x.reshape(-1, x.shape[-1]).contiguous()
        y, (h, pre) = be.mlp_fwd(x2d, w1, b1, w2, b2)
        ctx.save_for_backward(x2d, w1, w2, h, pre)
        ctx.in_shape = x.shape
        return y.reshape(*x.shape[:-1], w2.shape[0])

    @staticmethod
    def backward(ctx, dy):
        x2d, w1, w2, h, pre = ctx.saved_tensors
        be = _backend(dy)
        dy2d = dy.reshape(-1, dy.shape[-1]).contiguous()
        dx, dw1, db1, dw2, db2 = be.mlp_bwd(dy2d, x2d, w1, w2, h, pre)
        return dx.reshape(ctx.in_shape), dw1, db1, dw2, db2


def mlp(x, w1, b1, w2, b2):
    """Transformer MLP y = gelu(x@w1^T+b1)@w2^T+b2 (both biases
    required). GPU runs the fused-epilogue path; CPU composes the
    reference linears (same autograd surface)."""
    if x.is_cuda:
        return MlpFn.apply(x, w1, b1, w2, b2)
    return linear(linear(x, w1, b1, act="gelu"), w2, b2)


class MatmulFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b):
        be = _backend(a)
        ctx.save_for_backward(a, b)
        return be.matmul(a, b)

    @staticmethod
    def backward(ctx, dc):
        a, b = ctx.saved_tensors
        be = _backend(dc)
        dc = dc.contiguous()
        da = be.matmul(dc, b.transpose(-1, -2))
        db = be.matmul(a.transpose(-1, -2), dc)
        return da, db


def matmul(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    return MatmulFn.apply(a, b)


# --------------------------------------------------------------------------


class LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, beta, eps):
        be = _backend(x)
        x2d = x.reshape(-1, x.shape[-1])
        y, mean, rstd = be.layernorm_fwd(x2d, gamma, beta, eps)
        ctx.save_for_backward(x2d, gamma, mean, rstd)
        ctx.in_shape = x.shape
        return y.reshape(x.shape)

    @staticmethod
    def backward(ctx, dy):
        x2d, gamma, mean, rstd = ctx.saved_tensors
        be = _backend(dy)
        dy2d = dy.reshape(-1, dy.shape[-1]).contiguous()
        dx, dgamma, dbeta = be.layernorm_bwd(dy2d, x2d, gamma, mean, rstd)
        return dx.reshape(ctx.in_shape), dgamma, dbeta, None


def layernorm(x, gamma, beta, eps: float = 1e-5):
    return LayerNormFn.apply(x, gamma, beta, eps)


# --------------------------------------------------------------------------


class AddLayerNormFn(torch.autograd.Function):
    """Fused residual add + layernorm: one kernel produces BOTH the
    residual sum (bf16-rounded once, bit-identical to a separate add) and
    the normalized output; backward folds the straight-through gradient
    into the layernorm dx pass (kills the autograd grad-sum add)."""

    @staticmethod
    def forward(ctx, x, res, gamma, beta, eps):
        ctx.set_materialize_grads(False)
        be = _backend(x)
        x2d = x.reshape(-1, x.shape[-1])
        r2d = res.reshape(-1, res.shape[-1])
        s, y, mean, rstd = be.add_layernorm_fwd(x2d, r2d, gamma, beta, eps)
        ctx.save_for_backward(s, gamma, mean, rstd)
        ctx.in_shape = x.shape
        return s.reshape(x.shape), y.reshape(x.shape)

    @staticmethod
    def backward(ctx, dsum, dy):
        s, gamma, mean, rstd = ctx.saved_tensors
        be = _backend(dy if dy is not None else dsum)
        if dy is None:
            # normalized output unused: grads flow through the sum only
            d = dsum.reshape(ctx.in_shape)
            return d, d, None, None, None
        dy2d = dy.reshape(-1, dy.shape[-1])
        ds2d = None if dsum is None else dsum.reshape(-1, dsum.shape[-1])
        dx, dg, db = be.add_layernorm_bwd(dy2d, ds2d, s, gamma, mean, rstd)
        dx = dx.reshape(ctx.in_shape)
        return dx, dx, dg, db, None


def add_layernorm(x, res, gamma, beta, eps: float = 1e-5):
    """(s, y) = (x + res, layernorm(x + res)). Fused on GPU; composed on
    CPU with identical autograd semantics."""
    if x.is_cuda:
        return AddLayerNormFn.apply(x, res, gamma, beta, eps)
    s = x + res
    return s, layernorm(s, gamma, beta, eps)


class SoftmaxFn(torch.autograd.Function):
    """Fused scale + optional causal mask + softmax over the last dim."""

    @staticmethod
    def forward(ctx, scores, scale, causal):
        be = _backend(scores)
        p = be.softmax_fwd(scores, scale=scale, causal=causal)
        ctx.save_for_backward(p)
        ctx.scale = scale
        return p

    @staticmethod
    def backward(ctx, dp):
        (p,) = ctx.saved_tensors
        be = _backend(dp)
        ds = be.softmax_bwd(dp.contiguous(), p, scale=ctx.scale)
        return ds, None, None


def softmax(scores, scale: float = 1.0, causal: bool = False):
    return SoftmaxFn.apply(scores, scale, causal)


# --------------------------------------------------------------------------


class AttentionFn(torch.autograd.Function):
    """Causal multi-head attention on [B, H, S, D] tensors."""

    @staticmethod
    def forward(ctx, q, k, v, causal):
        be = _backend(q)
        out, residuals = be.attention_fwd(q, k, v, causal=causal)
        ctx.save_for_backward(q, k, v, *residuals)
        ctx.causal = causal
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, *residuals = ctx.saved_tensors
        be = _backend(dout)
        dq, dk, dv = be.attention_bwd(dout.contiguous(), q, k, v,
                                      tuple(residuals), causal=ctx.causal)
        return dq, dk, dv, None


def attention(q, k, v, causal: bool = True):
    return AttentionFn.apply(q, k, v, causal)


class AttentionQKVFn(torch.autograd.Function):
    """Attention on the packed qkv projection [B, S, 3*H*D]: the GPU flash
    kernels consume/produce the packed layout directly (no transposes)."""

    @staticmethod
    def forward(ctx, qkv, heads, causal):
        be = _backend(qkv)
        r = be.attention_qkv_fwd(qkv, heads, causal=causal) \
            if hasattr(be, "attention_qkv_fwd") else None
        if r is None:  # hip fallback for unsupported head dims
            r = ref.attention_qkv_fwd(qkv.cpu(), heads, causal) \
                if not qkv.is_cuda else _qkv_composed(be, qkv, heads, causal)
        out, residuals = r
        ctx.save_for_backward(qkv, *residuals)
        ctx.heads = heads
        ctx.causal = causal
        return out

    @staticmethod
    def backward(ctx, dout):
        qkv, *residuals = ctx.saved_tensors
        be = _backend(dout)
        dqkv = be.attention_qkv_bwd(dout.contiguous(), qkv, ctx.heads,
                                    tuple(residuals), causal=ctx.causal)
        return dqkv, None, None


def _qkv_composed(be, qkv, heads, causal):
    B, S, d3 = qkv.shape
    d = d3 // 3
    D = d // heads
    q, k, v = (t.reshape(B, S, heads, D).transpose(1, 2).contiguous()
               for t in qkv.split(d, dim=-1))
    out4, res = be.attention_fwd(q, k, v, causal=causal)
    return out4.transpose(1, 2).reshape(B, S, d), (q, k, v) + res


def attention_qkv(qkv, heads: int, causal: bool = True):
    return AttentionQKVFn.apply(qkv, heads, causal)


# --------------------------------------------------------------------------


class GeluFn(torch.autograd.Function):
    """Standalone tanh-gelu (the IR's `gelu` op; linear fuses its own)."""

    @staticmethod
    def forward(ctx, x):
        ctx.save_for_backward(x)
        return _backend(x).gelu_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        (x,) = ctx.saved_tensors
        return _backend(x).gelu_bwd(dy.contiguous(), x)


def gelu(x):
    return GeluFn.apply(x)


class EmbeddingFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, ids, table):
        be = _backend(table)
        ctx.save_for_backward(ids)
        ctx.vocab = table.shape[0]
        return be.embedding_fwd(ids, table)

    @staticmethod
    def backward(ctx, dy):
        (ids,) = ctx.saved_tensors
        be = _backend(dy)
        return None, be.embedding_bwd(dy.contiguous(), ids, ctx.vocab)


def embedding(ids, table):
    return EmbeddingFn.apply(ids, table)


# --------------------------------------------------------------------------


class CrossEntropyFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, targets, ignore_index):
        be = _backend(logits)
        loss, lse = be.cross_entropy_fwd(logits, targets, ignore_index)
        ctx.save_for_backward(logits, targets, lse)
        ctx.ignore_index = ignore_index
        return loss

    @staticmethod
    def backward(ctx, dloss):
        logits, targets, lse = ctx.saved_tensors
        be = _backend(logits)
        dl = be.cross_entropy_bwd(dloss, logits, targets, lse, ctx.ignore_index)
        return dl, None, None


def cross_entropy(logits, targets, ignore_index: int = -1):
    """Mean NLL over non-ignored targets; logits [M, V], targets [M]."""
    return CrossEntropyFn.apply(logits, targets, ignore_index)


# --------------------------------------------------------------------------


class DropoutFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, p, seed, offset):
        be = _backend(x)
        y, mask = be.dropout_fwd(x, p, seed, offset)
        ctx.save_for_backward(mask if mask is not None else torch.empty(0))
        ctx.p = p
        ctx.has_mask = mask is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        (mask,) = ctx.saved_tensors
        be = _backend(dy)
        dx = be.dropout_bwd(dy.contiguous(), mask if ctx.has_mask else None, ctx.p)
        return dx, None, None, None


def dropout(x, p: float, seed: int = 0, offset: int = 0):
    if p == 0.0:
        return x
    return DropoutFn.apply(x, p, seed, offset)


def adamw_step(param, master, grad, exp_avg, exp_avg_sq, *, lr, beta1=0.9,
               beta2=0.999, eps=1e-8, weight_decay=0.01, step=1):
    be = _backend(param)
    be.adamw_step(param, master, grad, exp_avg, exp_avg_sq, lr, beta1, beta2,
                  eps, weight_decay, step)


# --------------------------------------------------------------------------
# RMSNorm / RoPE / SwiGLU (llama family)
# --------------------------------------------------------------------------


class RMSNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, gamma, eps):
        be = _backend(x)
        x2d = x.reshape(-1, x.shape[-1])
        y, rstd = be.rmsnorm_fwd(x2d, gamma, eps)
        ctx.save_for_backward(x2d, gamma, rstd)
        ctx.in_shape = x.shape
        return y.reshape(x.shape)

    @staticmethod
    def backward(ctx, dy):
        x2d, gamma, rstd = ctx.saved_tensors
        be = _backend(dy)
        dy2d = dy.reshape(-1, dy.shape[-1]).contiguous()
        dx, dg = be.rmsnorm_bwd(dy2d, x2d, gamma, rstd)
        return dx.reshape(ctx.in_shape), dg, None


def rmsnorm(x, gamma, eps: float = 1e-6):
    return RMSNormFn.apply(x, gamma, eps)


class RoPEFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, seq_len, theta):
        be = _backend(x)
        ctx.seq_len, ctx.theta = seq_len, theta
        return be.rope_fwd(x.contiguous(), seq_len, theta)

    @staticmethod
    def backward(ctx, dy):
        be = _backend(dy)
        return be.rope_bwd(dy.contiguous(), ctx.seq_len, ctx.theta), \
            None, None


def rope(x, seq_len: int, theta: float = 10000.0):
    """Rotary position embedding over [tokens, heads, head_dim]
    (rotate-half convention; position = token index % seq_len)."""
    return RoPEFn.apply(x, seq_len, theta)


class SwiGLUFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b):
        be = _backend(a)
        ctx.save_for_backward(a, b)
        return be.swiglu_fwd(a.contiguous(), b.contiguous())

    @staticmethod
    def backward(ctx, dy):
        a, b = ctx.saved_tensors
        be = _backend(dy)
        da, db = be.swiglu_bwd(dy.contiguous(), a.contiguous(),
                               b.contiguous())
        return da, db


def swiglu(a, b):
    return SwiGLUFn.apply(a, b)

