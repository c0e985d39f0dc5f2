"""Pure-PyTorch reference implementations of every planner-sharded op.

These define the exact numerical semantics of the hand-written CDNA4 HIP
kernels in tepdist_amd/ops/csrc/. They are the CPU execution path and the
fp32 baseline that tests/test_*_numerics.py compare the HIP kernels against
(numerics tests for a HIP kernel compare against a plain
PyTorch fp32 reference of the same op).

Conventions:
- Compute dtype bf16, statistics/accumulation fp32 (kernels accumulate in
  fp32 MFMA accumulators / fp32 VGPRs).
- `linear` takes weight in PyTorch layout [out_features, in_features] so the
  forward GEMM is A[M,K] @ B^T with B stored [N,K] (the MFMA kernel's
  preferred k-contiguous layout).
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import torch

SQRT_2_OVER_PI = math.sqrt(2.0 / math.pi)


# --------------------------------------------------------------------------
# GEMM / linear
# --------------------------------------------------------------------------

def linear_fwd(x: torch.Tensor, w: torch.Tensor, bias: Optional[torch.Tensor],
               act: str = "none") -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
    """y = act(x @ w^T + bias). Returns (y, pre_act) where pre_act is saved
    only when act != 'none' (needed for backward)."""
    y32 = x.float() @ w.float().t()
    if bias is not None:
        y32 = y32 + bias.float()
    if act == "none":
        return y32.to(x.dtype), None
    elif act == "gelu":
        pre = y32.to(x.dtype)
        return gelu_fwd(pre), pre
    else:
        raise ValueError(f"unknown activation {act}")


def linear_bwd(dy: torch.Tensor, x: torch.Tensor, w: torch.Tensor,
               has_bias: bool, act: str, pre_act: Optional[torch.Tensor]):
    """Returns (dx, dw, dbias)."""
    dy32 = dy.float()
    if act == "gelu":
        dy32 = (gelu_bwd(dy, pre_act)).float()
    dx = (dy32 @ w.float()).to(x.dtype)
    dw = (dy32.t() @ x.float()).to(w.dtype)
    db = dy32.sum(dim=0).to(w.dtype) if has_bias else None
    return dx, dw, db


def matmul(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Plain C = A @ B with fp32 accumulation, output in input dtype.
    Supports batched inputs with broadcasting like torch.matmul."""
    return (a.float() @ b.float()).to(a.dtype)


# --------------------------------------------------------------------------
# GELU (tanh approximation, as GPT-2 uses)
# --------------------------------------------------------------------------

def gelu_fwd(x: torch.Tensor) -> torch.Tensor:
    x32 = x.float()
    y = 0.5 * x32 * (1.0 + torch.tanh(SQRT_2_OVER_PI * (x32 + 0.044715 * x32 ** 3)))
    return y.to(x.dtype)


def gelu_bwd(dy: torch.Tensor, x: torch.Tensor) -> torch.Tensor:
    x32 = x.float()
    t = torch.tanh(SQRT_2_OVER_PI * (x32 + 0.044715 * x32 ** 3))
    dt = (1.0 - t * t) * SQRT_2_OVER_PI * (1.0 + 3 * 0.044715 * x32 ** 2)
    dgelu = 0.5 * (1.0 + t) + 0.5 * x32 * dt
    return (dy.float() * dgelu).to(dy.dtype)


# --------------------------------------------------------------------------
# LayerNorm
# --------------------------------------------------------------------------

def layernorm_fwd(x: torch.Tensor, gamma: torch.Tensor, beta: torch.Tensor,
                  eps: float = 1e-5):
    """Row-wise LN over the last dim. Returns (y, mean[f32], rstd[f32])."""
    x32 = x.float()
    mean = x32.mean(dim=-1)
    var = x32.var(dim=-1, unbiased=False)
    rstd = torch.rsqrt(var + eps)
    xhat = (x32 - mean.unsqueeze(-1)) * rstd.unsqueeze(-1)
    y = xhat * gamma.float() + beta.float()
    return y.to(x.dtype), mean, rstd


def layernorm_bwd(dy: torch.Tensor, x: torch.Tensor, gamma: torch.Tensor,
                  mean: torch.Tensor, rstd: torch.Tensor):
    """Returns (dx, dgamma, dbeta)."""
    x32 = x.float()
    dy32 = dy.float()
    xhat = (x32 - mean.unsqueeze(-1)) * rstd.unsqueeze(-1)
    dgamma = (dy32 * xhat).sum(dim=tuple(range(dy.dim() - 1)))
    dbeta = dy32.sum(dim=tuple(range(dy.dim() - 1)))
    D = x.shape[-1]
    g = dy32 * gamma.float()
    c1 = g.mean(dim=-1, keepdim=True)
    c2 = (g * xhat).mean(dim=-1, keepdim=True)
    dx = (g - c1 - xhat * c2) * rstd.unsqueeze(-1)
    return dx.to(x.dtype), dgamma.to(gamma.dtype), dbeta.to(gamma.dtype)


# --------------------------------------------------------------------------
# Softmax (fused scale + causal mask + softmax over last dim)
# --------------------------------------------------------------------------

def softmax_fwd(scores: torch.Tensor, scale: float = 1.0,
                causal: bool = False) -> torch.Tensor:
    """p = softmax(scores * scale [+ causal mask]) over the last dim.
    scores: [..., S_q, S_k]."""
    s32 = scores.float() * scale
    if causal:
        sq, sk = scores.shape[-2], scores.shape[-1]
        mask = torch.ones(sq, sk, dtype=torch.bool, device=scores.device).tril(sk - sq)
        s32 = s32.masked_fill(~mask, float("-inf"))
    p = torch.softmax(s32, dim=-1)
    return p.to(scores.dtype)


def softmax_bwd(dp: torch.Tensor, p: torch.Tensor, scale: float = 1.0) -> torch.Tensor:
    """ds = scale * p * (dp - sum(dp*p, -1)). The causal mask needs no special
    handling: masked p entries are exactly 0."""
    p32 = p.float()
    dp32 = dp.float()
    dot = (dp32 * p32).sum(dim=-1, keepdim=True)
    return (scale * p32 * (dp32 - dot)).to(p.dtype)


# --------------------------------------------------------------------------
# Attention (composed reference; the GPU path uses batched GEMM kernels +
# the fused softmax kernel, later a fused flash-style kernel)
# --------------------------------------------------------------------------

def attention_fwd(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                  causal: bool = True):
    """q,k,v: [B, H, S, D]. Returns (out, residuals) with the residuals
    saved for backward (here: the attention probabilities)."""
    scale = 1.0 / math.sqrt(q.shape[-1])
    scores = matmul(q, k.transpose(-1, -2))
    p = softmax_fwd(scores, scale=scale, causal=causal)
    out = matmul(p, v)
    return out, (p,)


def attention_bwd(dout: torch.Tensor, q, k, v, residuals,
                  causal: bool = True):
    (p,) = residuals
    scale = 1.0 / math.sqrt(q.shape[-1])
    dv = matmul(p.transpose(-1, -2), dout)
    dp = matmul(dout, v.transpose(-1, -2))
    ds = softmax_bwd(dp, p, scale=scale)
    dq = matmul(ds, k)
    dk = matmul(ds.transpose(-1, -2), q)
    return dq, dk, dv


def attention_qkv_fwd(qkv: torch.Tensor, heads: int, causal: bool = True):
    """Packed-projection attention (reference): split + transpose, run the
    composed path, return residuals carrying what backward needs."""
    B, S, d3 = qkv.shape
    d = d3 // 3
    D = d // heads
    q, k, v = (t.reshape(B, S, heads, D).transpose(1, 2).contiguous()
               for t in qkv.split(d, dim=-1))
    out4, (p,) = attention_fwd(q, k, v, causal=causal)
    out = out4.transpose(1, 2).reshape(B, S, d)
    return out, (q, k, v, p)


def attention_qkv_bwd(dout: torch.Tensor, qkv: torch.Tensor, heads: int,
                      residuals, causal: bool = True):
    B, S, d3 = qkv.shape
    d = d3 // 3
    D = d // heads
    q, k, v, p = residuals
    dout4 = dout.reshape(B, S, heads, D).transpose(1, 2).contiguous()
    dq, dk, dv = attention_bwd(dout4, q, k, v, (p,), causal=causal)
    def back(t):
        return t.transpose(1, 2).reshape(B, S, d)
    return torch.cat([back(dq), back(dk), back(dv)], dim=-1)


# --------------------------------------------------------------------------
# Embedding
# --------------------------------------------------------------------------

def embedding_fwd(ids: torch.Tensor, table: torch.Tensor) -> torch.Tensor:
    return table[ids]


def embedding_bwd(dy: torch.Tensor, ids: torch.Tensor, vocab: int) -> torch.Tensor:
    """Scatter-add gradients into a [vocab, D] table gradient (fp32 accum)."""
    D = dy.shape[-1]
    grad = torch.zeros(vocab, D, dtype=torch.float32, device=dy.device)
    grad.index_add_(0, ids.reshape(-1), dy.reshape(-1, D).float())
    return grad.to(dy.dtype)


# --------------------------------------------------------------------------
# Cross entropy (fused log-softmax + NLL; bwd writes d_logits directly)
# --------------------------------------------------------------------------

def cross_entropy_fwd(logits: torch.Tensor, targets: torch.Tensor,
                      ignore_index: int = -1):
    """logits [M, V], targets [M]. Returns (mean_loss[f32 scalar],
    logsumexp[f32, M]) with lse saved for backward."""
    l32 = logits.float()
    lse = torch.logsumexp(l32, dim=-1)
    valid = targets != ignore_index
    has_tgt = targets >= 0  # negative non-ignore = "no target here"
    tgt = targets.clamp_min(0)
    nll = lse - l32.gather(-1, tgt.unsqueeze(-1)).squeeze(-1)
    nll = torch.where(valid & has_tgt, nll, torch.zeros_like(nll))
    n = valid.sum().clamp_min(1)
    return nll.sum() / n, lse


def cross_entropy_bwd(dloss: torch.Tensor, logits: torch.Tensor,
                      targets: torch.Tensor, lse: torch.Tensor,
                      ignore_index: int = -1) -> torch.Tensor:
    l32 = logits.float()
    p = torch.exp(l32 - lse.unsqueeze(-1))
    valid = (targets != ignore_index)
    has_tgt = targets >= 0  # negative non-ignore: softmax grad, no onehot
    tgt = targets.clamp_min(0)
    p.scatter_add_(-1, tgt.unsqueeze(-1),
                   -has_tgt.to(torch.float32).unsqueeze(-1))
    n = valid.sum().clamp_min(1).float()
    p = p * (dloss.float() / n)
    p = torch.where(valid.unsqueeze(-1), p, torch.zeros_like(p))
    return p.to(logits.dtype)


# --------------------------------------------------------------------------
# Dropout (counter-based Philox so the mask is reproducible from (seed,
# offset) without storing it; the HIP kernel uses the same construction)
# --------------------------------------------------------------------------

def _philox_mask(shape, p: float, seed: int, offset: int, device) -> torch.Tensor:
    g = torch.Generator(device="cpu")
    g.manual_seed(seed * 1000003 + offset)
    return (torch.rand(shape, generator=g, device="cpu") >= p).to(device)


def dropout_fwd(x: torch.Tensor, p: float, seed: int, offset: int):
    if p == 0.0:
        return x, None
    mask = _philox_mask(x.shape, p, seed, offset, x.device)
    scale = 1.0 / (1.0 - p)
    return (x.float() * mask.float() * scale).to(x.dtype), mask


def dropout_bwd(dy: torch.Tensor, mask: Optional[torch.Tensor], p: float):
    if p == 0.0 or mask is None:
        return dy
    scale = 1.0 / (1.0 - p)
    return (dy.float() * mask.float() * scale).to(dy.dtype)


# --------------------------------------------------------------------------
# AdamW fused step (bf16 params + fp32 master/moments)
# --------------------------------------------------------------------------

def adamw_step(param_bf16: torch.Tensor, master: torch.Tensor,
               grad: torch.Tensor, exp_avg: torch.Tensor,
               exp_avg_sq: torch.Tensor, lr: float, beta1: float,
               beta2: float, eps: float, weight_decay: float, step: int):
    """In-place AdamW on the fp32 master copy; bf16 param refreshed from it.
    grad may be bf16 or fp32."""
    g = grad.float()
    exp_avg.mul_(beta1).add_(g, alpha=1 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(g, g, value=1 - beta2)
    bc1 = 1 - beta1 ** step
    bc2 = 1 - beta2 ** step
    denom = (exp_avg_sq / bc2).sqrt().add_(eps)
    master.mul_(1 - lr * weight_decay)
    master.addcdiv_(exp_avg / bc1, denom, value=-lr)
    param_bf16.copy_(master.to(param_bf16.dtype))


# --------------------------------------------------------------------------
# RMSNorm / RoPE / SwiGLU (llama-family ops; torch fp32-semantics references)
# --------------------------------------------------------------------------

def rmsnorm_fwd(x: torch.Tensor, gamma: torch.Tensor, eps: float = 1e-6):
    xf = x.float()
    rstd = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    y = (xf * rstd) * gamma.float()
    return y.to(x.dtype), rstd.squeeze(-1)


def rmsnorm_bwd(dy: torch.Tensor, x: torch.Tensor, gamma: torch.Tensor,
                rstd: torch.Tensor):
    xf, dyf, gf = x.float(), dy.float(), gamma.float()
    r = rstd.unsqueeze(-1)
    xh = xf * r
    dg = (dyf * xh).sum(0)
    t = dyf * gf
    dx = r * (t - xh * (t * xh).mean(-1, keepdim=True))
    return dx.to(x.dtype), dg.to(gamma.dtype)


def rope_fwd(x: torch.Tensor, seq_len: int, theta: float = 10000.0):
    """x [tokens, heads, D] (tokens = batch*seq flattened, position =
    token % seq_len); rotate-half convention."""
    T, H, D = x.shape
    pos = (torch.arange(T, device=x.device) % seq_len).float()
    i = torch.arange(D // 2, device=x.device).float()
    freq = theta ** (-2.0 * i / D)
    ang = pos[:, None] * freq[None, :]
    cos, sin = ang.cos()[:, None, :], ang.sin()[:, None, :]
    xf = x.float()
    x1, x2 = xf[..., :D // 2], xf[..., D // 2:]
    y = torch.cat([x1 * cos - x2 * sin, x2 * cos + x1 * sin], dim=-1)
    return y.to(x.dtype)


def rope_bwd(dy: torch.Tensor, seq_len: int, theta: float = 10000.0):
    """The inverse rotation (transpose of an orthogonal map)."""
    T, H, D = dy.shape
    pos = (torch.arange(T, device=dy.device) % seq_len).float()
    i = torch.arange(D // 2, device=dy.device).float()
    freq = theta ** (-2.0 * i / D)
    ang = pos[:, None] * freq[None, :]
    cos, sin = ang.cos()[:, None, :], ang.sin()[:, None, :]
    df = dy.float()
    d1, d2 = df[..., :D // 2], df[..., D // 2:]
    dx = torch.cat([d1 * cos + d2 * sin, d2 * cos - d1 * sin], dim=-1)
    return dx.to(dy.dtype)


def swiglu_fwd(a: torch.Tensor, b: torch.Tensor):
    af = a.float()
    return (af * torch.sigmoid(af) * b.float()).to(a.dtype)


def swiglu_bwd(dy: torch.Tensor, a: torch.Tensor, b: torch.Tensor):
    af, bf, dyf = a.float(), b.float(), dy.float()
    sig = torch.sigmoid(af)
    silu = af * sig
    da = dyf * bf * (sig + silu * (1.0 - sig))
    db = dyf * silu
    return da.to(a.dtype), db.to(b.dtype)
