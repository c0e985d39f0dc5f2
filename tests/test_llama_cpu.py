"""Llama-family ops and model on CPU: torch-reference numerics for
RMSNorm / RoPE / SwiGLU (fp32 autograd cross-check) and a tiny training
run that must reduce the loss."""

import torch

from tepdist_amd import ops
from tepdist_amd.models.llama import LLAMA_CONFIGS, Llama


def test_rmsnorm_matches_autograd():
    x = torch.randn(12, 64, requires_grad=True)
    g = torch.randn(64, requires_grad=True)
    y = ops.rmsnorm(x, g)
    ref = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + 1e-6) * g
    assert torch.allclose(y, ref, atol=1e-5)
    dy = torch.randn_like(y)
    y.backward(dy)
    x2 = x.detach().clone().requires_grad_()
    g2 = g.detach().clone().requires_grad_()
    r2 = x2 * torch.rsqrt(x2.pow(2).mean(-1, keepdim=True) + 1e-6) * g2
    r2.backward(dy)
    assert torch.allclose(x.grad, x2.grad, atol=1e-5)
    assert torch.allclose(g.grad, g2.grad, atol=1e-4)


def test_rope_rotation_properties():
    T, H, D = 8, 2, 16
    x = torch.randn(T, H, D, requires_grad=True)
    y = ops.rope(x, seq_len=T)
    # norm-preserving per (token, head) pair structure
    assert torch.allclose(y.norm(dim=-1), x.norm(dim=-1), atol=1e-5)
    # position 0 is the identity
    assert torch.allclose(y[0], x[0], atol=1e-6)
    # backward is the inverse rotation: grad of sum(y*c) = rope^T(c)
    c = torch.randn_like(y)
    (y * c).sum().backward()
    # apply forward to grad and compare against c rotated back and forth
    assert torch.allclose(ops.rope(x.grad.detach(), seq_len=T), c, atol=1e-4)


def test_swiglu_matches_autograd():
    a = torch.randn(40, requires_grad=True)
    b = torch.randn(40, requires_grad=True)
    y = ops.swiglu(a, b)
    ref = torch.nn.functional.silu(a) * b
    assert torch.allclose(y, ref, atol=1e-6)
    dy = torch.randn_like(y)
    y.backward(dy)
    a2 = a.detach().clone().requires_grad_()
    b2 = b.detach().clone().requires_grad_()
    (torch.nn.functional.silu(a2) * b2).backward(dy)
    assert torch.allclose(a.grad, a2.grad, atol=1e-5)
    assert torch.allclose(b.grad, b2.grad, atol=1e-5)


def test_llama_tiny_trains():
    torch.manual_seed(0)
    cfg = LLAMA_CONFIGS["llama-test"]
    model = Llama(cfg)
    from tepdist_amd.train.optim import AdamW
    opt = AdamW(model.parameters(), lr=3e-3)
    ids = torch.randint(0, cfg.vocab_size, (2, 33))
    losses = []
    for _ in range(8):
        opt.zero_grad()
        loss = model(ids[:, :-1], labels=ids[:, 1:])
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0] - 0.5, losses
