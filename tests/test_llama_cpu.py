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


# -- tensor-parallel llama (gloo world 2) ---------------------------------


def _llama_tp_worker(rank, world, port):
    import os

    import torch.distributed as dist

    from tepdist_amd.models.gpt2 import shard_qkv_weight
    from tepdist_amd.parallel.tp import ParallelEnv

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(0)
    cfg = LLAMA_CONFIGS["llama-test"]
    master = Llama(cfg)     # replicated reference weights (seeded)

    env = ParallelEnv.create(tp_size=world)
    torch.manual_seed(0)    # same init draw order as master
    model = Llama(cfg, env=env)

    # copy master weights into this rank's shards
    with torch.no_grad():
        vl = cfg.vocab_size // world
        model.wte_mod.weight.copy_(
            master.wte.narrow(0, rank * vl, vl))
        model.head.weight.copy_(
            master.lm_head.narrow(0, rank * vl, vl))
        model.ln_f_g.copy_(master.ln_f_g)
        hl = cfg.ffn_mult * cfg.n_embd // world
        dl = cfg.n_embd // world
        for blk, mb in zip(model.blocks, master.blocks):
            blk.ln1_g.copy_(mb.ln1_g)
            blk.ln2_g.copy_(mb.ln2_g)
            wq, _ = shard_qkv_weight(mb.w_qkv, torch.zeros(3 * cfg.n_embd),
                                     cfg.n_head, rank, world)
            blk.qkv.weight.copy_(wq)
            blk.o.weight.copy_(mb.w_o.narrow(1, rank * dl, dl))
            blk.gate.weight.copy_(mb.w_gate.narrow(0, rank * hl, hl))
            blk.up.weight.copy_(mb.w_up.narrow(0, rank * hl, hl))
            blk.down.weight.copy_(mb.w_down.narrow(1, rank * hl, hl))

    gen = torch.Generator().manual_seed(9)
    ids = torch.randint(0, cfg.vocab_size, (2, 33), generator=gen)
    loss = model(ids[:, :-1], labels=ids[:, 1:])
    ref = master(ids[:, :-1], labels=ids[:, 1:])
    assert torch.allclose(loss, ref, rtol=1e-4, atol=1e-5), \
        (loss.item(), ref.item())
    loss.backward()
    for n, p in model.named_parameters():
        assert p.grad is None or torch.isfinite(p.grad).all(), n
    dist.destroy_process_group()


def test_llama_tensor_parallel_matches_single():
    import torch.multiprocessing as mp

    from tests.conftest import free_port
    port = free_port()
    mp.spawn(_llama_tp_worker, args=(2, port), nprocs=2, join=True)
