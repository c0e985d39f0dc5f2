"""Multi-process CPU tests (gloo, world_size=2) of the distributed execution
layer: DP gradient reduction, TP layer equivalence vs a single-process
master model, vocab-parallel cross entropy, all-to-all. These exercise the
same code paths that run over RCCL/xGMI on the GPU node (SURVEY.md §4.5:
multi-worker is tested on one machine with several processes)."""

import os

import pytest
import torch

from tests.conftest import free_port
import torch.distributed as dist
import torch.multiprocessing as mp

from tepdist_amd.models import GPT2, GPT2_CONFIGS
from tepdist_amd.models.gpt2 import shard_qkv_weight

WORLD = 2


def _run(fn, world=WORLD):
    port = free_port()
    mp.spawn(fn, args=(world, port), nprocs=world, join=True)


def _init(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(0)


# --------------------------------------------------------------------------


def _dp_worker(rank, world, port):
    _init(rank, world, port)
    from tepdist_amd.parallel import GradReducer
    from tepdist_amd.train import AdamW, Trainer

    cfg = GPT2_CONFIGS["gpt2-test"]
    model = GPT2(cfg, dtype=torch.float32)
    opt = AdamW(model.parameters(), lr=1e-3)
    reducer = GradReducer(model.parameters(), bucket_bytes=1 << 20)
    trainer = Trainer(model, opt, grad_accum_steps=1, reducer=reducer)

    g = torch.Generator().manual_seed(99)
    ids = torch.randint(0, cfg.vocab_size, (2 * world, 17), generator=g)
    my = ids[rank * 2:(rank + 1) * 2]
    loss = trainer.train_step(lambda i: (my[:, :-1], my[:, 1:]))

    # single-process full-batch reference
    ref = GPT2(cfg, dtype=torch.float32)
    ref_loss = ref(ids[:, :-1], labels=ids[:, 1:])
    ref_loss.backward()
    # averaged DP grads == full-batch grads (equal shard sizes)
    for (n, p), (_, rp) in zip(model.named_parameters(),
                               ref.named_parameters()):
        if p.grad is None:
            continue
        # grads were averaged across ranks then consumed by opt.step; use
        # the reducer-written grad (still in p.grad)
        torch.testing.assert_close(p.grad, rp.grad, rtol=1e-4, atol=1e-5), n
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_dp_grad_reduce_matches_full_batch():
    _run(_dp_worker)


# --------------------------------------------------------------------------


def _tp_from_master(master: GPT2, cfg, env):
    tp = GPT2(cfg, dtype=torch.float32, env=env)
    r, ts = env.tp_rank, env.tp_size
    d = cfg.n_embd
    H = cfg.n_head
    hd = d // H
    hl = H // ts
    with torch.no_grad():
        tp.wte_mod.weight.copy_(
            master.wte[r * tp.wte_mod.vocab_local:(r + 1) * tp.wte_mod.vocab_local])
        tp.wpe.copy_(master.wpe)
        tp.lnf_g.copy_(master.lnf_g)
        tp.lnf_b.copy_(master.lnf_b)
        for tb, mb in zip(tp.blocks, master.blocks):
            for nm in ("ln1_g", "ln1_b", "ln2_g", "ln2_b"):
                getattr(tb, nm).copy_(getattr(mb, nm))
            wq, bq = shard_qkv_weight(mb.w_qkv, mb.b_qkv, H, r, ts)
            tb.qkv.weight.copy_(wq)
            tb.qkv.bias.copy_(bq)
            wp = mb.w_proj.reshape(d, H, hd)[:, r * hl:(r + 1) * hl, :]
            tb.proj.weight.copy_(wp.reshape(d, hl * hd))
            tb.proj.bias.copy_(mb.b_proj)
            c = 4 * d // ts
            tb.fc.weight.copy_(mb.w_fc[r * c:(r + 1) * c])
            tb.fc.bias.copy_(mb.b_fc[r * c:(r + 1) * c])
            tb.out.weight.copy_(mb.w_out[:, r * c:(r + 1) * c])
            tb.out.bias.copy_(mb.b_out)
    return tp


def _tp_worker(rank, world, port):
    _init(rank, world, port)
    from tepdist_amd.parallel.tp import ParallelEnv

    cfg = GPT2_CONFIGS["gpt2-test"]
    env = ParallelEnv.create(tp_size=world)
    torch.manual_seed(7)
    master = GPT2(cfg, dtype=torch.float32)
    tp = _tp_from_master(master, cfg, env)

    g = torch.Generator().manual_seed(123)
    ids = torch.randint(0, cfg.vocab_size, (3, 19), generator=g)
    loss_tp = tp(ids[:, :-1], labels=ids[:, 1:])
    loss_tp.backward()
    loss_ref = master(ids[:, :-1], labels=ids[:, 1:])
    loss_ref.backward()

    torch.testing.assert_close(loss_tp, loss_ref, rtol=1e-4, atol=1e-5)
    # spot-check sharded grads against master grads
    mb = master.blocks[0]
    tb = tp.blocks[0]
    c = 4 * cfg.n_embd // world
    torch.testing.assert_close(tb.fc.weight.grad,
                               mb.w_fc.grad[rank * c:(rank + 1) * c],
                               rtol=1e-3, atol=1e-5)
    torch.testing.assert_close(tb.ln1_g.grad, mb.ln1_g.grad, rtol=1e-3,
                               atol=1e-5)
    vs = tp.wte_mod.vocab_local
    torch.testing.assert_close(tp.wte_mod.weight.grad,
                               master.wte.grad[rank * vs:(rank + 1) * vs],
                               rtol=1e-3, atol=1e-5)
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_tp2_matches_single_process():
    _run(_tp_worker)


# --------------------------------------------------------------------------


def _a2a_worker(rank, world, port):
    _init(rank, world, port)
    from tepdist_amd.parallel.mappings import all_to_all
    x = torch.arange(8, dtype=torch.float32).reshape(world, 4) + 100 * rank
    x.requires_grad_()
    y = all_to_all(x, None)
    # row i of rank r's output = row r of rank i's input
    expect = torch.stack([torch.arange(4) + 100 * i + 4 * rank * 0 for i in range(world)]).float()
    expect = torch.stack(
        [torch.arange(4, dtype=torch.float32) + 4 * rank + 100 * i
         for i in range(world)])
    torch.testing.assert_close(y.detach(), expect)
    y.sum().backward()
    torch.testing.assert_close(x.grad, torch.ones_like(x))
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_all_to_all_roundtrip():
    _run(_a2a_worker)


def _zero_worker(rank, world, port):
    _init(rank, world, port)
    from tepdist_amd.parallel import GradReducer
    from tepdist_amd.train import AdamW, Trainer
    from tepdist_amd.train.optim import ZeroAdamW

    cfg = GPT2_CONFIGS["gpt2-test"]
    model = GPT2(cfg, dtype=torch.float32)
    opt = ZeroAdamW(model.parameters(), lr=1e-3)
    reducer = GradReducer(model.parameters(), bucket_bytes=1 << 20)
    trainer = Trainer(model, opt, reducer=reducer)

    ref = GPT2(cfg, dtype=torch.float32)
    ref_opt = AdamW(ref.parameters(), lr=1e-3)

    g = torch.Generator().manual_seed(31)
    ids = torch.randint(0, cfg.vocab_size, (2 * world, 17), generator=g)
    my = ids[rank * 2:(rank + 1) * 2]
    for _ in range(3):
        trainer.train_step(lambda i: (my[:, :-1], my[:, 1:]))
        # reference: full batch, plain AdamW
        ref_opt.zero_grad()
        loss = ref(ids[:, :-1], labels=ids[:, 1:])
        loss.backward()
        ref_opt.step()
    # sharded-optimizer training must match full-state training
    for (n, p), (_, rp) in zip(model.named_parameters(),
                               ref.named_parameters()):
        torch.testing.assert_close(p.data, rp.data, rtol=1e-4, atol=1e-5)
    # optimizer state really is sharded
    assert opt.state_bytes() < sum(p.numel() * 12 for p in
                                   model.parameters())
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_zero1_optimizer_matches_full():
    _run(_zero_worker)


def _fp16_comm_worker(rank, world, port):
    import os as _os
    _os.environ["MASTER_ADDR"] = "127.0.0.1"
    _os.environ["MASTER_PORT"] = str(port)
    _os.environ["FP16_COMM"] = "true"
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from tepdist_amd.parallel.dp import GradReducer
        torch.manual_seed(rank)
        m = torch.nn.Linear(8, 8)  # fp32 params -> bf16 comm under the flag
        red = GradReducer(m.parameters(), bucket_bytes=1 << 10)
        assert red.comm_dtype == torch.bfloat16
        red.reset()
        red.arm()
        m(torch.randn(4, 8)).sum().backward()
        red.finalize()
        # grads are averaged across ranks (through the bf16 wire)
        ref = m.weight.grad.clone()
        dist.all_reduce(ref)
        torch.testing.assert_close(m.weight.grad, ref / world, rtol=1e-2,
                                   atol=1e-2)
    finally:
        del _os.environ["FP16_COMM"]
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_fp16_comm_flag():
    port = free_port()
    mp.spawn(_fp16_comm_worker, args=(2, port), nprocs=2, join=True)
