"""Context-parallel GPT-2 (parallel/cp.py): sequence-sharded training
step over ring attention must match the full-sequence single-process
model exactly (fp32) — loss AND parameter gradients (after the CP-group
SUM reduction, since parameters are replicated)."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from tests.conftest import free_port


def _worker(rank, world, port, zigzag):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from tepdist_amd.models import GPT2, GPT2Config
        from tepdist_amd.parallel.cp import ContextParallelGPT2
        from tepdist_amd.parallel.dp import GradReducer
        cfg = GPT2Config("cp-test", n_layer=2, n_embd=32, n_head=2,
                         n_ctx=64, vocab_size=128)
        torch.manual_seed(7)
        ref = GPT2(cfg, dtype=torch.float32)
        cp = ContextParallelGPT2(cfg, zigzag=zigzag, dtype=torch.float32)
        cp.load_state_dict(ref.state_dict())

        g = torch.Generator().manual_seed(11)
        B, S = 2, 16 * world
        ids = torch.randint(0, 128, (B, S + 1), generator=g)
        inp, lab = ids[:, :-1], ids[:, 1:].contiguous()

        loss_ref = ref(inp, lab)
        loss_ref.backward()

        red = GradReducer(cp.parameters(), average=False,
                          bucket_bytes=1 << 16)
        li, ll, pos = cp.shard_inputs(inp, lab)
        red.reset()
        red.arm()
        loss = cp(li, ll, pos=pos)
        loss.backward()
        red.finalize()

        torch.testing.assert_close(loss, loss_ref, rtol=1e-5, atol=1e-6)
        for (n1, p1), (n2, p2) in zip(ref.named_parameters(),
                                      cp.named_parameters()):
            assert n1 == n2
            torch.testing.assert_close(p2.grad, p1.grad, rtol=1e-4,
                                       atol=1e-5, msg=n1)
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("world,zigzag", [(2, False), (2, True),
                                          (4, True)])
@pytest.mark.timeout(600)
def test_cp_gpt2_matches_full(world, zigzag):
    port = free_port()
    mp.spawn(_worker, args=(world, port, zigzag), nprocs=world, join=True)


def _learn_worker(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from tepdist_amd.models.gpt2 import GPT2Config
        from tepdist_amd.parallel.cp import ContextParallelGPT2
        from tepdist_amd.parallel.dp import GradReducer
        from tepdist_amd.train.optim import AdamW
        cfg = GPT2Config("cp-learn", n_layer=2, n_embd=32, n_head=2,
                         n_ctx=64, vocab_size=128)
        torch.manual_seed(3)
        m = ContextParallelGPT2(cfg, zigzag=True, dtype=torch.float32)
        red = GradReducer(m.parameters(), average=False,
                          bucket_bytes=1 << 16)
        opt = AdamW(m.parameters(), lr=3e-3)
        g = torch.Generator().manual_seed(9)
        ids = torch.randint(0, 128, (2, 33), generator=g)
        li, ll, pos = m.shard_inputs(ids[:, :-1], ids[:, 1:].contiguous())
        first = last = None
        for step in range(25):
            opt.zero_grad()
            red.reset()
            red.arm()
            loss = m(li, ll, pos=pos)
            loss.backward()
            red.finalize()
            opt.step()
            if step == 0:
                first = float(loss.detach())
            last = float(loss.detach())
        assert last < first * 0.7, (rank, first, last)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_cp_learns():
    """CP grad-sync end-to-end sanity: sequence-sharded training
    memorizes a fixed batch (sign/scale of the SUM reduction over
    replicated params is what this pins)."""
    port = free_port()
    mp.spawn(_learn_worker, args=(2, port), nprocs=2, join=True)
