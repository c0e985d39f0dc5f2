"""GPT-MoE tests: top-2 gating semantics, training on CPU, and
expert-parallel all-to-all equivalence across 2 gloo ranks."""

import os

import pytest
import torch

from tests.conftest import free_port
import torch.distributed as dist
import torch.multiprocessing as mp

from tepdist_amd.models.configs import MOE_CONFIGS
from tepdist_amd.models.moe import GPTMoE, MoELayer


def test_moe_layer_shapes_and_grads():
    torch.manual_seed(0)
    layer = MoELayer(32, num_experts=4, top_k=2, dtype=torch.float32)
    with torch.no_grad():
        for p in layer.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.05)
    x = torch.randn(2, 8, 32, requires_grad=True)
    y = layer(x)
    assert y.shape == x.shape
    y.sum().backward()
    assert x.grad is not None and torch.isfinite(x.grad).all()
    assert layer.w_gate.grad is not None
    assert layer.w1.grad is not None and layer.w1.grad.abs().sum() > 0
    assert float(layer.aux_loss.detach()) > 0


def test_gpt_moe_trains():
    cfg = MOE_CONFIGS["gpt-moe-test"]
    torch.manual_seed(0)
    model = GPTMoE(cfg, dtype=torch.float32)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    g = torch.Generator().manual_seed(3)
    ids = torch.randint(0, cfg.vocab_size, (4, 17), generator=g)
    losses = []
    for _ in range(10):
        opt.zero_grad()
        loss = model(ids[:, :-1], labels=ids[:, 1:])
        loss.backward()
        opt.step()
        losses.append(loss.item())
    assert losses[-1] < losses[0], losses


def _ep_worker(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(0)
    E = 4
    d = 16
    # master layer (all experts local)
    full = MoELayer(d, E, top_k=2, dtype=torch.float32)
    with torch.no_grad():
        for p in full.parameters():
            if p.dim() >= 2:
                p.normal_(0, 0.05)
    # EP layer: this rank owns E/world experts, weights sliced from master
    ep = MoELayer(d, E, top_k=2, ep_group=None, ep_size=world, ep_rank=rank,
                  dtype=torch.float32)
    el = E // world
    with torch.no_grad():
        ep.w_gate.copy_(full.w_gate)
        ep.w1.copy_(full.w1[rank * el:(rank + 1) * el])
        ep.b1.copy_(full.b1[rank * el:(rank + 1) * el])
        ep.w2.copy_(full.w2[rank * el:(rank + 1) * el])
        ep.b2.copy_(full.b2[rank * el:(rank + 1) * el])
    x = torch.randn(2, 6, d, generator=torch.Generator().manual_seed(9))
    y_ref = full(x.clone())
    y_ep = ep(x.clone())
    torch.testing.assert_close(y_ep, y_ref, rtol=1e-4, atol=1e-5)
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_expert_parallel_matches_single():
    port = free_port()
    mp.spawn(_ep_worker, args=(2, port), nprocs=2, join=True)
