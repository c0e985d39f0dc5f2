"""Property-style sweep: for a range of model shapes and shard counts,
CostSpmdStrategy's plan must transform into a per-rank graph whose
2-process (gloo) execution reproduces the single-device loss and leaves
finite gradients. This is the end-to-end consistency contract between
rules.py (strategy enumeration), spmd.py (search), transform.py
(graph rewriting) and the interpreter's collective lowering."""

import os

import pytest
import torch

from tests.conftest import free_port
import torch.distributed as dist
import torch.multiprocessing as mp

from tepdist_amd.ir.capture import gpt2_ir, llama_ir
from tepdist_amd.ir.graph import Graph
from tepdist_amd.ir.interpreter import GraphInterpreter
from tepdist_amd.models.configs import GPT2Config
from tepdist_amd.models.llama import LlamaConfig
from tepdist_amd.planner.spmd import CostSpmdStrategy
from tepdist_amd.planner.transform import SpmdTransform
from tepdist_amd.runtime.initializers import InitSpec, init_shard

CASES = [
    ("gpt2", dict(n_layer=2, n_embd=64, n_head=4, vocab_size=128,
                  n_ctx=64), 4, 16),
    ("gpt2", dict(n_layer=1, n_embd=128, n_head=4, vocab_size=256,
                  n_ctx=32), 2, 32),
    ("llama", dict(n_layer=2, n_embd=64, n_head=4, vocab_size=128,
                   n_ctx=64, ffn_mult=2), 4, 16),
    # world-4 case (divisibility/spec-consistency at higher shard counts)
    ("gpt2", dict(n_layer=1, n_embd=128, n_head=8, vocab_size=256,
                  n_ctx=32), 8, 16),
    # MoE graph: the cost search must produce an executable plan over the
    # moe_dispatch/combine + batched-expert-matmul op family
    ("moe", dict(), 4, 16),
]


def _build(kind, cfg_kw, batch, seq):
    if kind == "gpt2":
        cfg = GPT2Config(name="t", **cfg_kw)
        return gpt2_ir(cfg, batch=batch, seq=seq), cfg.padded_vocab
    if kind == "moe":
        from tepdist_amd.ir.capture import moe_ir
        from tepdist_amd.models.configs import MOE_CONFIGS
        cfg = MOE_CONFIGS["gpt-moe-test"]
        return moe_ir(cfg, batch=batch, seq=seq,
                      capacity_factor=4.0), cfg.vocab_size
    cfg = LlamaConfig(name="t", **cfg_kw)
    return llama_ir(cfg, batch=batch, seq=seq), cfg.vocab_size


def _variables(g: Graph, device="cpu"):
    out = {}
    for name, nid in g.params.items():
        shape = g.nodes[nid].shape
        spec = InitSpec("ones") if name.endswith("_g") else (
            InitSpec("zeros") if name.endswith("_b")
            else InitSpec("random_normal", std=0.05))
        out[name] = init_shard(name, shape, spec,
                               dtype=torch.float32).requires_grad_()
    return out


def _feeds(g, vocab, seed=3):
    gen = torch.Generator().manual_seed(seed)
    bs = g.nodes[g.inputs[0]].shape[0]
    return {"input_ids": torch.randint(0, vocab, (bs,), generator=gen),
            "labels": torch.randint(0, vocab, (bs,), generator=gen)}


def _shard(t, dim, n, rank):
    if dim < 0 or n == 1:
        return t
    sz = t.shape[dim] // n
    return t.narrow(dim, rank * sz, sz).contiguous()


def _worker(rank, world, port, case_idx):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    kind, cfg_kw, batch, seq = CASES[case_idx]
    g, vocab = _build(kind, cfg_kw, batch, seq)
    plan = CostSpmdStrategy(g, world).run()
    res = SpmdTransform(g, plan.node_specs, world).run()

    full = _variables(g)
    local = {k: _shard(v.detach().clone(), *res.param_specs[k],
                       rank).requires_grad_() for k, v in full.items()}
    feeds = _feeds(g, vocab)
    loss = list(GraphInterpreter(res.graph).run(feeds, local).values())[0]
    loss.backward()

    ref = list(GraphInterpreter(g).run(feeds, full).values())[0]
    assert torch.allclose(loss.detach(), ref.detach(), rtol=2e-4,
                          atol=1e-5), (kind, loss.item(), ref.item())
    for k, v in local.items():
        if v.grad is not None:
            assert torch.isfinite(v.grad).all(), k
    dist.destroy_process_group()


@pytest.mark.parametrize("case_idx", range(len(CASES)))
@pytest.mark.timeout(600)
def test_plan_transform_execute_matches_single(case_idx):
    world = 4 if case_idx == len(CASES) - 1 else 2
    port = free_port()
    mp.spawn(_worker, args=(world, port, case_idx), nprocs=world, join=True)
