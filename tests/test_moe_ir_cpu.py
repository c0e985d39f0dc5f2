"""Planner-driven expert parallelism on the MoE IR (ir/capture.moe_ir):
the SpmdTransform inserts the expert all-to-all at the
capacity-split/expert-split mismatch (reference kDAPPLEAllToAll,
SURVEY §2.7 EP), and the transformed graph executes on 2 gloo ranks with
the exact single-device loss (ample capacity: nothing drops)."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from tests.conftest import free_port

from tepdist_amd.ir.capture import moe_ir
from tepdist_amd.models.configs import MOE_CONFIGS
from tepdist_amd.planner.dist_spec import DimStrategy, DistSpec


def _graph():
    cfg = MOE_CONFIGS["gpt-moe-test"]
    # capacity factor 4: top-2 of 4 experts can never overflow -> exact
    return moe_ir(cfg, batch=4, seq=16, capacity_factor=4.0), cfg


def _ep_specs(g, n):
    """Hand EP plan: token-split everywhere, expert-split expert matmuls
    (what the cost search would express as DimStrategies)."""
    specs = {}
    for nid, node in g.nodes.items():
        if node.op == "moe_dispatch":
            specs[nid] = DimStrategy.split(1, n)
        elif node.op in ("matmul", "gelu") and len(node.shape) == 3:
            specs[nid] = DimStrategy.split(0, n)
        elif node.op == "param" and len(node.shape) == 3:
            specs[nid] = DimStrategy.split(0, n)   # expert weights
        elif node.op == "cross_entropy":
            specs[nid] = DimStrategy.partial(n)
        elif node.op in ("param", "data"):
            specs[nid] = DimStrategy.replicated(n)
        elif node.shape and node.shape[0] % n == 0 and \
                node.attrs.get("batch", 0) % n == 0:
            specs[nid] = DimStrategy.split(0, n)
        else:
            specs[nid] = DimStrategy.replicated(n)
    return specs


def test_transform_inserts_expert_all_to_all():
    from tepdist_amd.planner.transform import SpmdTransform
    g, cfg = _graph()
    res = SpmdTransform(g, _ep_specs(g, 2), 2).run()
    a2a = [n for n in res.graph.nodes.values() if n.op == "all_to_all"]
    n_moe = sum(1 for n in g.nodes.values() if n.op == "moe_dispatch")
    # one a2a into the expert matmuls + one back per MoE layer
    assert len(a2a) == 2 * n_moe, (len(a2a), n_moe)
    dims = {(n.attrs["src_dim"], n.attrs["dst_dim"]) for n in a2a}
    assert (1, 0) in dims and (0, 1) in dims
    # expert weights are sharded on the expert dim
    for name, (dim, nsh) in res.param_specs.items():
        if "moe_w" in name:
            assert (dim, nsh) == (0, 2), (name, dim, nsh)


def _worker(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from tepdist_amd.ir.interpreter import GraphInterpreter
        from tepdist_amd.planner.transform import SpmdTransform
        from tepdist_amd.runtime.initializers import (InitSpec,
                                                      default_init_spec,
                                                      init_shard)
        g, cfg = _graph()
        res = SpmdTransform(g, _ep_specs(g, world), world).run()

        def make_vars(graph, param_specs=None):
            out = {}
            for name, nid in graph.params.items():
                node = graph.nodes[nid]
                full = list(node.shape)
                dim, nsh = (param_specs or {}).get(name, (-1, 1))
                if dim >= 0:
                    full[dim] *= nsh
                t = init_shard(name, tuple(full),
                               default_init_spec(name, full),
                               shard_dim=dim, shard_index=rank,
                               num_shards=nsh, dtype=torch.float32)
                out[name] = t.requires_grad_()
            return out

        local = make_vars(res.graph, res.param_specs)
        gen = torch.Generator().manual_seed(3)
        feeds = {"input_ids": torch.randint(0, cfg.vocab_size, (64,),
                                            generator=gen),
                 "labels": torch.randint(0, cfg.vocab_size, (64,),
                                         generator=gen)}
        loss = list(GraphInterpreter(res.graph).run(feeds, local).values())[0]
        loss.backward()
        full = make_vars(g)
        ref = list(GraphInterpreter(g).run(feeds, full).values())[0]
        assert torch.allclose(loss.detach(), ref.detach(), rtol=1e-4,
                              atol=1e-5), (rank, float(loss), float(ref))
        # expert-weight grads are finite and local-shaped
        for name, t in local.items():
            if "moe_w" in name and t.grad is not None:
                assert torch.isfinite(t.grad).all(), name
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_ep_plan_executes_two_ranks():
    port = free_port()
    mp.spawn(_worker, args=(2, port), nprocs=2, join=True)
