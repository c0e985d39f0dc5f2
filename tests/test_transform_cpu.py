"""SpmdTransform tests: applying a planned sharding to the IR must produce
a per-rank graph that computes the SAME loss and gradients as the
unsharded graph (the reference validates SpmdTransform the same way:
sharded module outputs == single-device outputs).

Runs on CPU with gloo world_size=2 (the multi-process harness the GPU
node reuses over RCCL)."""

import os

import torch

from tests.conftest import free_port
import torch.distributed as dist
import torch.multiprocessing as mp

from tepdist_amd.ir.graph import Graph
from tepdist_amd.ir.interpreter import GraphInterpreter
from tepdist_amd.planner.dist_spec import DimStrategy
from tepdist_amd.planner.transform import SpmdTransform

WORLD = 2
R, S, P = (DimStrategy.replicated, DimStrategy.split, DimStrategy.partial)


def _run(fn, world=WORLD):
    port = free_port()
    mp.spawn(fn, args=(world, port), nprocs=world, join=True)


def _init(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    torch.manual_seed(0)


# -- a 2-layer MLP + CE in IR form ----------------------------------------


def _mlp_ir(rows=8, din=16, dh=32, klass=10):
    g = Graph()
    x = g.add_input("x", (rows, din), "f32")
    t = g.add_input("t", (rows,), "i64")
    w1 = g.add_param("w1", (dh, din), "f32")
    b1 = g.add_param("b1", (dh,), "f32")
    w2 = g.add_param("w2", (klass, dh), "f32")
    h = g.add("linear", [x, w1, b1], (rows, dh), "f32", {"act": "gelu"})
    y = g.add("linear", [h, w2], (rows, klass), "f32")
    loss = g.add("cross_entropy", [y, t], (), "f32")
    g.outputs = [loss.id]
    return g, (x, t, w1, b1, w2, h, y, loss)


def _mlp_data(rows=8, din=16, dh=32, klass=10):
    gen = torch.Generator().manual_seed(7)
    feeds = {"x": torch.randn(rows, din, generator=gen),
             "t": torch.randint(0, klass, (rows,), generator=gen)}
    variables = {"w1": torch.randn(dh, din, generator=gen) * 0.3,
                 "b1": torch.zeros(dh),
                 "w2": torch.randn(klass, dh, generator=gen) * 0.3}
    return feeds, variables


def _reference_loss_and_grads():
    g, _ = _mlp_ir()
    feeds, variables = _mlp_data()
    for v in variables.values():
        v.requires_grad_()
    loss = list(GraphInterpreter(g).run(feeds, variables).values())[0]
    loss.backward()
    return (loss.detach(),
            {k: v.grad.clone() for k, v in variables.items()})


def _shard_param(t, dim, n, rank):
    if dim < 0 or n == 1:
        return t
    sz = t.shape[dim] // n
    return t.narrow(dim, rank * sz, sz).contiguous()


# -- tensor parallel (Megatron column->row) --------------------------------


def _tp_specs(nodes, n):
    x, t, w1, b1, w2, h, y, loss = nodes
    return {x.id: R(n), t.id: R(n),
            w1.id: S(0, n), b1.id: S(0, n),   # column parallel
            h.id: S(1, n),
            w2.id: S(1, n),                    # row parallel (K split)
            y.id: P(n),                        # partial sums
            loss.id: R(n)}


def _tp_worker(rank, world, port):
    _init(rank, world, port)
    g, nodes = _mlp_ir()
    res = SpmdTransform(g, _tp_specs(nodes, world), world).run()
    feeds, variables = _mlp_data()
    full = {k: v.clone() for k, v in variables.items()}
    local = {k: _shard_param(v, *res.param_specs[k], rank).requires_grad_()
             for k, v in variables.items()}
    loss = list(GraphInterpreter(res.graph).run(feeds, local).values())[0]
    loss.backward()

    ref_loss, ref_grads = _reference_loss_and_grads()
    assert torch.allclose(loss.detach(), ref_loss, rtol=1e-5, atol=1e-6), \
        (loss.item(), ref_loss.item())
    for k, v in local.items():
        dim, n = res.param_specs[k]
        want = _shard_param(ref_grads[k], dim, n, rank)
        assert torch.allclose(v.grad, want, rtol=1e-4, atol=1e-6), k
    dist.destroy_process_group()


def test_transform_tensor_parallel_matches_single():
    _run(_tp_worker)


# -- data parallel (row split via planner) ---------------------------------


def _dp_specs(nodes, n):
    x, t, w1, b1, w2, h, y, loss = nodes
    return {x.id: S(0, n), t.id: S(0, n),
            w1.id: R(n), b1.id: R(n), w2.id: R(n),
            h.id: S(0, n), y.id: S(0, n), loss.id: P(n)}


def _dp_worker(rank, world, port):
    _init(rank, world, port)
    g, nodes = _mlp_ir()
    res = SpmdTransform(g, _dp_specs(nodes, world), world).run()
    feeds, variables = _mlp_data()
    local = {k: v.clone().requires_grad_() for k, v in variables.items()}
    loss = list(GraphInterpreter(res.graph).run(feeds, local).values())[0]
    loss.backward()

    ref_loss, ref_grads = _reference_loss_and_grads()
    assert torch.allclose(loss.detach(), ref_loss, rtol=1e-5, atol=1e-6)
    # replicated params: grad contributions were all-reduced (copy_to);
    # each rank computed mean-loss over its half, summed over ranks by the
    # backward all-reduce, and the output all-reduce averaged the loss —
    # so grads match the full-batch mean-loss grads exactly
    for k, v in local.items():
        assert torch.allclose(v.grad, ref_grads[k], rtol=1e-4, atol=1e-6), k
    dist.destroy_process_group()


def test_transform_data_parallel_matches_single():
    _run(_dp_worker)


# -- planner-driven end to end ---------------------------------------------


def _planned_worker(rank, world, port):
    _init(rank, world, port)
    from tepdist_amd.planner.spmd import CostSpmdStrategy
    g, _ = _mlp_ir(rows=8, din=16, dh=32, klass=10)
    plan = CostSpmdStrategy(g, world).run()
    res = SpmdTransform(g, plan.node_specs, world).run()
    feeds, variables = _mlp_data()
    local = {k: _shard_param(v, *res.param_specs[k], rank).requires_grad_()
             for k, v in variables.items()}
    loss = list(GraphInterpreter(res.graph).run(feeds, local).values())[0]
    loss.backward()
    ref_loss, _ = _reference_loss_and_grads()
    assert torch.allclose(loss.detach(), ref_loss, rtol=1e-4, atol=1e-5), \
        (loss.item(), ref_loss.item())
    for v in local.values():
        assert v.grad is None or torch.isfinite(v.grad).all()
    dist.destroy_process_group()


def test_transform_planner_driven_matches_single():
    _run(_planned_worker)


# -- single-process shape/structure checks ---------------------------------


def test_transform_inserts_expected_collectives():
    g, nodes = _mlp_ir()
    res = SpmdTransform(g, _tp_specs(nodes, 2), 2).run()
    ops = [n.op for n in res.graph.topo()]
    assert "all_reduce" in ops          # partial y -> replicated CE input
    # x carries no gradient, so no copy_to wrapper for it; replicated
    # PARAMS in a sharded region do get one (DP case below)
    dp = SpmdTransform(g, _dp_specs(nodes, 2), 2).run()
    assert "copy_to" in [n.op for n in dp.graph.topo()]
    assert res.param_specs["w1"] == (0, 2)
    assert res.param_specs["w2"] == (1, 2)
    # local shapes halved on the sharded dims
    w1l = res.graph.nodes[res.graph.params["w1"]]
    assert tuple(w1l.shape) == (16, 16)


def test_transform_single_shard_is_identity():
    g, _ = _mlp_ir()
    res = SpmdTransform(g, {}, 1).run()
    assert res.graph is g
