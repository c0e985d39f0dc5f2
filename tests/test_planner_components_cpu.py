"""Planner sub-component tests: InstAffinityMap, NeighborVote fallback,
liveness optimizer, all-reduce combiner, resolve utils (the reference's
inst_affinity_map.h / cost_spmd_strategy.cc:708 /
hlo_liveness_optimizer.h / dapple_all_reduce_combiner.h /
resolve_utils.h counterparts)."""

import torch

from tepdist_amd.ir.graph import Graph
from tepdist_amd.ir.interpreter import GraphInterpreter
from tepdist_amd.planner.affinity import InstAffinityMap
from tepdist_amd.planner.combiner import combine_all_reduces
from tepdist_amd.planner.dist_spec import DimStrategy
from tepdist_amd.planner.liveness import optimize_liveness
from tepdist_amd.planner.resolve import (find_backward_insts,
                                         find_forward_insts,
                                         resolve_gradients)

R, S = DimStrategy.replicated, DimStrategy.split


def test_affinity_groups_aux_vars():
    g = Graph()
    w = g.add_param("w", (8, 8))
    m = g.add_param("opt.m.w", (8, 8))
    v = g.add_param("opt.v.w", (8, 8))
    am = InstAffinityMap(g, aux_affinity=True)
    assert any(set(grp) >= {w.id, m.id, v.id} for grp in am.groups)
    specs = {w.id: S(0, 2)}
    am.apply(specs)
    assert specs[m.id] == S(0, 2) and specs[v.id] == S(0, 2)


def test_affinity_union():
    g = Graph()
    a = g.add_param("a", (4,))
    b = g.add_param("b", (4,))
    c = g.add_param("c", (4,))
    am = InstAffinityMap(g, aux_affinity=False)
    am.add_affinity(a.id, b.id)
    am.add_affinity(b.id, c.id)
    specs = am.apply({b.id: S(0, 2)})
    assert specs[a.id] == specs[c.id] == S(0, 2)


def test_liveness_clone_preserves_semantics():
    g = Graph()
    x = g.add_input("x", (4, 8), "f32")
    sc = g.add("scale", [x], (4, 8), "f32", {"scale": 2.0})
    cur = sc
    for _ in range(10):  # long chain so sc's consumers are far apart
        cur = g.add("gelu", [cur], (4, 8), "f32")
    out = g.add("add", [cur, sc], (4, 8), "f32")   # distant second use
    g.outputs = [out.id]
    feeds = {"x": torch.randn(4, 8)}
    before = list(GraphInterpreter(g).run(feeds, {}).values())[0]
    n = optimize_liveness(g, span_threshold=4)
    assert n >= 1
    after = list(GraphInterpreter(g).run(feeds, {}).values())[0]
    assert torch.allclose(before, after)


def test_combiner_bundles_small_all_reduces():
    g = Graph()
    a = g.add_input("a", (8,), "f32")
    b = g.add_input("b", (16,), "f32")
    ra = g.add("all_reduce", [a], (8,), "f32")
    rb = g.add("all_reduce", [b], (16,), "f32")
    g.outputs = [ra.id, rb.id]
    made = combine_all_reduces(g)
    assert made == 1
    ops = [n.op for n in g.topo()]
    assert ops.count("all_reduce") == 1
    assert ops.count("bundle_get") == 2
    # single-process execution: all_reduce is identity; outputs preserved
    feeds = {"a": torch.arange(8.0), "b": torch.arange(16.0)}
    outs = GraphInterpreter(g).run(feeds, {})
    vals = list(outs.values())
    assert torch.allclose(vals[0], torch.arange(8.0))
    assert torch.allclose(vals[1], torch.arange(16.0))


def test_combiner_respects_dependence():
    g = Graph()
    a = g.add_input("a", (8,), "f32")
    r1 = g.add("all_reduce", [a], (8,), "f32")
    mid = g.add("gelu", [r1], (8,), "f32")
    r2 = g.add("all_reduce", [mid], (8,), "f32")  # depends on r1
    g.outputs = [r2.id]
    assert combine_all_reduces(g) == 0


def test_resolve_forward_backward_apply():
    g = Graph()
    x = g.add_input("x", (4, 8), "f32")
    w = g.add_param("w", (8, 8), "f32")
    y = g.add("linear", [x, w], (4, 8), "f32")
    loss = g.add("cross_entropy", [y, g.add_input("t", (4,), "i64")], (),
                 "f32")
    gw = g.add("matmul", [y, x], (8, 8), "f32")
    gw.is_backward = True
    upd = g.add("adamw_update", [w, gw], (8, 8), "f32")
    g.outputs = [loss.id]
    fwd = find_forward_insts(g)
    bwd = find_backward_insts(g)
    assert y.id in fwd and loss.id in fwd
    assert gw.id in bwd and y.id not in bwd
    assert resolve_gradients(g) == {"w": gw.id}


def test_auto_parallel_dump_artifacts(tmp_path, monkeypatch):
    import json

    from tepdist_amd.ir.capture import gpt2_ir
    from tepdist_amd.models.configs import GPT2_CONFIGS
    from tepdist_amd.planner.auto_parallel import AutoParallel
    import tepdist_amd.config as cfgmod

    monkeypatch.setenv("TEPDIST_DUMP_DIR", str(tmp_path))
    monkeypatch.setattr(cfgmod, "_GLOBAL_ENV", None)  # drop the env cache
    g = gpt2_ir(GPT2_CONFIGS["gpt2-test"], batch=4, seq=16)
    try:
        AutoParallel(g, 2).run()
    finally:
        cfgmod._GLOBAL_ENV = None
    d = json.load(open(tmp_path / "strategies.json"))
    assert d["dp"] * d["tp"] * d["pp"] == 2
    assert d["node_specs"]
    assert (tmp_path / "dag.dot").exists()
