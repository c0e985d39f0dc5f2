"""Planner tests: strategy rules, cone/ILP SPMD search, pipeline stage-cut
ILP, sync-free analysis, ZeRO decision, and the AutoParallel driver's three
modes — all headless on the IR (the reference tests its planner passes as
pure HLO->HLO transforms)."""

import os

import pytest

from tepdist_amd.config import ServiceEnv, set_env, get_env
from tepdist_amd.ir import Graph, gpt2_ir
from tepdist_amd.models.configs import GPT2_CONFIGS
from tepdist_amd.planner import AutoParallel
from tepdist_amd.planner.dist_spec import DimStrategy, reshard_collective
from tepdist_amd.planner.pipeline import GraphSketch
from tepdist_amd.planner.rules import op_strategies
from tepdist_amd.planner.spmd import CostSpmdStrategy
from tepdist_amd.planner.sync_free import SyncFreeSplittingAnalysis
from tepdist_amd.planner.zero import plan_zero


@pytest.fixture(autouse=True)
def fresh_env():
    set_env(ServiceEnv())
    yield
    set_env(ServiceEnv())


def tiny_graph():
    return gpt2_ir(GPT2_CONFIGS["gpt2-test"], batch=8, seq=32)


def test_rules_linear_strategies():
    g = tiny_graph()
    lin = next(n for n in g.topo() if n.op == "linear")
    sts = op_strategies(g, lin, 2)
    notes = {s.note for s in sts}
    assert {"row", "col_tp", "row_tp", "rep"} <= notes
    # row split: x split dim0, weight replicated
    row = next(s for s in sts if s.note == "row")
    assert row.out.is_split and row.out.partition_dim == 0
    assert row.ins[1].is_replicated


def test_reshard_classification():
    S, R, P = DimStrategy.split, DimStrategy.replicated, DimStrategy.partial
    assert reshard_collective(P(4), R(4)) == "all_reduce"
    assert reshard_collective(P(4), S(0, 4)) == "reduce_scatter"
    assert reshard_collective(S(0, 4), R(4)) == "all_gather"
    assert reshard_collective(S(0, 4), S(1, 4)) == "all_to_all"
    assert reshard_collective(R(4), S(0, 4)) == "dynamic_slice"
    assert reshard_collective(S(0, 4), S(0, 4)) is None


def test_spmd_search_consistent_specs():
    g = tiny_graph()
    res = CostSpmdStrategy(g, 2, time_limit_s=5.0).run()
    assert res.cost > 0
    # every node got a spec and splits divide the dims
    for nid, sp in res.node_specs.items():
        n = g.nodes[nid]
        if sp.is_split:
            assert n.shape[sp.partition_dim] % 2 == 0, (n.op, n.shape, str(sp))


def test_pipeline_stage_cut_balance():
    g = gpt2_ir(GPT2_CONFIGS["gpt2-345m"], batch=8, seq=128)
    sk = GraphSketch(g)
    plan = sk.stage_plan(4)
    stages = set(plan.cluster_stage.values())
    assert stages == {0, 1, 2, 3}
    # monotone along layers
    layer_stage = {c: s for c, s in plan.cluster_stage.items()}
    keys = sorted(layer_stage)
    assert all(layer_stage[keys[i]] <= layer_stage[keys[i + 1]]
               for i in range(len(keys) - 1))
    # flop balance within slack
    cl = {c.id: c.flops for c in sk.build_clusters()}
    per_stage = {}
    for cid, s in plan.cluster_stage.items():
        per_stage[s] = per_stage.get(s, 0.0) + cl[cid]
    total = sum(per_stage.values())
    for s, f in per_stage.items():
        assert abs(f - total / 4) <= total / 4 * 0.6, per_stage


def test_sync_free_analysis():
    g = tiny_graph()
    sf = SyncFreeSplittingAnalysis(g).run()
    assert sf is not None
    assert 1 in sf.micro_batches and 4 in sf.micro_batches
    assert sf.batch_dim == 0


def test_zero_plan_triggers_on_memory():
    g = gpt2_ir(GPT2_CONFIGS["gpt2-345m"], batch=8, seq=128)
    zp = plan_zero(g, dp_degree=8, var_mem_limit=1 << 30)  # 1 GB: too small
    assert zp.shard_optimizer and zp.shard_degree == 8
    assert zp.per_device_state_bytes < zp.param_bytes * 8
    zp2 = plan_zero(g, dp_degree=8, var_mem_limit=1 << 40)
    assert not zp2.shard_optimizer


def test_auto_parallel_exploration_picks_dp_for_small_model():
    g = gpt2_ir(GPT2_CONFIGS["gpt2-345m"], batch=32, seq=256)
    plan = AutoParallel(g, 8).run()
    assert plan.dp * plan.tp * plan.pp == 8
    assert plan.cost.total_duration < float("inf")
    assert plan.dp >= 2  # DP should dominate for a 345M model at batch 32
    assert plan.search_time_s < 60


def test_auto_parallel_config_mode_stages():
    os.environ["NUM_STAGES"] = "4"
    os.environ["NUM_MICRO_BATCHES"] = "8"
    set_env(ServiceEnv())
    try:
        g = gpt2_ir(GPT2_CONFIGS["gpt2-345m"], batch=32, seq=128)
        plan = AutoParallel(g, 8).run()
        assert plan.mode == "config"
        assert plan.pp == 4
        assert plan.micro_batches == 8
        assert set(plan.node_stage.values()) == {0, 1, 2, 3}
        assert plan.cost.bubble_ratio > 0
    finally:
        del os.environ["NUM_STAGES"]
        del os.environ["NUM_MICRO_BATCHES"]


def test_auto_parallel_rule_mode():
    os.environ["RULE_MODE"] = "1"
    set_env(ServiceEnv())
    try:
        g = tiny_graph()
        plan = AutoParallel(g, 4).run()
        assert plan.mode == "rule"
        assert plan.dp * plan.tp * plan.pp == 4
    finally:
        del os.environ["RULE_MODE"]


def test_def_tree_structure():
    g = tiny_graph()
    plan = AutoParallel(g, 4).run()
    tree = plan.def_tree
    entry = tree.entry()
    kinds = {tree.contexts[c].kind for c in entry.children}
    assert {"CG", "GA_INIT", "GA", "AG"} <= kinds
    s = tree.to_json()
    t2 = type(tree).from_json(s)
    assert len(t2.contexts) == len(tree.contexts)


def test_memory_pressure_prefers_weight_sharding():
    """A model whose optimizer state cannot fit replicated must come back
    weight-sharded (the SplitPlanByMemCost bias) with an explicit
    min-memory fallback when nothing fits."""
    from tepdist_amd.ir.capture import gpt2_ir
    from tepdist_amd.models.configs import GPT2_CONFIGS
    from tepdist_amd.planner.auto_parallel import AutoParallel

    cfg = GPT2_CONFIGS["gpt2-175b"]
    g = gpt2_ir(cfg, batch=64, seq=64)
    plan = AutoParallel(g, 8).run()
    # dp-only would replicate 2.8 TB of state per device; the planner must
    # shift the mesh toward tensor sharding
    assert plan.tp > 1, plan.summary()
