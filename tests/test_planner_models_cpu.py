"""Planner coverage across the BASELINE model families: Wide-ResNet
auto-SPMD planning (image path) and the GPT-2 175B memory-infeasibility
fallback (288 GB HBM3E per device is not enough for fp32 optimizer
states at that size — the planner must say so, not silently pick)."""

import math

import pytest

from tepdist_amd.ir.capture import wrn_ir
from tepdist_amd.models.configs import WIDE_RESNET_CONFIGS
from tepdist_amd.planner.auto_parallel import AutoParallel


@pytest.mark.timeout(600)
def test_wide_resnet_auto_spmd_plan():
    g = wrn_ir(WIDE_RESNET_CONFIGS["wrn-250m"], batch=64 * 8)
    plan = AutoParallel(g, 8).run()
    assert math.isfinite(plan.cost.total_duration)
    assert plan.dp * plan.tp * plan.pp == 8
    # image nets split the batch: most conv nodes carry a dim-0 split
    convs = [i for i, n in g.nodes.items() if n.op == "conv2d"]
    n_split = 0
    for i in convs:
        ds = plan.node_specs.get(i)
        for r in range(len(plan.mesh_rounds) or 1):
            s = ds.round(r)
            if s.is_split and s.partition_dim == 0:
                n_split += 1
                break
    assert n_split >= len(convs) * 0.5, (n_split, len(convs))


@pytest.mark.timeout(900)
def test_gpt2_175b_infeasible_on_8_gpus(capsys):
    """96 x 12288 GPT-2: ~175B params x 16 B/param of state is ~2.8 TB —
    over 8 x 288 GB even ZeRO-sharded. The planner returns the
    minimum-memory layout WITH an explicit warning (round-1 behavior,
    kept honest by the transformed-graph memory model)."""
    from tepdist_amd.ir.capture import gpt2_ir
    from tepdist_amd.models.configs import GPT2_CONFIGS
    cfg = GPT2_CONFIGS["gpt2-175b"]
    g = gpt2_ir(cfg, batch=8, seq=2048)
    plan = AutoParallel(g, 8).run()
    assert plan is not None
    out = capsys.readouterr().out
    infeasible = not math.isfinite(plan.cost.total_duration)
    assert infeasible, plan.summary()
    assert "WARNING" in out and "fits" in out


@pytest.mark.timeout(900)
def test_gpt2_175b_plans_on_64_gpus():
    """The multi-node scale the reference targets: 96 x 12288 GPT-2 over
    64 devices. ~2.8 TB of parameter+optimizer state against 64 x 288 GB
    = 18.4 TB is feasible — the planner must return a FINITE plan that
    uses model parallelism (pp and/or a sharding mesh), and its memory
    model must agree it fits."""
    from tepdist_amd.ir.capture import gpt2_ir
    from tepdist_amd.models.configs import GPT2_CONFIGS
    cfg = GPT2_CONFIGS["gpt2-175b"]
    g = gpt2_ir(cfg, batch=64, seq=2048)
    plan = AutoParallel(g, 64).run()
    assert plan is not None
    assert math.isfinite(plan.cost.total_duration), plan.summary()
    assert plan.dp * plan.tp * plan.pp == 64
    # pure data parallelism cannot hold 175B: some model split must exist
    assert plan.tp * plan.pp > 1 or plan.zero.shard_optimizer, \
        plan.summary()
