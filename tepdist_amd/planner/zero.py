"""ZeRO-style variable-sharding decision under a memory budget.

Re-implements the intent of the reference's SplitPlanByMemCost
(cost_spmd_strategy.cc:1487) gated by VAR_MEM_LIMIT: when a device's
parameter + optimizer-state + gradient footprint exceeds the budget, shard
optimizer states (and optionally parameters) across the data-parallel
group, paying an all-gather on use. Returns the chosen shard degree and
the predicted per-device memory."""

from __future__ import annotations

from dataclasses import dataclass

from tepdist_amd.config import get_env
from tepdist_amd.ir.graph import Graph


@dataclass
class ZeroPlan:
    shard_optimizer: bool
    shard_degree: int
    param_bytes: float
    per_device_state_bytes: float


def plan_zero(graph: Graph, dp_degree: int,
              var_mem_limit: int = None) -> ZeroPlan:
    limit = var_mem_limit if var_mem_limit is not None \
        else get_env().var_mem_limit_bytes
    param_bytes = sum(graph.bytes_of(graph.nodes[i])
                      for i in graph.params.values())
    # bf16 param + bf16 grad + fp32 master + 2x fp32 Adam moments
    full_state = param_bytes * (1 + 1 + 2 + 4)
    if full_state <= limit or dp_degree <= 1:
        return ZeroPlan(False, 1, param_bytes, full_state)
    # ZeRO-1: master + moments sharded across dp
    sharded = param_bytes * 2 + param_bytes * 6 / dp_degree
    return ZeroPlan(True, dp_degree, param_bytes, sharded)
