"""User sharding annotations on IR tensors.

Mirrors the reference's xla_sharding API
(experimental/xla_sharding/xla_sharding.py:234-290: split / replicate
annotations attached to tensors, consumed by
CostSpmdStrategy::ExtractUserSplit, cost_spmd_strategy.cc:588-680, and by
the rule-mode AnnotFastSpmdStrategy). Annotations are attrs on Graph
nodes; `IGNORE_ANNOTATION=1` (config.py) makes every planner drop them —
the same kill-switch the reference has.

    from tepdist_amd.ir.sharding import split, replicate
    wte = g.add_param("wte", (V, d))
    split(wte, 0)            # pin: shard the vocab dim
    replicate(gate_w)        # pin: keep the MoE gate replicated
"""

from __future__ import annotations

from typing import Optional

from tepdist_amd.ir.graph import Graph, Node
from tepdist_amd.planner.dist_spec import DimStrategy


def split(node: Node, dim: int) -> Node:
    """Request this tensor be partitioned on `dim` (the reference's
    xla_sharding.split)."""
    node.attrs["user_spec"] = ["split", int(dim)]
    return node


def replicate(node: Node) -> Node:
    """Request this tensor stay replicated on every shard."""
    node.attrs["user_spec"] = ["rep"]
    return node


def clear(node: Node) -> Node:
    node.attrs.pop("user_spec", None)
    return node


def user_spec(node: Node, nshards: int) -> Optional[DimStrategy]:
    """The pinned DimStrategy for `node` at `nshards`, or None (also None
    when the requested dim does not divide — annotation is a request, not
    a proof obligation)."""
    us = node.attrs.get("user_spec")
    if not us:
        return None
    if us[0] == "rep":
        return DimStrategy.replicated(nshards)
    if us[0] == "split":
        d = int(us[1])
        if 0 <= d < len(node.shape) and node.shape[d] % nshards == 0:
            return DimStrategy.split(d, nshards)
    return None


def collect_pins(g: Graph, nshards: int):
    """{node id: pinned DimStrategy} for every annotated node (the
    reference's ExtractUserSplit)."""
    out = {}
    for nid, n in g.nodes.items():
        s = user_spec(n, nshards)
        if s is not None:
            out[nid] = s
    return out
