"""Liveness optimization: duplicate cheap long-live-range nodes.

The reference's HloLivenessOptimizer (service/parallel/
hlo_liveness_optimizer.h:22-31) clones inexpensive instructions whose value
stays live across a long stretch of the schedule, so each consumer gets a
fresh short-lived copy and peak activation memory drops. Same pass on our
IR: an elementwise/broadcast node whose consumers are far apart in
topological order is cloned per consumer."""

from __future__ import annotations

from tepdist_amd.ir.graph import ELEMENTWISE, Graph

CHEAP = ELEMENTWISE | {"const", "reshape", "split", "scale"}


def optimize_liveness(g: Graph, span_threshold: int = 8) -> int:
    """Clones cheap multi-consumer nodes whose consumer span (topo-index
    distance between first and last consumer) exceeds the threshold.
    Mutates g in place; returns the number of clones made."""
    topo_pos = {n.id: i for i, n in enumerate(g.topo())}
    cons = g.consumers()
    clones = 0
    for nid in list(g.nodes):
        n = g.nodes[nid]
        if n.op not in CHEAP or nid in g.outputs:
            continue
        users = cons[nid]
        if len(users) < 2:
            continue
        span = max(topo_pos[u] for u in users) - \
            min(topo_pos[u] for u in users)
        if span < span_threshold:
            continue
        # keep the original for the earliest consumer; clone for the rest
        ordered = sorted(users, key=lambda u: topo_pos[u])
        for u in ordered[1:]:
            dup = g.add(n.op, [g.nodes[i] for i in n.inputs], n.shape,
                        n.dtype, dict(n.attrs), n.op_group,
                        n.name + f".dup{clones}")
            dup.is_backward = n.is_backward
            un = g.nodes[u]
            un.inputs = [dup.id if i == nid else i for i in un.inputs]
            clones += 1
    return clones
