"""Graph-structure recognition: forward / backward / apply partition.

The reference's resolve_utils (service/parallel/resolve_utils.h:30-49)
recognizes the optimizer structure inside the HLO module
(`ResolveGradients` for GradientDescent / AdamWeightDecay patterns) and
partitions instructions into forward, backward and apply-gradient sets
(`ResolveForwardBackwardAndApplyGradients`, utils.cc:124-234: forward =
reachability to the loss output, backward = reachability to the variable
updates). Our IR carries the client's op_group/backward metadata when the
capture provides it; these helpers recover the partition structurally when
it does not."""

from __future__ import annotations

from typing import Dict, List, Set

from tepdist_amd.ir.graph import Graph

APPLY_OPS = {"adamw_update", "sgd_update", "apply_gradient"}


def _ancestors(g: Graph, roots: List[int]) -> Set[int]:
    seen: Set[int] = set()
    stack = list(roots)
    while stack:
        i = stack.pop()
        if i in seen:
            continue
        seen.add(i)
        stack.extend(g.nodes[i].inputs)
    return seen


def find_apply_insts(g: Graph) -> Set[int]:
    return {n.id for n in g.topo() if n.op in APPLY_OPS}


def find_backward_insts(g: Graph) -> Set[int]:
    """Backward = flagged nodes, plus everything only reachable from the
    apply/update outputs and not from the loss."""
    flagged = {n.id for n in g.topo() if n.is_backward}
    if flagged:
        return flagged
    apply_ = find_apply_insts(g)
    if not apply_:
        return set()
    fwd = find_forward_insts(g)
    return _ancestors(g, list(apply_)) - fwd - apply_


def find_forward_insts(g: Graph) -> Set[int]:
    """Forward = ancestors of the loss (first graph output), the
    reference's FindForwardInsts reachability."""
    if not g.outputs:
        return set(g.nodes)
    loss = [g.outputs[0]]
    back = {n.id for n in g.topo() if n.is_backward}
    return _ancestors(g, loss) - back


def resolve_gradients(g: Graph) -> Dict[str, int]:
    """param name -> gradient-producing node id, for graphs that encode
    the backward explicitly (apply ops consume (param, grad, ...))."""
    grads: Dict[str, int] = {}
    rev_params = {v: k for k, v in g.params.items()}
    for n in g.topo():
        if n.op not in APPLY_OPS or len(n.inputs) < 2:
            continue
        pname = rev_params.get(n.inputs[0])
        if pname is not None:
            grads[pname] = n.inputs[1]
    return grads
