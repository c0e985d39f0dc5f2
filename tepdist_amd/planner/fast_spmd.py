"""FastSpmdStrategy: one-pass annotation/rule-driven sharding inference.

Re-implements the reference's AnnotFastSpmdStrategy
(service/parallel/fast_spmd_strategy.{h,cc}, SURVEY.md §2.3: RULE_MODE
runs a single forward+backward inference sweep over every opcode instead
of the cone/ILP search). Round-1's "rule mode" was just the cost search
with the ILP budget zeroed; this is the real engine:

  1. SEED: user annotations pin their nodes (ir/sharding.py, unless
     IGNORE_ANNOTATION); un-annotated batch-carrying data inputs seed
     batch-dim splits — the reference's default token-parallel bias.
  2. FORWARD sweep (one pass, topological): each op picks the rule-table
     strategy (planner/rules.py: the Infer* tables) whose input demands
     agree with the producers' already-inferred specs on the most inputs;
     ties prefer split > partial > replicated outputs (keep parallelism).
  3. BACKWARD sweep: producers whose spec stayed replicated while every
     consumer demands the same split adopt it (the BackInfer* family),
     removing gather/slice pairs the forward pass left behind.

No ILP, no cone extraction: linear in graph size. The result is a
node_specs map the same SpmdTransform consumes.
"""

from __future__ import annotations

from typing import Dict

from tepdist_amd.config import get_env
from tepdist_amd.ir.graph import Graph
from tepdist_amd.planner.cost_model import CostModel
from tepdist_amd.planner.dist_spec import DimStrategy
from tepdist_amd.planner.rules import op_strategies
from tepdist_amd.planner.spmd import SpmdResult


def _rank(spec: DimStrategy) -> int:
    if spec.is_split:
        return 2
    if spec.is_partial:
        return 1
    return 0


class FastSpmdStrategy:
    def __init__(self, graph: Graph, nshards: int, cm: CostModel = None):
        self.g = graph
        self.n = nshards
        self.cm = cm or CostModel()
        if get_env().ignore_annotation:
            self.pins = {}
        else:
            from tepdist_amd.ir.sharding import collect_pins
            self.pins = collect_pins(graph, nshards)

    def run(self) -> SpmdResult:
        n = self.n
        if n <= 1:
            return SpmdResult({i: DimStrategy.replicated(1)
                               for i in self.g.nodes}, 0.0, False)
        g = self.g
        specs: Dict[int, DimStrategy] = {}

        # 1. seeds
        for nid, node in g.nodes.items():
            if nid in self.pins:
                specs[nid] = self.pins[nid]
            elif node.op == "data":
                b = node.attrs.get("batch", 0)
                if node.shape and b and b % n == 0 and \
                        node.shape[0] % n == 0:
                    specs[nid] = DimStrategy.split(0, n)
                else:
                    specs[nid] = DimStrategy.replicated(n)

        # 2. forward sweep
        for node in g.topo():
            if node.id in specs:
                continue
            if node.op == "param":
                specs[node.id] = DimStrategy.replicated(n)
                continue
            best = None
            best_key = None
            for st in op_strategies(g, node, n):
                ins = st.ins if st.ins else ()
                agree = sum(1 for i, want in zip(node.inputs, ins)
                            if specs.get(i) == want)
                key = (agree, _rank(st.out))
                if best is None or key > best_key:
                    best, best_key = st, key
            specs[node.id] = best.out if best is not None \
                else DimStrategy.replicated(n)

        # 3. backward sweep: lift replicated producers whose consumers all
        # demand one split (kills gather/slice churn)
        cons = g.consumers()
        for node in reversed(g.topo()):
            cur = specs.get(node.id)
            if cur is None or cur.is_split or node.id in self.pins:
                continue
            demands = set()
            for cid in cons.get(node.id, ()):  # what consumers want of us
                c = g.nodes[cid]
                cs = specs.get(cid)
                if cs is None:
                    continue
                for st in op_strategies(g, c, n):
                    if st.out != cs or not st.ins:
                        continue
                    for i, want in zip(c.inputs, st.ins):
                        if i == node.id:
                            demands.add(want)
                    break
            if len(demands) == 1:
                want = demands.pop()
                if want.is_split and any(
                        st.out == want
                        for st in op_strategies(g, node, n)):
                    specs[node.id] = want

        cost = sum(self.cm.compute_time(g, nd,
                                        n if specs[i].is_split else 1)
                   for i, nd in g.nodes.items())
        return SpmdResult(specs, cost, used_ilp=False)
