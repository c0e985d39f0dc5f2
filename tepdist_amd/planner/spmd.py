"""Cost-based auto-SPMD search for one device-mesh dimension.

A faithful re-implementation of the reference CostSpmdStrategy's structure
(service/parallel/cost_spmd_strategy.cc) on our IR:

  1. split the graph into sub-graphs at critical nodes (single-tensor cut
     points, GraphSketch::FindCriticalInsts);
  2. inside each sub-graph, extract CONES rooted at compute-sensitive ops
     (dot/conv class) — every other node joins its unique consumer's cone;
  3. enumerate per-cone strategies by back-inference from the root's
     candidate shardings (rules.py);
  4. choose one strategy per cone with an ILP minimizing compute +
     inter-cone reshard cost (scipy.optimize.milp = HiGHS, standing in for
     the reference's COIN-OR CBC), linearized with edge product variables;
  5. combine sub-graphs with dynamic programming over the boundary
     tensor's spec (dp_combine_graph).

Falls back to greedy forward inference when the ILP exceeds the time limit
(the reference's InferByRank heuristics)."""

from __future__ import annotations

import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import numpy as np

from tepdist_amd.config import get_env
from tepdist_amd.ir.graph import COMPUTE_SENSITIVE, Graph, Node
from tepdist_amd.planner.cost_model import CostModel
from tepdist_amd.planner.dist_spec import DimStrategy, reshard_collective
from tepdist_amd.planner.rules import OpStrategy, back_infer, op_strategies


@dataclass
class Cone:
    root: int
    members: List[int] = field(default_factory=list)  # includes root
    strategies: List[Dict[int, DimStrategy]] = field(default_factory=list)
    strat_in: List[Dict[int, DimStrategy]] = field(default_factory=list)
    costs: List[float] = field(default_factory=list)


@dataclass
class SpmdResult:
    node_specs: Dict[int, DimStrategy]
    cost: float
    used_ilp: bool


class CostSpmdStrategy:
    def __init__(self, graph: Graph, nshards: int, cm: CostModel = None,
                 time_limit_s: float = None,
                 param_mem_penalty: float = 0.0):
        self.g = graph
        self.n = nshards
        self.cm = cm or CostModel()
        self.time_limit = time_limit_s if time_limit_s is not None \
            else get_env().ilp_time_limit_s
        # SplitPlanByMemCost's role (reference cost_spmd_strategy.cc:1487):
        # under memory pressure, an UNsplit parameter pays this many
        # seconds per byte of replicated weight, steering the ILP toward
        # weight-sharded (tensor-parallel) strategies
        self.param_mem_penalty = param_mem_penalty
        self.cons = graph.consumers()
        # user annotations (the reference's ExtractUserSplit,
        # cost_spmd_strategy.cc:588-680; IGNORE_ANNOTATION drops them)
        if get_env().ignore_annotation:
            self.pins = {}
        else:
            from tepdist_amd.ir.sharding import collect_pins
            self.pins = collect_pins(graph, nshards)

    # ---------------------------------------------------------------------

    def run(self) -> SpmdResult:
        if self.n <= 1:
            return SpmdResult({i: DimStrategy.replicated(1)
                               for i in self.g.nodes}, 0.0, False)
        t0 = time.time()
        subgraphs = self._split_subgraphs()
        all_specs: Dict[int, DimStrategy] = {}
        total = 0.0
        used_ilp = True
        prev_best: Dict[str, Tuple[float, Dict[int, DimStrategy]]] = {"": (0.0, {})}
        prev_out_node: Optional[int] = None

        for sub in subgraphs:
            cones = self._extract_cones(sub)
            self._populate_cone_strategies(cones)
            budget = self.time_limit - (time.time() - t0)
            cands = self._subgraph_candidates(cones, sub, budget)
            if not cands:
                used_ilp = False
                cands = [self._greedy(cones)]
            # DP over boundary spec: combine with previous best
            new_best: Dict[str, Tuple[float, Dict[int, DimStrategy]]] = {}
            for (out_spec, in_demand, icost, specs) in cands:
                for key, (pcost, pspecs) in prev_best.items():
                    rcost = 0.0
                    if prev_out_node is not None and prev_out_node in pspecs:
                        src = pspecs[prev_out_node]
                        dst = in_demand.get(prev_out_node, src)
                        kind = reshard_collective(src, dst)
                        rcost = self.cm.collective(
                            kind, self.g.bytes_of(self.g.nodes[prev_out_node]),
                            self.n)
                    cost = pcost + rcost + icost
                    k = str(out_spec)
                    if k not in new_best or cost < new_best[k][0]:
                        merged = dict(pspecs)
                        merged.update(specs)
                        new_best[k] = (cost, merged)
            prev_best = new_best
            prev_out_node = sub[-1]

        best_cost, best_specs = min(prev_best.values(), key=lambda v: v[0])
        # uncovered nodes (dead code etc.): NeighborVote fallback (the
        # reference's cost_spmd_strategy.cc:708) then replicated
        self._neighbor_vote(best_specs)
        for i in self.g.nodes:
            best_specs.setdefault(i, DimStrategy.replicated(self.n))
        # user annotations are a CONTRACT: enforce them on the result
        # (SpmdTransform reshards any producer/consumer mismatch)
        best_specs.update(self.pins)
        # co-location affinity (InstAffinityMap): aux optimizer vars and
        # aliased pairs adopt their leader's spec
        from tepdist_amd.planner.affinity import InstAffinityMap
        InstAffinityMap(self.g).apply(best_specs)
        return SpmdResult(best_specs, best_cost, used_ilp)

    def _neighbor_vote(self, specs) -> None:
        """Nodes without a planned spec adopt the majority strategy of
        their planned neighbors (producers + consumers)."""
        from collections import Counter
        for n in self.g.topo():
            if n.id in specs:
                continue
            votes = Counter()
            for i in list(n.inputs) + self.cons[n.id]:
                sp = specs.get(i)
                if sp is not None and not sp.is_glue:
                    votes[sp] += 1
            if votes:
                cand = votes.most_common(1)[0][0]
                # only adopt if the op supports it exactly
                from tepdist_amd.planner.rules import back_infer
                if back_infer(self.g, n, cand, self.n) is not None:
                    specs[n.id] = cand

    # ---------------------------------------------------------------------

    def _split_subgraphs(self) -> List[List[int]]:
        crit = set(self.g.critical_nodes())
        subs: List[List[int]] = []
        cur: List[int] = []
        for n in self.g.topo():
            cur.append(n.id)
            if n.id in crit:
                subs.append(cur)
                cur = []
        if cur:
            subs.append(cur)
        return subs

    def _extract_cones(self, sub: List[int]) -> List[Cone]:
        inset = set(sub)
        cone_of: Dict[int, int] = {}
        cones: Dict[int, Cone] = {}
        # walk in reverse topological order
        for nid in reversed(sub):
            n = self.g.nodes[nid]
            consumers = [c for c in self.cons[nid] if c in inset]
            if (n.op in COMPUTE_SENSITIVE or not consumers or
                    len({cone_of.get(c) for c in consumers}) != 1 or
                    cone_of.get(consumers[0]) is None):
                cones[nid] = Cone(root=nid, members=[nid])
                cone_of[nid] = nid
            else:
                root = cone_of[consumers[0]]
                cones[root].members.append(nid)
                cone_of[nid] = root
        self.cone_of = cone_of
        return [cones[r] for r in sorted(cones)]

    def _populate_cone_strategies(self, cones: List[Cone]):
        for cone in cones:
            root = self.g.nodes[cone.root]
            pin = self.pins.get(cone.root)
            cands = op_strategies(self.g, root, self.n)
            if pin is not None:
                # annotated root: only strategies producing the pinned
                # spec survive (fall back to all if none match — the
                # final enforcement pass still applies the pin)
                matched = [st for st in cands if st.out == pin]
                cands = matched or cands
            for st in cands:
                specs, in_demand, cost = self._grow(cone, st)
                cone.strategies.append(specs)
                cone.strat_in.append(in_demand)
                cone.costs.append(cost)

    def _grow(self, cone: Cone, root_st: OpStrategy):
        """Back-infer specs for every cone member from the root strategy.
        Returns (member specs, demanded specs of external inputs, cost)."""
        specs: Dict[int, DimStrategy] = {cone.root: root_st.out}
        demand: Dict[int, DimStrategy] = {}
        cost = 0.0
        members = set(cone.members)
        # record the root's demands
        root = self.g.nodes[cone.root]
        for inp, want in zip(root.inputs, root_st.ins):
            (specs if inp in members else demand)[inp] = want
        # members are stored root-first (reverse topo); process forward
        for nid in cone.members:
            if nid == cone.root:
                cost += self._node_cost(root, root_st.out)
                continue
            n = self.g.nodes[nid]
            want = specs.get(nid, DimStrategy.replicated(self.n))
            st = back_infer(self.g, n, want, self.n)
            if st is None:
                # no exact rule: replicate this member and pay a gather
                st = back_infer(self.g, n, DimStrategy.replicated(self.n),
                                self.n)
                specs[nid] = DimStrategy.replicated(self.n)
                cost += self.cm.collective("all_gather", self.g.bytes_of(n),
                                           self.n)
            else:
                specs[nid] = want
            for inp, w in zip(n.inputs, st.ins):
                (specs if inp in members else demand).setdefault(inp, w)
            cost += self._node_cost(n, specs[nid])
        return specs, demand, cost

    def _node_cost(self, n: Node, spec: DimStrategy) -> float:
        shards = self.n if (spec.is_split or spec.is_partial) else 1
        c = self.cm.compute_time(self.g, n, shards)
        if self.param_mem_penalty > 0 and n.op == "param" \
                and not spec.is_split:
            c += self.param_mem_penalty * self.g.bytes_of(n)
        return c

    # ---------------------------------------------------------------------

    def _inter_cone_edges(self, cones: List[Cone], sub: List[int]):
        inset = set(sub)
        edges = []
        for cone in cones:
            for nid in cone.members:
                for inp in self.g.nodes[nid].inputs:
                    if inp in inset and self.cone_of[inp] != cone.root:
                        edges.append((self.cone_of[inp], cone.root, inp, nid))
        return edges

    def _reshard_cost(self, src: DimStrategy, dst: DimStrategy,
                      nbytes: float) -> float:
        kind = reshard_collective(src, dst)
        return self.cm.collective(kind, nbytes, self.n)

    def _subgraph_candidates(self, cones: List[Cone], sub: List[int],
                             budget_s: float):
        """Solve the intra-subgraph ILP once per boundary-out strategy of
        the last cone; returns [(out_spec, in_demand, cost, specs)]."""
        out_cone = cones[-1]
        cands = []
        for si in range(len(out_cone.strategies)):
            if budget_s <= 0:
                return cands  # fall back to greedy upstream
            r = self._solve_ilp(cones, sub, pin=(len(cones) - 1, si),
                                budget_s=max(budget_s / max(
                                    len(out_cone.strategies), 1), 0.2))
            if r is not None:
                cost, choice = r
                specs: Dict[int, DimStrategy] = {}
                demand: Dict[int, DimStrategy] = {}
                for ci, cone in enumerate(cones):
                    specs.update(cone.strategies[choice[ci]])
                    for k, v in cone.strat_in[choice[ci]].items():
                        demand.setdefault(k, v)
                out_spec = out_cone.strategies[si][out_cone.root]
                cands.append((out_spec, demand, cost, specs))
        return cands

    def _solve_ilp(self, cones: List[Cone], sub: List[int],
                   pin: Tuple[int, int], budget_s: float):
        from scipy.optimize import Bounds, LinearConstraint, milp
        nc = len(cones)
        idx = {}
        costs = []
        for ci, cone in enumerate(cones):
            for si in range(len(cone.strategies)):
                idx[(ci, si)] = len(costs)
                costs.append(cone.costs[si])
        nx = len(costs)
        edges = self._inter_cone_edges(cones, sub)
        cone_index = {c.root: i for i, c in enumerate(cones)}
        # edge product variables
        evars = []
        ecosts = []
        for (src_root, dst_root, prod_nid, cons_nid) in edges:
            c1, c2 = cone_index[src_root], cone_index[dst_root]
            nbytes = self.g.bytes_of(self.g.nodes[prod_nid])
            for s1 in range(len(cones[c1].strategies)):
                sp1 = cones[c1].strategies[s1].get(
                    prod_nid, DimStrategy.replicated(self.n))
                for s2 in range(len(cones[c2].strategies)):
                    want = cones[c2].strat_in[s2].get(prod_nid, sp1)
                    rc = self._reshard_cost(sp1, want, nbytes)
                    if rc > 0:
                        evars.append((c1, s1, c2, s2))
                        ecosts.append(rc)
        ne = len(evars)
        cvec = np.array(costs + ecosts)
        # constraints
        rows_A, rows_lb, rows_ub = [], [], []
        for ci, cone in enumerate(cones):
            row = np.zeros(nx + ne)
            for si in range(len(cone.strategies)):
                row[idx[(ci, si)]] = 1.0
            rows_A.append(row)
            rows_lb.append(1.0)
            rows_ub.append(1.0)
        for ei, (c1, s1, c2, s2) in enumerate(evars):
            row = np.zeros(nx + ne)
            row[idx[(c1, s1)]] = 1.0
            row[idx[(c2, s2)]] = 1.0
            row[nx + ei] = -1.0
            rows_A.append(row)       # x1 + x2 - y <= 1
            rows_lb.append(-np.inf)
            rows_ub.append(1.0)
        lb = np.zeros(nx + ne)
        ub = np.ones(nx + ne)
        pi, psi = pin
        for si in range(len(cones[pi].strategies)):
            if si != psi:
                ub[idx[(pi, si)]] = 0.0
        lb[idx[(pi, psi)]] = 1.0
        integrality = np.concatenate([np.ones(nx), np.zeros(ne)])
        try:
            res = milp(c=cvec,
                       constraints=LinearConstraint(
                           np.array(rows_A), np.array(rows_lb),
                           np.array(rows_ub)),
                       bounds=Bounds(lb, ub), integrality=integrality,
                       options={"time_limit": max(budget_s, 0.1)})
        except Exception:
            return None
        if not res.success:
            return None
        choice = {}
        for (ci, si), j in idx.items():
            if res.x[j] > 0.5:
                choice[ci] = si
        if len(choice) != nc:
            return None
        return float(res.fun), choice

    def _greedy(self, cones: List[Cone]):
        """Forward greedy: each cone picks its locally cheapest strategy
        (the reference's InferByRank-class fallback)."""
        specs: Dict[int, DimStrategy] = {}
        demand: Dict[int, DimStrategy] = {}
        cost = 0.0
        for cone in cones:
            best = min(range(len(cone.strategies)), key=lambda s: cone.costs[s])
            specs.update(cone.strategies[best])
            for k, v in cone.strat_in[best].items():
                demand.setdefault(k, v)
            cost += cone.costs[best]
        out_cone = cones[-1]
        out_spec = specs.get(out_cone.root, DimStrategy.replicated(self.n))
        return (out_spec, demand, cost, specs)
