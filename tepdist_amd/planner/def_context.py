"""DefContext: the compile-time decomposition artifact.

Mirrors the reference's DefContext tree (hlo_module.h:63-275, SURVEY.md
§2.3): the entry program decomposes into child contexts
  ENTRY -> { CG (compute gradients, one instance per micro-batch),
             GA_INIT (zero accumulators), GA (accumulate), AG (apply
             gradients / optimizer) }
and each of CG/GA/GA_INIT/AG further splits into per-pipeline-stage
*_SLICE contexts. Each context records its instance slice ids (micro
ordinals x spmd shards) and input/output maps; the runtime builds its
TaskDAG from this tree (runtime/task_graph.py)."""

from __future__ import annotations

import json
from dataclasses import dataclass, field
from typing import Dict, List, Optional

ENTRY = "ENTRY"
CG = "CG"
GA_INIT = "GA_INIT"
GA = "GA"
AG = "AG"
CG_SLICE = "CG_SLICE"
AG_SLICE = "AG_SLICE"


@dataclass
class DefContext:
    name: str
    kind: str
    def_id: int
    parent: Optional[int] = None
    children: List[int] = field(default_factory=list)
    stage: int = -1                     # for *_SLICE contexts
    micro: int = -1                     # micro-batch ordinal for CG slices
    node_ids: List[int] = field(default_factory=list)  # IR nodes it runs
    # arg index -> producing (def_id, output index) across the tree
    input_def_map: Dict[int, List[int]] = field(default_factory=dict)
    gflops: float = 0.0


@dataclass
class DefContextTree:
    contexts: Dict[int, DefContext] = field(default_factory=dict)
    _next: int = 0

    def new(self, name: str, kind: str, parent: Optional[int] = None,
            **kw) -> DefContext:
        ctx = DefContext(name, kind, self._next, parent, **kw)
        self.contexts[ctx.def_id] = ctx
        if parent is not None:
            self.contexts[parent].children.append(ctx.def_id)
        self._next += 1
        return ctx

    def entry(self) -> DefContext:
        return self.contexts[0]

    def to_json(self) -> str:
        return json.dumps({
            str(i): {
                "name": c.name, "kind": c.kind, "parent": c.parent,
                "children": c.children, "stage": c.stage, "micro": c.micro,
                "node_ids": c.node_ids, "input_def_map": c.input_def_map,
                "gflops": c.gflops,
            } for i, c in self.contexts.items()
        })

    @staticmethod
    def from_json(s: str) -> "DefContextTree":
        d = json.loads(s)
        t = DefContextTree()
        for i, cd in d.items():
            c = DefContext(cd["name"], cd["kind"], int(i), cd["parent"],
                           cd["children"], cd["stage"], cd["micro"],
                           cd["node_ids"],
                           {int(k): v for k, v in cd["input_def_map"].items()},
                           cd["gflops"])
            t.contexts[int(i)] = c
            t._next = max(t._next, int(i) + 1)
        return t


def build_def_tree(graph, num_stages: int, num_micro: int,
                   node_stage: Dict[int, int]) -> DefContextTree:
    """Builds the ENTRY -> {CG x micro x stage, GA_INIT, GA, AG} tree from
    a planned graph (the reference's SyncFreeDecomposition +
    StageDecomposition combined)."""
    t = DefContextTree()
    entry = t.new("entry", ENTRY)
    per_stage_nodes: Dict[int, List[int]] = {s: [] for s in range(num_stages)}
    for nid, st in node_stage.items():
        per_stage_nodes.setdefault(st, []).append(nid)

    cg = t.new("cg", CG, entry.def_id)
    for m in range(num_micro):
        for s in range(num_stages):
            nodes = per_stage_nodes.get(s, [])
            gf = sum(graph.flops(graph.nodes[n]) for n in nodes) / 1e9
            t.new(f"cg_m{m}_s{s}", CG_SLICE, cg.def_id, stage=s, micro=m,
                  node_ids=nodes, gflops=gf)
    t.new("ga_init", GA_INIT, entry.def_id)
    t.new("ga", GA, entry.def_id)
    ag = t.new("ag", AG, entry.def_id)
    for s in range(num_stages):
        t.new(f"ag_s{s}", AG_SLICE, ag.def_id, stage=s)
    return t
