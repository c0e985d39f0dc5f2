"""Generic stage decomposition: split a planned IR graph into per-stage
subgraphs with inferred cross-stage I/O maps.

Re-implements the reference's StageDecomposition
(service/parallel/stage_decomposition.cc:718): the
planner's per-node stage assignment (GraphSketch::StagePlan) becomes one
executable subgraph per pipeline stage, with the tensors crossing each
stage boundary identified so the runtime can insert Send/Recv tasks for
them. Round-1 had only a GPT-2-specific layer-range split
(models/gpt2.py); this works on ANY planned graph — llama and the fx
captures get pipeline decomposition for free, and the per-stage graphs
compose with the multi-round SpmdTransform (stage-decompose first, then
shard each stage over the plan's mesh rounds).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Tuple

from tepdist_amd.ir.graph import Graph


@dataclass
class StageBoundary:
    """One tensor crossing from `src_stage` to `dst_stage`: produced by
    src graph node `src_node` (original id), consumed in the destination
    stage through input placeholder `input_name`."""
    src_stage: int
    dst_stage: int
    src_node: int
    input_name: str
    shape: Tuple[int, ...]
    dtype: str
    src_output_idx: int = -1   # index into the source stage graph outputs


@dataclass
class StagePlan:
    stages: List[Graph]
    boundaries: List[StageBoundary] = field(default_factory=list)
    # per stage: original graph node id -> local node id (for remapping
    # the plan's node specs onto a stage subgraph before SpmdTransform)
    local_maps: List[Dict[int, int]] = field(default_factory=list)

    def inputs_of(self, stage: int) -> List[StageBoundary]:
        return [b for b in self.boundaries if b.dst_stage == stage]

    def outputs_of(self, stage: int) -> List[StageBoundary]:
        return [b for b in self.boundaries if b.src_stage == stage]


class DecompositionError(RuntimeError):
    pass


def decompose_stages(g: Graph, node_stage: Dict[int, int],
                     num_stages: int) -> StagePlan:
    """Splits `g` into `num_stages` subgraphs along the planner's per-node
    stage map. Params and data inputs replicate into every stage that
    consumes them (data is fed per stage; variables live on their stage).
    Producer stage must be <= consumer stage (the stage ILP's topology
    constraint); an edge crossing more than one stage is relayed through
    the intermediate stages."""
    if num_stages <= 1:
        return StagePlan([g])

    def stage_of(nid: int) -> int:
        n = g.nodes[nid]
        if n.op in ("param", "data"):
            # producers follow their first consumer
            cons = [c for c in g.nodes.values() if nid in c.inputs]
            if cons:
                return min(stage_of_cache.get(c.id, node_stage.get(c.id, 0))
                           for c in cons)
            return 0
        return node_stage.get(nid, 0)

    stage_of_cache: Dict[int, int] = {}
    for n in g.topo():
        stage_of_cache[n.id] = stage_of(n.id)

    stages = [Graph() for _ in range(num_stages)]
    # per stage: src node id -> node in that stage's graph
    local: List[Dict[int, object]] = [dict() for _ in range(num_stages)]
    boundaries: List[StageBoundary] = []
    # (src node id, stage) -> local node holding its value in that stage
    arrived: Dict[Tuple[int, int], object] = {}

    def materialize_input(nid: int, s: int):
        """Makes src node nid's value available in stage s, relaying
        through intermediate stages when the producer is further back."""
        if (nid, s) in arrived:
            return arrived[(nid, s)]
        src = g.nodes[nid]
        ps = stage_of_cache[nid]
        if ps > s:
            raise DecompositionError(
                f"edge from stage {ps} to earlier stage {s} "
                f"({src.op}#{nid})")
        if src.op in ("param", "data"):
            # replicate the producer into this stage
            sg = stages[s]
            if src.op == "param":
                nn = sg.add_param(src.name, src.shape, src.dtype,
                                  src.op_group)
            else:
                nn = sg.add_input(src.name, src.shape, src.dtype)
            nn.attrs = dict(src.attrs)
            arrived[(nid, s)] = nn
            return nn
        # relay hop-by-hop so every boundary is between adjacent stages
        # (the runtime's Send/Recv tasks connect neighbors)
        if ps < s - 1:
            materialize_input(nid, s - 1)
        name = f"stage_in_{nid}_{s}"
        nn = stages[s].add_input(name, src.shape, src.dtype)
        nn.attrs = dict(src.attrs)
        # boundary activations arrive ALREADY sharded from the previous
        # stage's identical mesh coordinates: SpmdTransform must localize
        # the shape without inserting a dynamic_slice
        nn.attrs["stage_boundary"] = True
        boundaries.append(StageBoundary(s - 1, s, nid, name,
                                        tuple(src.shape), src.dtype))
        arrived[(nid, s)] = nn
        return nn

    for n in g.topo():
        s = stage_of_cache[n.id]
        if n.op in ("param", "data"):
            continue  # materialized on demand by consumers
        ins = []
        for i in n.inputs:
            pi = stage_of_cache[i]
            if pi == s or g.nodes[i].op in ("param", "data"):
                x = arrived.get((i, s))
                if x is None:
                    if g.nodes[i].op in ("param", "data"):
                        x = materialize_input(i, s)
                    else:
                        x = local[s][i]
            else:
                x = materialize_input(i, s)
            ins.append(x)
        sg = stages[s]
        nn = sg.add(n.op, ins, n.shape, n.dtype, dict(n.attrs),
                    n.op_group, n.name)
        nn.is_backward = n.is_backward
        local[s][n.id] = nn
        arrived[(n.id, s)] = nn

    # graph outputs live on their producing stage
    for o in g.outputs:
        s = stage_of_cache[o]
        stages[s].outputs.append(local[s][o].id)
    # every boundary's source value becomes an output of its source stage
    # graph, so the runtime can fetch it for the Send task
    for b in boundaries:
        src_local = arrived[(b.src_node, b.src_stage)]
        sg = stages[b.src_stage]
        if src_local.id not in sg.outputs:
            sg.outputs.append(src_local.id)
        b.src_output_idx = sg.outputs.index(src_local.id)
    lmaps = []
    for s in range(num_stages):
        lm = {oid: nn.id for (oid, st), nn in arrived.items() if st == s}
        lm.update({oid: nn.id for oid, nn in local[s].items()})
        lmaps.append(lm)
    return StagePlan(stages, boundaries, lmaps)
