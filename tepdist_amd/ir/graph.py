"""Tensor-program IR: the whole-graph representation the planner consumes.

Plays the role HLO plays in the reference (input = whole
HLO module): an SSA op graph with shapes, parameter identity, and the
client-side metadata the reference pipes through OpMetadata (op_group layer
tags + backward flags, reference xla_data.proto:272-273). Captured from
PyTorch models (ir/capture.py) or built by a model's export_ir().
"""

from __future__ import annotations

import json
from dataclasses import dataclass, field
from typing import Dict, List, Tuple

# ops whose cost is matmul-like (the planner's "compute sensitive" set,
# reference cost_spmd_strategy.h:42-49 IsComputeSensitive = dot/conv)
COMPUTE_SENSITIVE = {"linear", "matmul", "attention", "attention_qkv", "conv2d"}

ELEMENTWISE = {"add", "mul", "gelu", "dropout", "scale", "cast", "bias_add"}


@dataclass
class Node:
    id: int
    op: str
    inputs: List[int]
    shape: Tuple[int, ...]
    dtype: str = "bf16"
    attrs: dict = field(default_factory=dict)
    op_group: int = -1        # client layer tag (dapple_scope equivalent)
    is_backward: bool = False
    name: str = ""

    @property
    def numel(self) -> int:
        n = 1
        for s in self.shape:
            n *= s
        return n


@dataclass
class Graph:
    nodes: Dict[int, Node] = field(default_factory=dict)
    params: Dict[str, int] = field(default_factory=dict)   # name -> node id
    inputs: List[int] = field(default_factory=list)        # sample inputs
    outputs: List[int] = field(default_factory=list)
    _next_id: int = 0

    # -- construction -------------------------------------------------------

    def add(self, op: str, inputs: List["Node"], shape, dtype="bf16",
            attrs=None, op_group=-1, name="") -> Node:
        n = Node(self._next_id, op, [i.id for i in inputs], tuple(shape),
                 dtype, attrs or {}, op_group, False, name)
        self.nodes[n.id] = n
        self._next_id += 1
        return n

    def add_param(self, name: str, shape, dtype="bf16", op_group=-1) -> Node:
        n = self.add("param", [], shape, dtype, op_group=op_group, name=name)
        self.params[name] = n.id
        return n

    def add_input(self, name: str, shape, dtype="bf16") -> Node:
        n = self.add("data", [], shape, dtype, name=name)
        self.inputs.append(n.id)
        return n

    # -- structure ----------------------------------------------------------

    def topo(self) -> List[Node]:
        """Topological order, smallest-id-first among ready nodes (ids are
        creation order, so graphs built front-to-back keep their layout;
        passes that append nodes mid-graph — liveness cloning — still
        serialize correctly)."""
        import heapq
        indeg = {i: 0 for i in self.nodes}
        cons: Dict[int, List[int]] = {i: [] for i in self.nodes}
        for n in self.nodes.values():
            for i in n.inputs:
                indeg[n.id] += 1
                cons[i].append(n.id)
        heap = [i for i, d in indeg.items() if d == 0]
        heapq.heapify(heap)
        out = []
        while heap:
            i = heapq.heappop(heap)
            out.append(self.nodes[i])
            for c in cons[i]:
                indeg[c] -= 1
                if indeg[c] == 0:
                    heapq.heappush(heap, c)
        assert len(out) == len(self.nodes), "cycle in graph"
        return out

    def consumers(self) -> Dict[int, List[int]]:
        cons: Dict[int, List[int]] = {i: [] for i in self.nodes}
        for n in self.topo():
            for i in n.inputs:
                cons[i].append(n.id)
        return cons

    def flops(self, n: Node) -> float:
        """Forward flops of a node (training ~3x for fw+bw handled by the
        cost model)."""
        if n.op in ("linear", "matmul"):
            k = n.attrs.get("k")
            if k is None:
                k = self.nodes[n.inputs[0]].shape[-1]
            return 2.0 * n.numel * k
        if n.op in ("attention", "attention_qkv"):
            if len(n.shape) == 4:
                b, h, s, d = n.shape
            else:  # flattened (B*S, hidden) form with attrs
                s = n.attrs.get("seq", 1)
                b = n.shape[0] // max(s, 1)
                h = n.attrs.get("heads", 1)
                d = n.shape[1] // max(h, 1)
            return 2.0 * b * h * s * s * d * 2
        if n.op == "conv2d":
            cin = self.nodes[n.inputs[0]].shape[1]
            kh, kw = n.attrs.get("kernel", (3, 3))
            return 2.0 * n.numel * cin * kh * kw
        return float(n.numel)

    def bytes_of(self, n: Node) -> float:
        esize = 2 if n.dtype == "bf16" else 4
        return float(n.numel) * esize

    # -- critical nodes (subgraph cut points) -------------------------------

    def critical_nodes(self) -> List[int]:
        """Nodes through which every live value passes in topological order
        (single-tensor cut points; the reference's GraphSketch
        FindCriticalInsts, hlo_graph_sketch.cc:1289-1331). A node is
        critical if, right after it executes, it is the ONLY live non-param
        value that later nodes still need."""
        cons = self.consumers()
        order = self.topo()
        last_use = {}
        for n in order:
            for i in n.inputs:
                last_use[i] = n.id
        live: set = set()
        crit = []
        skip = {"param", "data"}
        for n in order:
            for i in n.inputs:
                if last_use.get(i) == n.id and i in live:
                    live.discard(i)
            if n.op in skip:
                continue
            if cons[n.id]:
                live.add(n.id)
            live_vals = [v for v in live if self.nodes[v].op not in skip]
            if len(live_vals) == 1 and live_vals[0] == n.id:
                crit.append(n.id)
        return crit

    # -- serialization (the wire format sent client->server, playing
    #    HloModuleProto's role) -------------------------------------------

    def to_json(self) -> str:
        return json.dumps({
            "nodes": [
                {"id": n.id, "op": n.op, "inputs": n.inputs,
                 "shape": list(n.shape), "dtype": n.dtype, "attrs": n.attrs,
                 "op_group": n.op_group, "backward": n.is_backward,
                 "name": n.name}
                for n in self.topo()
            ],
            "params": self.params,
            "inputs": self.inputs,
            "outputs": self.outputs,
        })

    @staticmethod
    def from_json(s: str) -> "Graph":
        d = json.loads(s)
        g = Graph()
        for nd in d["nodes"]:
            n = Node(nd["id"], nd["op"], nd["inputs"], tuple(nd["shape"]),
                     nd["dtype"], nd["attrs"], nd["op_group"], nd["backward"],
                     nd["name"])
            g.nodes[n.id] = n
            g._next_id = max(g._next_id, n.id + 1)
        g.params = dict(d["params"])
        g.inputs = list(d["inputs"])
        g.outputs = list(d["outputs"])
        return g
