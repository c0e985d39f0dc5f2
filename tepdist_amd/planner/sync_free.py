"""Micro-batch ("SyncFree") gradient-accumulation analysis.

Re-implements the intent of the reference's SyncFreeSplittingAnalysis +
SyncFreeDecomposition (sync_free_splitting_analysis.cc:229,
sync_free_decomposition.cc:457): find the sample-input batch dimension,
propose micro-batch counts, and validate that splitting the step into
micro-batches is "sync-free" — i.e. parameter gradients combine purely
additively across micro-batches, so the only cross-micro work is
elementwise accumulation (GA) and the optimizer (AG) runs once.

For our op set additivity holds whenever every path from a sample input to
the loss treats the batch dim element-wise or reduces it only at the final
mean loss (true for GPT-2/MLP/MoE/WRN graphs: batched matmuls, row-wise
norms/softmax, per-row CE). The validator walks the graph to confirm no op
mixes information ACROSS the batch dim before the loss."""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

from tepdist_amd.ir.graph import Graph, Node

# ops that mix rows of their input's dim0 (would break sync-freeness if the
# batch dim reaches them as a non-batch axis)
_BATCH_MIXING = {"linear": [1], "matmul": []}


@dataclass
class SyncFreeProposal:
    batch_input: int                # node id of the sample input
    batch_dim: int                  # its batch dimension (0)
    micro_batches: List[int]        # valid micro-batch counts
    activation_bytes_full: float    # activation memory at micro=1


class SyncFreeSplittingAnalysis:
    def __init__(self, graph: Graph):
        self.g = graph

    def run(self) -> Optional[SyncFreeProposal]:
        if not self.g.inputs:
            return None
        inp = self.g.nodes[self.g.inputs[0]]
        if not inp.shape:
            return None
        b = inp.shape[0]
        if b < 1:
            return None
        if not self._validate(inp):
            return None
        micros = [m for m in (1, 2, 4, 8, 16, 32) if b % m == 0 and b // m >= 1]
        act = sum(self.g.bytes_of(n) for n in self.g.topo()
                  if n.op not in ("param",))
        return SyncFreeProposal(inp.id, 0, micros, act)

    def _validate(self, inp: Node) -> bool:
        """Track which dim of each tensor carries the batch index; fail if
        an op contracts over it anywhere but the loss."""
        bdim = {inp.id: 0}
        for n in self.g.topo():
            if n.id == inp.id or not n.inputs:
                continue
            dims = [bdim.get(i) for i in n.inputs]
            if all(d is None for d in dims):
                continue
            if n.op == "cross_entropy":
                continue  # final mean over rows: handled by loss rescale
            if n.op in ("linear", "matmul"):
                # contraction over the last dim of input0 / dim -2 of input1
                d0 = dims[0]
                a = self.g.nodes[n.inputs[0]]
                if d0 is not None and d0 == len(a.shape) - 1:
                    return False
                bdim[n.id] = d0 if d0 is not None else None
            elif n.op in ("layernorm", "softmax"):
                d0 = dims[0]
                if d0 == len(self.g.nodes[n.inputs[0]].shape) - 1:
                    return False
                bdim[n.id] = d0
            elif n.op == "reshape":
                dm = n.attrs.get("dim_map", {})
                d0 = dims[0]
                mapped = dm.get(d0, dm.get(str(d0))) if d0 is not None else None
                bdim[n.id] = mapped
            elif n.op == "transpose":
                perm = n.attrs.get("perm")
                d0 = dims[0]
                if perm and d0 is not None and d0 in perm:
                    bdim[n.id] = perm.index(d0)
                else:
                    bdim[n.id] = d0
            else:
                ds = [d for d in dims if d is not None]
                bdim[n.id] = ds[0] if ds else None
        return True
