"""SpmdTransform: applies a planned sharding to the IR, producing the
per-rank executable graph with reshard collectives inserted.

The reference rewrites every HLO instruction to its sharded shape and
inserts kCustomCollective placeholders where producer/consumer strategies
mismatch (SpmdTransform, service/parallel/spmd_transform.cc:1840,2155),
then lowers them to concrete DAPPLE collectives
(custom_collective_expander.cc:95-175). Here the same two steps run over
our Graph: shapes are rewritten to their local shard, and edges get
explicit collective nodes ("all_reduce", "all_gather", "all_to_all",
"dynamic_slice", "copy_to") that the interpreter executes through the
autograd-aware RCCL mappings (parallel/mappings.py), so the backward pass
of the transformed graph is automatically correct (the reference transforms
the backward HLO explicitly; we let autograd differentiate the collective).

The transform is rank-agnostic: one graph serves every rank of the mesh
dim; rank-dependent ops (dynamic_slice) read the rank from the process
group at execution time.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, Tuple

from tepdist_amd.ir.graph import Graph, Node
from tepdist_amd.planner.dist_spec import DimStrategy
from tepdist_amd.planner.rules import back_infer


class TransformError(RuntimeError):
    pass


@dataclass
class TransformResult:
    graph: Graph
    # param name -> (partition_dim, num_shards); dim = -1 means replicated
    param_specs: Dict[str, Tuple[int, int]] = field(default_factory=dict)
    # data input name -> (partition_dim, num_shards) of the slice each rank
    # takes (the graph itself contains the dynamic_slice node; this is for
    # callers that want to pre-shard feeds instead)
    input_specs: Dict[str, Tuple[int, int]] = field(default_factory=dict)


def _local_shape(shape, spec: DimStrategy):
    if not spec.is_split:
        return tuple(shape)
    sh = list(shape)
    d = spec.partition_dim
    if sh[d] % spec.num_shards != 0:
        raise TransformError(f"dim {d} of {shape} not divisible by "
                             f"{spec.num_shards}")
    sh[d] //= spec.num_shards
    return tuple(sh)


class SpmdTransform:
    def __init__(self, graph: Graph, node_specs: Dict[int, DimStrategy],
                 nshards: int):
        self.src = graph
        self.specs = node_specs
        self.n = nshards

    def run(self) -> TransformResult:
        if self.n <= 1:
            return TransformResult(self.src)
        g = self.src
        out = Graph()
        res = TransformResult(out)
        new_of: Dict[int, Node] = {}      # src id -> new node
        cur_spec: Dict[int, DimStrategy] = {}  # src id -> spec of new node

        for node in g.topo():
            spec = self.specs.get(node.id, DimStrategy.replicated(self.n))
            if spec.is_glue:
                spec = DimStrategy.replicated(self.n)

            if node.op == "param":
                nn = out.add_param(node.name, _local_shape(node.shape, spec),
                                   node.dtype, node.op_group)
                nn.attrs = dict(node.attrs)
                nn.attrs["global_shape"] = list(node.shape)
                self._scale_attrs(nn, spec)
                res.param_specs[node.name] = (
                    spec.partition_dim if spec.is_split else -1,
                    spec.num_shards if spec.is_split else 1)
                new_of[node.id], cur_spec[node.id] = nn, spec
                continue
            if node.op == "data":
                # data arrives global; a dynamic_slice takes the rank's part
                nn = out.add_input(node.name, node.shape, node.dtype)
                nn.attrs = dict(node.attrs)
                if spec.is_split:
                    sl = out.add("dynamic_slice", [nn],
                                 _local_shape(node.shape, spec), node.dtype,
                                 {"dim": spec.partition_dim, "n": self.n})
                    sl.attrs.update({k: v for k, v in node.attrs.items()
                                     if k not in ("dim", "n")})
                    self._scale_attrs(sl, spec)
                    res.input_specs[node.name] = (spec.partition_dim, self.n)
                    new_of[node.id], cur_spec[node.id] = sl, spec
                else:
                    new_of[node.id], cur_spec[node.id] = nn, spec
                continue

            st = back_infer(g, node, spec, self.n)
            if st is None:
                raise TransformError(
                    f"no strategy for {node.op}#{node.id} -> {spec}")
            wanted = list(st.ins) if st.ins else \
                [DimStrategy.replicated(self.n)] * len(node.inputs)

            new_ins = []
            consumer_sharded = spec.is_split or spec.is_partial
            for src_id, want in zip(node.inputs, wanted):
                x = new_of[src_id]
                have = cur_spec[src_id]
                x = self._reshard(out, x, have, want, g.nodes[src_id])
                if (consumer_sharded and want.is_replicated
                        and not have.is_split and not have.is_partial
                        and self._needs_grad(src_id)):
                    # replicated tensor consumed inside a sharded region:
                    # identity forward, all-reduce backward (Megatron f)
                    x = out.add("copy_to", [x], x.shape, x.dtype)
                new_ins.append(x)

            nn = out.add(node.op, new_ins, _local_shape(node.shape, spec),
                         node.dtype, dict(node.attrs), node.op_group,
                         node.name)
            nn.is_backward = node.is_backward
            nn.attrs["global_shape"] = list(node.shape)
            self._scale_attrs(nn, spec)
            new_of[node.id], cur_spec[node.id] = nn, spec

        # outputs: resolve each to replicated so every rank returns the
        # full value (losses: partial -> all-reduce mean over shards)
        for o in g.outputs:
            x, have = new_of[o], cur_spec[o]
            if have.is_partial:
                x = out.add("all_reduce", [x], x.shape, x.dtype)
                if g.nodes[o].op == "cross_entropy" and \
                        self.specs[o].is_partial:
                    # row-sharded mean loss: average, not sum
                    x = out.add("scale", [x], x.shape, x.dtype,
                                {"scale": 1.0 / self.n})
            elif have.is_split:
                x = out.add("all_gather", [x], g.nodes[o].shape, x.dtype,
                            {"dim": have.partition_dim})
            out.outputs.append(x.id)
        return res

    # ------------------------------------------------------------------

    def _needs_grad(self, src_id: int) -> bool:
        """True if the source subtree contains a trainable param (only then
        does the backward-all-reduce wrapper matter)."""
        seen = set()
        stack = [src_id]
        while stack:
            i = stack.pop()
            if i in seen:
                continue
            seen.add(i)
            n = self.src.nodes[i]
            if n.op == "param":
                return True
            stack.extend(n.inputs)
        return False

    def _scale_attrs(self, nn: Node, spec: DimStrategy):
        """Keep flattened-batch and head attrs consistent with the local
        shard (dim-0 splits shrink `batch`; attention head splits shrink
        `heads`)."""
        if not spec.is_split:
            return
        if spec.partition_dim == 0 and "batch" in nn.attrs:
            nn.attrs["batch"] = max(1, nn.attrs["batch"] // spec.num_shards)
        if nn.op == "attention" and spec.partition_dim == 1 and \
                "heads" in nn.attrs:
            nn.attrs["heads"] = max(1, nn.attrs["heads"] // spec.num_shards)

    def _reshard(self, out: Graph, x: Node, have: DimStrategy,
                 want: DimStrategy, src_node: Node) -> Node:
        """Inserts the collective converting spec `have` to `want`
        (CustomCollectiveExpander's lowering table)."""
        if want.is_glue or have == want:
            return x
        if have.is_glue or have.is_replicated:
            if want.is_replicated or not want.is_split:
                return x
            sl = out.add("dynamic_slice", [x], _local_shape(
                x.shape, want), x.dtype,
                {"dim": want.partition_dim, "n": self.n})
            self._scale_attrs(sl, want)
            return sl
        if have.is_partial:
            ar = out.add("all_reduce", [x], x.shape, x.dtype)
            if want.is_split:
                sl = out.add("dynamic_slice", [ar], _local_shape(
                    x.shape, want), x.dtype,
                    {"dim": want.partition_dim, "n": self.n})
                self._scale_attrs(sl, want)
                return sl
            return ar
        if have.is_split:
            full_shape = list(x.shape)
            full_shape[have.partition_dim] *= have.num_shards
            if want.is_replicated:
                return out.add("all_gather", [x], tuple(full_shape), x.dtype,
                               {"dim": have.partition_dim})
            if want.is_split and want.partition_dim != have.partition_dim:
                sh = list(full_shape)
                sh[want.partition_dim] //= self.n
                a2a = out.add("all_to_all", [x], tuple(sh), x.dtype,
                              {"src_dim": have.partition_dim,
                               "dst_dim": want.partition_dim})
                self._scale_attrs(a2a, want)
                return a2a
            if want.is_split:
                return x
        raise TransformError(f"unsupported reshard {have} -> {want}")
