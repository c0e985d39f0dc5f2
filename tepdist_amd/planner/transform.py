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

Multi-round (hybrid) application — the reference applies one transform per
split ordinal and re-derives shapes between rounds
(spmd_transform.cc:2155, dist_spec.h:36-227): `multi_round_transform`
chains one SpmdTransform per mesh round over the previous round's output
graph, remapping the plan's original-node-id specs through each round's
id_map. Every collective node carries attrs["mesh_round"] so the executor
runs it on that round's process group (CommDevManager).

The transform is rank-agnostic: one graph serves every rank of the mesh
dim; rank-dependent ops (dynamic_slice) read the rank from the process
group at execution time.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence, Tuple

from tepdist_amd.ir.graph import Graph, Node
from tepdist_amd.planner.dist_spec import DimStrategy, DistSpec
from tepdist_amd.planner.rules import back_infer


class TransformError(RuntimeError):
    pass


@dataclass
class TransformResult:
    graph: Graph
    # param name -> (partition_dim, num_shards); dim = -1 means replicated.
    # For multi-round results this is the LAST splitting round's view; use
    # param_rounds for the full per-round stack.
    param_specs: Dict[str, Tuple[int, int]] = field(default_factory=dict)
    # param name -> [(round, partition_dim, num_shards), ...] splits applied
    # in round order (each narrows the previous round's local shard)
    param_rounds: Dict[str, List[Tuple[int, int, int]]] = \
        field(default_factory=dict)
    # data input name -> (partition_dim, num_shards) of the slice each rank
    # takes (the graph itself contains the dynamic_slice node; this is for
    # callers that want to pre-shard feeds instead)
    input_specs: Dict[str, Tuple[int, int]] = field(default_factory=dict)
    # src node id -> transformed node id (for chaining rounds)
    id_map: Dict[int, int] = field(default_factory=dict)
    # params whose gradients the EXECUTOR must all-reduce over the listed
    # rounds' groups (dp_reduce rounds skip the per-param copy_to wrapper so
    # a bucketed, backward-overlapped reducer can do it instead)
    grad_sync_params: Dict[str, List[int]] = field(default_factory=dict)


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


# shape-preserving ops whose this-round spec may be inherited from their
# input when the plan has no entry for them (nodes inserted by an earlier
# round of a multi-round transform)
_PASS_THROUGH = ("copy_to", "all_reduce", "scale", "gelu", "dropout",
                 "elementwise", "add", "mul")


class SpmdTransform:
    def __init__(self, graph: Graph, node_specs: Dict[int, DimStrategy],
                 nshards: int, round_ordinal: int = 0,
                 dp_reduce: bool = False, sharded_outputs=None):
        self.src = graph
        self.specs = node_specs
        self.n = nshards
        self.round = round_ordinal
        # graph-output node ids whose values STAY SHARDED (pipeline stage
        # boundaries: the next stage holds the same mesh coordinates and
        # consumes the local shard); everything else resolves to the full
        # value (losses)
        self.sharded_outputs = set(sharded_outputs or ())
        # dp_reduce: this round is a data-parallel round executed with a
        # bucketed gradient reducer — params consumed replicated inside
        # sharded regions are recorded in grad_sync_params instead of being
        # wrapped in copy_to (whose backward would all-reduce per param per
        # micro-batch)
        self.dp_reduce = dp_reduce

    def run(self) -> TransformResult:
        if self.n <= 1:
            return TransformResult(
                self.src, id_map={i: i for i in self.src.nodes})
        g = self.src
        out = Graph()
        res = TransformResult(out)
        new_of: Dict[int, Node] = {}      # src id -> new node
        cur_spec: Dict[int, DimStrategy] = {}  # src id -> spec of new node
        assigned: Dict[int, DimStrategy] = {}  # src id -> this-round spec
        self._idmap_override: Dict[int, int] = {}  # src id -> new id

        def _spec_of(node: Node) -> DimStrategy:
            spec = self.specs.get(node.id)
            if spec is not None and not spec.is_glue:
                return spec
            # no plan entry (inserted by an earlier round, or glue):
            # inherit the producer's spec when the op allows it
            if node.inputs:
                cand = assigned.get(node.inputs[0])
                if cand is not None and cand.is_split and \
                        back_infer(g, node, cand, self.n) is not None:
                    return cand
            return DimStrategy.replicated(self.n)

        for node in g.topo():
            spec = _spec_of(node)
            assigned[node.id] = spec

            if node.op == "param":
                nn = out.add_param(node.name, _local_shape(node.shape, spec),
                                   node.dtype, node.op_group)
                nn.attrs = dict(node.attrs)
                nn.attrs["global_shape"] = list(node.shape)
                self._scale_attrs(nn, spec)
                res.param_specs[node.name] = (
                    spec.partition_dim if spec.is_split else -1,
                    spec.num_shards if spec.is_split else 1)
                if spec.is_split:
                    res.param_rounds[node.name] = [
                        (self.round, spec.partition_dim, spec.num_shards)]
                new_of[node.id], cur_spec[node.id] = nn, spec
                continue
            if node.op == "data":
                if node.attrs.get("stage_boundary") and spec.is_split:
                    # cross-stage activation: the previous stage (same mesh
                    # coordinates) already holds the local shard — localize
                    # the shape, no slice
                    nn = out.add_input(node.name,
                                       _local_shape(node.shape, spec),
                                       node.dtype)
                    nn.attrs = dict(node.attrs)
                    self._scale_attrs(nn, spec)
                    new_of[node.id], cur_spec[node.id] = nn, spec
                    continue
                # data arrives global; a dynamic_slice takes the rank's part
                nn = out.add_input(node.name, node.shape, node.dtype)
                nn.attrs = dict(node.attrs)
                if spec.is_split:
                    sl = out.add("dynamic_slice", [nn],
                                 _local_shape(node.shape, spec), node.dtype,
                                 {"dim": spec.partition_dim, "n": self.n,
                                  "mesh_round": self.round})
                    sl.attrs.update({k: v for k, v in node.attrs.items()
                                     if k not in ("dim", "n", "mesh_round")})
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
            for k, (src_id, want) in enumerate(zip(node.inputs, wanted)):
                x = new_of[src_id]
                have = cur_spec[src_id]
                x = self._reshard(out, x, have, want, g.nodes[src_id])
                if (consumer_sharded and want.is_replicated
                        and not have.is_split and not have.is_partial
                        and self._needs_grad(src_id)):
                    # replicated tensor consumed inside a sharded region:
                    # its gradient must be summed over this round's group.
                    if self.dp_reduce and g.nodes[src_id].op == "param":
                        # bucketed reducer handles it (recorded below)
                        res.grad_sync_params.setdefault(
                            g.nodes[src_id].name, [self.round])
                    else:
                        # identity forward, all-reduce backward (Megatron f)
                        x = out.add("copy_to", [x], x.shape, x.dtype,
                                    {"mesh_round": self.round})
                if (node.op == "linear" and spec.is_partial and k == 2):
                    # K-split linear keeps its bias: every shard would add
                    # the full bias to its partial sums, so pre-scale it by
                    # 1/n (the post-all-reduce sum then adds it exactly once)
                    x = out.add("scale", [x], x.shape, x.dtype,
                                {"scale": 1.0 / self.n})
                new_ins.append(x)

            nn = out.add(node.op, new_ins, _local_shape(node.shape, spec),
                         node.dtype, dict(node.attrs), node.op_group,
                         node.name)
            nn.is_backward = node.is_backward
            nn.attrs["global_shape"] = list(node.shape)
            self._scale_attrs(nn, spec)
            if st.note == "vocab" and node.op in ("embedding",
                                                  "cross_entropy"):
                # vocab-split table/logits: the interpreter runs the
                # rank-aware (masked / distributed-lse) lowering
                nn.attrs["vocab_parallel"] = True
                nn.attrs["mesh_round"] = self.round
                if node.op == "cross_entropy":
                    # distributed CE already returns the GLOBAL loss —
                    # not a partial term: no output all-reduce needed
                    spec = DimStrategy.replicated(self.n)
                    assigned[node.id] = spec
            elif node.op == "cross_entropy" and spec.is_partial:
                # row-split CE yields the LOCAL MEAN; scale by 1/n so the
                # partials sum to the global mean (assumes balanced valid
                # counts per shard — exact for fully-valid batches) and any
                # downstream all_reduce (this round's or a later round's
                # reshard) is correct without output-special-casing.
                # id_map keeps the CE node itself so a LATER round's loss
                # spec lands on the CE, not on this helper scale.
                self._idmap_override[node.id] = nn.id
                nn = out.add("scale", [nn], nn.shape, nn.dtype,
                             {"scale": 1.0 / self.n})
            new_of[node.id], cur_spec[node.id] = nn, spec

        # outputs: resolve each to replicated so every rank returns the
        # full value (losses: partial -> all-reduce mean over shards)
        for o in g.outputs:
            x, have = new_of[o], cur_spec[o]
            if o in self.sharded_outputs:
                out.outputs.append(x.id)   # stage boundary: stays local
                continue
            if have.is_partial:
                # partials (incl. the 1/n-scaled row-split loss) sum to the
                # global value
                x = out.add("all_reduce", [x], x.shape, x.dtype,
                            {"mesh_round": self.round})
            elif have.is_split:
                x = out.add("all_gather", [x], g.nodes[o].shape, x.dtype,
                            {"dim": have.partition_dim,
                             "mesh_round": self.round})
            out.outputs.append(x.id)
        res.id_map = {sid: nn.id for sid, nn in new_of.items()}
        res.id_map.update(self._idmap_override)
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
                {"dim": want.partition_dim, "n": self.n,
                 "mesh_round": self.round})
            self._scale_attrs(sl, want)
            return sl
        if have.is_partial:
            ar = out.add("all_reduce", [x], x.shape, x.dtype,
                         {"mesh_round": self.round})
            if want.is_split:
                sl = out.add("dynamic_slice", [ar], _local_shape(
                    x.shape, want), x.dtype,
                    {"dim": want.partition_dim, "n": self.n,
                     "mesh_round": self.round})
                self._scale_attrs(sl, want)
                return sl
            return ar
        if have.is_split:
            full_shape = list(x.shape)
            full_shape[have.partition_dim] *= have.num_shards
            if want.is_replicated:
                return out.add("all_gather", [x], tuple(full_shape), x.dtype,
                               {"dim": have.partition_dim,
                                "mesh_round": self.round})
            if want.is_split and want.partition_dim != have.partition_dim:
                sh = list(full_shape)
                sh[want.partition_dim] //= self.n
                a2a = out.add("all_to_all", [x], tuple(sh), x.dtype,
                              {"src_dim": have.partition_dim,
                               "dst_dim": want.partition_dim,
                               "mesh_round": self.round})
                self._scale_attrs(a2a, want)
                return a2a
            if want.is_split:
                return x
        raise TransformError(f"unsupported reshard {have} -> {want}")


# --------------------------------------------------------------------------
# multi-round application
# --------------------------------------------------------------------------

def multi_round_transform(graph: Graph,
                          node_specs: Dict[int, DistSpec],
                          mesh: Sequence[int],
                          dp_rounds: Optional[Sequence[int]] = None,
                          sharded_outputs=None) -> TransformResult:
    """Applies one SpmdTransform per mesh round (reference: one DoTransform
    per split ordinal, spmd_transform.cc:2155), each over the previous
    round's output graph. `node_specs` is keyed by ORIGINAL graph node ids;
    round r's specs are remapped through the accumulated id_map. `mesh[r]`
    is round r's shard count; rounds in `dp_rounds` record their replicated
    params in grad_sync_params (executor uses a bucketed reducer) instead
    of inserting copy_to wrappers."""
    dp_rounds = set(dp_rounds or ())
    cur = graph
    idmap: Dict[int, int] = {i: i for i in graph.nodes}
    combined = TransformResult(cur)
    combined.id_map = dict(idmap)
    sharded_out = set(sharded_outputs or ())
    for r, n in enumerate(mesh):
        if n <= 1:
            continue
        specs_r: Dict[int, DimStrategy] = {}
        for oid, ds in node_specs.items():
            if oid not in idmap:
                continue
            s = ds.round(r) if isinstance(ds, DistSpec) else \
                (ds if r == 0 else DimStrategy.glue())
            specs_r[idmap[oid]] = s
        t = SpmdTransform(cur, specs_r, n, round_ordinal=r,
                          dp_reduce=(r in dp_rounds),
                          sharded_outputs={idmap[o] for o in sharded_out
                                           if o in idmap})
        res = t.run()
        # compose id maps and merge param/input specs
        idmap = {oid: res.id_map[mid] for oid, mid in idmap.items()
                 if mid in res.id_map}
        for name, (dim, nsh) in res.param_specs.items():
            if nsh > 1:
                combined.param_rounds.setdefault(name, []).append(
                    (r, dim, nsh))
                combined.param_specs[name] = (dim, nsh)
            else:
                combined.param_specs.setdefault(name, (dim, nsh))
        for name, rounds in res.grad_sync_params.items():
            combined.grad_sync_params.setdefault(name, []).extend(rounds)
        for name, (dim, nsh) in res.input_specs.items():
            combined.input_specs.setdefault(name, (dim, nsh))
        cur = res.graph
    _canonicalize_slice_chains(cur)
    combined.graph = cur
    combined.id_map = idmap
    return combined


def _canonicalize_slice_chains(g: Graph) -> None:
    """Orders nested same-dim dynamic_slice chains by mesh round (round 0
    outermost). Reshard insertion can nest the rounds' slices in either
    order depending on where in the chain the later round's slice landed;
    each order is a valid partition on its own, but two tensors meeting at
    a binary op (e.g. logits and labels at the loss) must map the SAME
    global rows to the same rank, so one convention is enforced
    everywhere."""
    changed = True
    while changed:
        changed = False
        for node in list(g.nodes.values()):
            if node.op != "dynamic_slice" or not node.inputs:
                continue
            parent = g.nodes[node.inputs[0]]
            if parent.op != "dynamic_slice":
                continue
            d = node.attrs.get("dim", 0)
            if parent.attrs.get("dim", 0) != d:
                continue  # different dims commute
            r_in = node.attrs.get("mesh_round", 0)
            r_out = parent.attrs.get("mesh_round", 0)
            if r_out <= r_in:
                continue
            # swap the two slices' (round, n) so the smaller round is outer
            for k in ("mesh_round", "n"):
                node.attrs[k], parent.attrs[k] = \
                    parent.attrs.get(k), node.attrs.get(k)
            gp = g.nodes[parent.inputs[0]]
            sh = list(parent.shape)
            sh[d] = gp.shape[d] // parent.attrs["n"]
            parent.shape = tuple(sh)
            for nd, pa in ((parent, gp), (node, parent)):
                if "batch" in nd.attrs and d == 0 and "batch" in pa.attrs:
                    nd.attrs["batch"] = max(
                        1, pa.attrs["batch"] // nd.attrs["n"])
            changed = True
