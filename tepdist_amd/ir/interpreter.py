"""IR interpreter: executes a client-shipped Graph on the server.

Plays the role of the reference's compiled sub-module executables (the
server compiles the received HLO and runs it) — here the
graph runs through the tepdist_amd.ops layer (hand-written CDNA4 kernels on
GPU, torch reference on CPU) with torch autograd providing the backward
pass and the fused AdamW applying gradients (the AG role)."""

from __future__ import annotations

import os
from typing import Dict

import torch

from tepdist_amd import ops
from tepdist_amd.ir.graph import Graph, Node
from tepdist_amd.parallel import mappings


def _fuse_mlp(g: Graph) -> None:
    """Backend peephole (GPU): linear(act=gelu) -> linear, single
    consumer, both biased, becomes ONE `mlp` node so the gelu rides the
    GEMM epilogues (ops.mlp: hipBLASLt GELU_AUX_BIAS / DGELU_BGRAD). Runs
    AFTER planning/transform — a mesh round that resharded between the
    two linears leaves a collective in between and the pattern simply
    does not match."""
    cons: Dict[int, int] = {}
    for n in g.nodes.values():
        for i in n.inputs:
            cons[i] = cons.get(i, 0) + 1
    outs = set(g.outputs)
    for n in list(g.nodes.values()):
        if (n.op != "linear" or n.attrs.get("act", "none") != "none"
                or len(n.inputs) != 3):
            continue
        p = g.nodes.get(n.inputs[0])
        if (p is None or p.op != "linear"
                or p.attrs.get("act") != "gelu" or len(p.inputs) != 3
                or cons.get(p.id, 0) != 1 or p.id in outs):
            continue
        n.op = "mlp"
        n.inputs = [p.inputs[0], p.inputs[1], p.inputs[2], n.inputs[1],
                    n.inputs[2]]
        n.attrs.pop("act", None)
        del g.nodes[p.id]


class GraphInterpreter:
    def __init__(self, graph: Graph, device: str = "cpu",
                 dtype=torch.float32, group=None, groups=None):
        self.g = graph
        if str(device).startswith("cuda") and \
                os.environ.get("TEPDIST_MLP_FUSE", "1") != "0":
            _fuse_mlp(self.g)
        self.device = device
        self.dtype = dtype
        # residual-add + layernorm pairs fused at eval time (GPU): the add
        # returns the sum as usual AND stashes the fused ln output for the
        # consumer node (one kernel, one HBM pass)
        self._addln = {}
        self._ln_cache = {}
        if str(device).startswith("cuda") and \
                os.environ.get("TEPDIST_ADDLN_FUSE", "1") != "0":
            for n in self.g.nodes.values():
                if n.op != "layernorm" or len(n.inputs) != 3:
                    continue
                a = self.g.nodes.get(n.inputs[0])
                g1 = self.g.nodes.get(n.inputs[1])
                b1 = self.g.nodes.get(n.inputs[2])
                if (a is not None and a.op == "add" and len(a.inputs) == 2
                        and g1 is not None and g1.op == "param"
                        and b1 is not None and b1.op == "param"
                        and a.id not in self._addln):
                    self._addln[a.id] = n.id
        self.group = group  # process group for reshard collective nodes
        # mesh-round ordinal -> process group (multi-round transforms tag
        # every collective with attrs["mesh_round"]; CommDevManager builds
        # the groups). `group` remains the single-round fallback.
        self.groups = groups or {}

    def _grp(self, n: Node):
        r = n.attrs.get("mesh_round")
        if r is not None and r in self.groups:
            return self.groups[r]
        return self.group

    def run(self, feeds: Dict[str, torch.Tensor],
            variables: Dict[str, torch.Tensor]) -> Dict[int, torch.Tensor]:
        """One forward pass. feeds keyed by input node name; variables by
        param name (tensors with requires_grad for training). Returns
        {output node id: tensor}."""
        env: Dict[int, torch.Tensor] = {}
        for n in self.g.topo():
            env[n.id] = self._eval(n, env, feeds, variables)
        return {o: env[o] for o in self.g.outputs}

    def _eval(self, n: Node, env, feeds, variables) -> torch.Tensor:
        ins = [env[i] for i in n.inputs]
        if n.op == "data":
            if n.name in feeds:
                return feeds[n.name].to(self.device)
            if n.name == "pos":  # position ids (B*S,) derived from seq attr
                bs = n.shape[0]
                b = n.attrs.get("batch", 1)
                s = bs // b
                return torch.arange(s, device=self.device).repeat(b)
            raise KeyError(f"missing feed {n.name}")
        if n.op == "param":
            return variables[n.name]
        if n.op == "embedding":
            if n.attrs.get("vocab_parallel"):
                # vocab-split table: mask out-of-shard ids; partial sums are
                # combined by the planner-inserted all_reduce downstream
                import torch.distributed as dist
                grp = self._grp(n)
                rank = dist.get_rank(grp) if dist.is_initialized() else 0
                vloc = ins[1].shape[0]
                local = ins[0] - rank * vloc
                in_shard = (local >= 0) & (local < vloc)
                local = local.clamp(0, vloc - 1)
                y = ops.embedding(local, ins[1])
                return y * in_shard.unsqueeze(-1).to(y.dtype)
            return ops.embedding(ins[0], ins[1])
        if n.op == "linear":
            bias = ins[2] if len(ins) > 2 else None
            return ops.linear(ins[0], ins[1], bias,
                              act=n.attrs.get("act", "none"))
        if n.op == "matmul":
            return ops.matmul(ins[0], ins[1])
        if n.op == "mlp":
            return ops.mlp(ins[0], ins[1], ins[2], ins[3], ins[4])
        if n.op == "layernorm":
            if n.id in self._ln_cache:
                return self._ln_cache.pop(n.id)
            return ops.layernorm(ins[0], ins[1], ins[2])
        if n.op == "softmax":
            return ops.softmax(ins[0], scale=n.attrs.get("scale", 1.0),
                               causal=n.attrs.get("causal", False))
        if n.op == "attention":
            x = ins  # q, k, v in flattened (B*S, d) form
            b = n.attrs["batch"]
            h = n.attrs["heads"]
            s = n.attrs["seq"]
            hd = n.shape[1] // h
            q, k, v = (t.reshape(b, s, h, hd).transpose(1, 2).contiguous()
                       for t in x)
            o = ops.attention(q, k, v, causal=True)
            return o.transpose(1, 2).reshape(b * s, h * hd).contiguous()
        if n.op == "attention_qkv":
            b = n.attrs["batch"]
            s = n.attrs["seq"]
            qkv = ins[0].reshape(b, s, -1)
            o = ops.attention_qkv(qkv, n.attrs["heads"], causal=True)
            return o.reshape(b * s, -1)
        if n.op == "moe_dispatch":
            # GShard static dispatch -> [E, C_local, d] (C_local from the
            # node's possibly-sharded shape: each token shard fills its
            # own capacity block — what makes EP a planner reshard)
            from tepdist_amd.models.moe import static_dispatch
            gates = torch.softmax(ins[1].float(), -1)
            D, _ = static_dispatch(ins[0], gates, n.attrs["k"], n.shape[1])
            return D
        if n.op == "moe_combine":
            from tepdist_amd.models.moe import static_combine
            gates = torch.softmax(ins[2].float(), -1)
            return static_combine(ins[0], ins[1], gates, n.attrs["k"])
        if n.op == "split":
            dim = n.attrs.get("dim", -1)
            idx = n.attrs.get("index", 0)
            size = n.shape[dim if dim >= 0 else len(n.shape) - 1]
            return ins[0].narrow(dim, idx * size, size)
        if n.op == "add":
            ln_id = self._addln.get(n.id)
            if (ln_id is not None and ins[0].is_cuda
                    and ins[0].dtype == torch.bfloat16
                    and ins[0].shape == ins[1].shape):
                ln = self.g.nodes[ln_id]
                gamma = variables[self.g.nodes[ln.inputs[1]].name]
                beta = variables[self.g.nodes[ln.inputs[2]].name]
                s, y = ops.add_layernorm(ins[0], ins[1], gamma, beta)
                self._ln_cache[ln_id] = y
                return s
            return ins[0] + ins[1]
        if n.op == "mul":
            return ins[0] * ins[1]
        if n.op == "gelu":
            return ops.gelu(ins[0])
        if n.op == "dropout":
            return ops.dropout(ins[0], n.attrs.get("p", 0.0))
        if n.op == "cross_entropy":
            if n.attrs.get("vocab_parallel"):
                # vocab-split logits: exact distributed CE (per-shard lse
                # combined over the group) — a plain CE over a vocab slice
                # would NOT be a partial term of the full loss
                import torch.distributed as dist
                from tepdist_amd.parallel.tp import \
                    vocab_parallel_cross_entropy
                grp = self._grp(n)
                rank = dist.get_rank(grp) if dist.is_initialized() else 0
                vloc = ins[0].shape[-1]
                return vocab_parallel_cross_entropy(
                    ins[0].reshape(-1, vloc), ins[1].reshape(-1),
                    rank * vloc, vloc, grp, ignore_index=-1)
            return ops.cross_entropy(ins[0], ins[1].reshape(-1),
                                     ignore_index=-1)
        if n.op == "reshape":
            return ins[0].reshape(n.shape)
        if n.op == "transpose":
            return ins[0].permute(n.attrs["perm"]).contiguous()
        if n.op == "elementwise":
            return ins[0]
        if n.op == "scale":
            return ins[0] * n.attrs.get("scale", 1.0)
        if n.op == "rmsnorm":
            return ops.rmsnorm(ins[0], ins[1])
        if n.op == "rope":
            heads = n.attrs["heads"]
            T, d = ins[0].shape
            hd = d // heads
            y = ops.rope(ins[0].reshape(T, heads, hd), n.attrs["seq"])
            return y.reshape(T, d)
        if n.op == "swiglu":
            return ops.swiglu(ins[0], ins[1])
        # reshard collectives inserted by the SpmdTransform (autograd-aware:
        # backward of the transformed graph is correct by construction)
        if n.op == "all_reduce":
            if len(ins) > 1:  # combined bundle (planner/combiner.py): one
                # flat collective covers all members; bundle_get projects
                flat = torch.cat([t.reshape(-1) for t in ins])
                return mappings.reduce_from_group(flat, self._grp(n))
            return mappings.reduce_from_group(ins[0], self._grp(n))
        if n.op == "bundle_get":
            offs = n.attrs["offsets"]
            idx = n.attrs["index"]
            start = int(sum(offs[:idx]))
            return ins[0].narrow(0, start, int(offs[idx])).reshape(n.shape)
        if n.op == "copy_to":
            return mappings.copy_to_group(ins[0], self._grp(n))
        if n.op == "all_gather":
            return mappings.gather_from_group(ins[0], self._grp(n),
                                              dim=n.attrs.get("dim", 0))
        if n.op == "dynamic_slice":
            return mappings.scatter_to_group(ins[0], self._grp(n),
                                             dim=n.attrs.get("dim", 0))
        if n.op == "all_to_all":
            # split-dim -> split-dim reshard; composed gather+slice keeps
            # autograd exact (a fused all_to_all_single path would halve the
            # bytes; the MoE layer uses that form)
            x = mappings.gather_from_group(ins[0], self._grp(n),
                                           dim=n.attrs["src_dim"])
            return mappings.scatter_to_group(x, self._grp(n),
                                             dim=n.attrs["dst_dim"])
        raise NotImplementedError(f"op {n.op}")
