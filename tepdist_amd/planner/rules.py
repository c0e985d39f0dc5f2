"""Per-opcode sharding-strategy enumeration and inference rules.

Plays the role of the reference's StrategyUtil Infer*/BackInfer* tables and
split-proposal generators (service/parallel/utils.h:31-291,
GenSplitProposals/GenDotProposals). For each op we enumerate the consistent
(output, inputs) sharding combinations for one mesh-dim round of n shards;
the cost search (spmd.py) picks among them."""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

from tepdist_amd.ir.graph import Graph, Node
from tepdist_amd.planner.dist_spec import DimStrategy

R = DimStrategy.replicated
S = DimStrategy.split
P = DimStrategy.partial


@dataclass(frozen=True)
class OpStrategy:
    """One consistent sharding choice for an op: the output DimStrategy and
    one DimStrategy per input (aligned with node.inputs)."""
    out: DimStrategy
    ins: tuple
    note: str = ""


def _splittable(shape, dim, n) -> bool:
    return 0 <= dim < len(shape) and shape[dim] % n == 0 and shape[dim] >= n


def _split0_ok(node: Node, n: int) -> bool:
    """dim-0 (row) splits must preserve sequences: when the node carries a
    `batch` attr (rows = batch*seq flattened), the true batch count must
    divide n-ways."""
    if not node.shape:
        return False
    b = node.attrs.get("batch", node.shape[0])
    return b % n == 0 and b >= n and node.shape[0] % n == 0


def op_strategies(g: Graph, node: Node, n: int) -> List[OpStrategy]:
    """All candidate strategies for `node` at shard count n (plus the
    always-available fully-replicated fallback)."""
    sh = node.shape
    ins = [g.nodes[i] for i in node.inputs]
    out: List[OpStrategy] = []

    def rep():
        return OpStrategy(R(n), tuple(R(n) for _ in ins), "rep")

    if node.op in ("param", "data", "const"):
        # producers: free to be replicated; consumers decide splits
        cands = [rep()]
        for d in range(len(sh)):
            if _splittable(sh, d, n):
                cands.append(OpStrategy(S(d, n), (), f"S{d}"))
        return cands

    if node.op == "linear":
        # x[M,K] @ w[N,K]^T (+b[N])
        nb = len(ins)
        if _split0_ok(node, n):    # token/DP split
            out.append(OpStrategy(S(0, n),
                                  tuple([S(0, n)] + [R(n)] * (nb - 1)),
                                  "row"))
        if _splittable(sh, 1, n):  # column TP (weight out-dim split)
            ii = [R(n), S(0, n)] + ([S(0, n)] if nb > 2 else [])
            out.append(OpStrategy(S(1, n), tuple(ii), "col_tp"))
        xK = ins[0].shape[-1]
        if xK % n == 0:            # K split (row-parallel, partial out)
            ii = [S(len(ins[0].shape) - 1, n), S(1, n)] + \
                ([R(n)] if nb > 2 else [])
            out.append(OpStrategy(P(n), tuple(ii), "row_tp"))
        out.append(rep())
        return out

    if node.op == "matmul":
        nd = len(sh)
        a, b = ins[0], ins[1]
        for d in range(nd - 2):    # batch dims
            if _splittable(sh, d, n):
                out.append(OpStrategy(S(d, n), (S(d, n), S(d, n)), f"b{d}"))
        if _splittable(sh, nd - 2, n):
            out.append(OpStrategy(S(nd - 2, n),
                                  (S(len(a.shape) - 2, n), R(n)), "m"))
        if _splittable(sh, nd - 1, n):
            out.append(OpStrategy(S(nd - 1, n),
                                  (R(n), S(len(b.shape) - 1, n)), "n"))
        if a.shape[-1] % n == 0:
            out.append(OpStrategy(P(n), (S(len(a.shape) - 1, n),
                                         S(len(b.shape) - 2, n)), "k"))
        out.append(rep())
        return out

    if node.op in ("layernorm", "rmsnorm", "softmax"):
        # reduction over the last dim: split any other dim; aux params rep
        for d in range(len(sh) - 1):
            if (_splittable(sh, d, n) if d > 0 else _split0_ok(node, n)):
                ii = [S(d, n)] + [R(n)] * (len(ins) - 1)
                out.append(OpStrategy(S(d, n), tuple(ii), f"S{d}"))
        out.append(rep())
        return out

    if node.op == "attention":
        # 2-D form (BS, d) with batch/heads attrs, or 4-D [B,H,S,D]:
        # batch or head split (sequence split = ring attention is out of
        # parity scope)
        if len(sh) == 2:
            if _split0_ok(node, n):
                out.append(OpStrategy(S(0, n),
                                      tuple(S(0, n) for _ in ins), "batch"))
            heads = node.attrs.get("heads", 0)
            if heads and heads % n == 0 and _splittable(sh, 1, n):
                out.append(OpStrategy(S(1, n),
                                      tuple(S(1, n) for _ in ins), "head"))
        else:
            for d in (0, 1):
                if _splittable(sh, d, n):
                    out.append(OpStrategy(S(d, n),
                                          tuple(S(d, n) for _ in ins),
                                          f"S{d}"))
        out.append(rep())
        return out

    if node.op == "moe_dispatch":
        # [T,d] x [T,E] -> [E, C, d] group-blocked capacity: a token
        # shard fills ITS capacity block, so the capacity dim (1) splits
        # with the token inputs; the downstream expert matmuls want
        # expert-dim (0) splits — the planner reshard between them is the
        # MoE all-to-all (reference kDAPPLEAllToAll, SURVEY §2.7 EP)
        if node.shape[1] % n == 0 and _split0_ok(g.nodes[node.inputs[0]], n):
            out.append(OpStrategy(S(1, n), (S(0, n), S(0, n)), "ep"))
        out.append(rep())
        return out

    if node.op == "moe_combine":
        # [E,C,d] x [T,d] x [T,E] -> [T,d]: token-split output consumes
        # the capacity-split expert results for ITS token shard
        if _split0_ok(node, n) and g.nodes[node.inputs[0]].shape[1] % n == 0:
            out.append(OpStrategy(S(0, n), (S(1, n), S(0, n), S(0, n)),
                                  "ep"))
        out.append(rep())
        return out

    if node.op == "attention_qkv":
        # packed [BS, 3d] projection in, [BS, d] out: batch split only (a
        # head split would need a strided/per-section weight shard — the
        # reference's stride_on_dim; not expressible as a contiguous
        # DimStrategy split of the packed projection)
        if _split0_ok(node, n):
            out.append(OpStrategy(S(0, n), (S(0, n),), "batch"))
        out.append(rep())
        return out

    if node.op == "embedding":
        table = ins[1]
        if _split0_ok(node, n):
            out.append(OpStrategy(S(0, n), (S(0, n), R(n)), "ids"))
        if table.shape[0] % n == 0:   # vocab split -> partial (masked)
            out.append(OpStrategy(P(n), (R(n), S(0, n)), "vocab"))
        if table.shape[1] % n == 0:   # hidden split
            out.append(OpStrategy(S(len(sh) - 1, n), (R(n), S(1, n)), "dim"))
        out.append(rep())
        return out

    if node.op == "cross_entropy":
        logits = ins[0]
        if _split0_ok(node, n) or (not node.shape and
                                   logits.attrs.get("batch",
                                                    logits.shape[0]) % n == 0):
            out.append(OpStrategy(P(n), (S(0, n), S(0, n)), "rows"))
        if logits.shape[1] % n == 0:
            out.append(OpStrategy(P(n), (S(1, n), R(n)), "vocab"))
        out.append(rep())
        return out

    if node.op in ("add", "mul", "bias_add", "swiglu"):
        for d in range(len(sh)):
            if (_splittable(sh, d, n) if d > 0 else _split0_ok(node, n)):
                ii = []
                ok = True
                for i in ins:
                    off = len(sh) - len(i.shape)
                    di = d - off
                    if di < 0 or i.shape[di] == 1:
                        ii.append(R(n))      # broadcast input
                    elif _splittable(i.shape, di, n):
                        ii.append(S(di, n))
                    else:
                        ok = False
                        break
                if ok:
                    out.append(OpStrategy(S(d, n), tuple(ii), f"S{d}"))
        out.append(rep())
        return out

    if node.op == "rope":
        # position depends on the token index: only sequence-preserving
        # batch splits keep the rotation angles right
        if _split0_ok(node, n):
            out.append(OpStrategy(S(0, n), (S(0, n),), "S0"))
        out.append(rep())
        return out

    if node.op in ("gelu", "dropout", "cast", "scale"):
        for d in range(len(sh)):
            if (_splittable(sh, d, n) if d > 0 else _split0_ok(node, n)):
                out.append(OpStrategy(S(d, n),
                                      tuple([S(d, n)] +
                                            [R(n)] * (len(ins) - 1)), f"S{d}"))
        out.append(rep())
        return out

    if node.op == "reshape":
        dim_map = node.attrs.get("dim_map", {})  # {in_dim: out_dim}
        src = ins[0]
        for din_s, dout in dim_map.items():
            din = int(din_s) if isinstance(din_s, str) else din_s
            if _splittable(src.shape, din, n) and _splittable(sh, dout, n):
                out.append(OpStrategy(S(dout, n), (S(din, n),), f"m{dout}"))
        out.append(rep())
        return out

    if node.op == "transpose":
        perm = node.attrs.get("perm")
        if perm:
            for dout, din in enumerate(perm):
                if _splittable(sh, dout, n):
                    out.append(OpStrategy(S(dout, n), (S(din, n),),
                                          f"t{dout}"))
        out.append(rep())
        return out

    if node.op == "split":
        # chunk along attrs['dim']: any OTHER dim may be split
        cd = node.attrs.get("dim", len(sh) - 1) % len(sh)
        for d in range(len(sh)):
            if d != cd and (_splittable(sh, d, n) if d > 0
                            else _split0_ok(node, n)):
                out.append(OpStrategy(S(d, n), (S(d, n),), f"S{d}"))
        out.append(rep())
        return out

    # collectives inserted by an EARLIER mesh round (multi-round transform:
    # a later round treats them as ops; their own round's group semantics
    # are orthogonal to this round's split)
    if node.op in ("copy_to", "all_reduce"):
        # shape-preserving; reducing/broadcasting over another round's group
        # commutes with splitting on any dim of this round
        for d in range(len(sh)):
            if (_splittable(sh, d, n) if d > 0 else _split0_ok(node, n)):
                out.append(OpStrategy(S(d, n),
                                      tuple(S(d, n) for _ in ins), f"S{d}"))
        out.append(rep())
        return out

    if node.op == "dynamic_slice":
        # slices along attrs['dim'] on another round's group: any dim may
        # be split this round — even the slice dim itself (nested narrows
        # compose into a hierarchical slice)
        for d in range(len(sh)):
            if (_splittable(sh, d, n) if d > 0 else _split0_ok(node, n)):
                out.append(OpStrategy(S(d, n), (S(d, n),), f"S{d}"))
        out.append(rep())
        return out

    if node.op == "all_gather":
        # concatenates along attrs['dim'] on another round's group: this
        # round may split any OTHER dim
        cd = node.attrs.get("dim", 0) % max(len(sh), 1)
        for d in range(len(sh)):
            if d != cd and (_splittable(sh, d, n) if d > 0
                            else _split0_ok(node, n)):
                out.append(OpStrategy(S(d, n), (S(d, n),), f"S{d}"))
        out.append(rep())
        return out

    if node.op == "all_to_all":
        sd = node.attrs.get("src_dim", 0)
        dd = node.attrs.get("dst_dim", 0)
        for d in range(len(sh)):
            if d in (sd, dd):
                continue
            if (_splittable(sh, d, n) if d > 0 else _split0_ok(node, n)):
                out.append(OpStrategy(S(d, n), (S(d, n),), f"S{d}"))
        out.append(rep())
        return out

    if node.op == "conv2d":
        # [B,C,H,W]: batch split or out-channel split
        if _splittable(sh, 0, n):
            out.append(OpStrategy(S(0, n),
                                  tuple([S(0, n)] + [R(n)] * (len(ins) - 1)),
                                  "b"))
        if _splittable(sh, 1, n):
            out.append(OpStrategy(S(1, n),
                                  tuple([R(n), S(0, n)] +
                                        [S(0, n)] * (len(ins) - 2)), "oc"))
        out.append(rep())
        return out

    # default: replicated only
    return [rep()]


def back_infer(g: Graph, node: Node, want: DimStrategy,
               n: int) -> Optional[OpStrategy]:
    """Finds an op strategy whose output matches `want` (the reference's
    BackInfer* family). Returns None if no exact match."""
    for st in op_strategies(g, node, n):
        if st.out == want:
            return st
    return None
