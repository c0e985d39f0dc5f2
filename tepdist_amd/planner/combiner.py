"""All-reduce combiner: fuse small independent all-reduces into one.

The reference's DAPPLEAllReduceCombiner (service/parallel/
dapple_all_reduce_combiner.h:32-50) merges independent kDAPPLEAllReduce
instructions into a single variadic op under byte/count thresholds, so one
RCCL call covers many small tensors (launch + ring-latency amortization —
on xGMI each ring step is bound by a single ~153 GB/s link, so fewer,
larger collectives win). On our IR the bundle is one `all_reduce` node
with k inputs followed by k `bundle_get` projections; the interpreter
flattens, reduces once, and splits."""

from __future__ import annotations

from typing import Dict, List

from tepdist_amd.ir.graph import Graph


def combine_all_reduces(g: Graph, max_bytes: float = 32 * 1024 * 1024,
                        max_count: int = 64) -> int:
    """Combines independent single-input all_reduce nodes into bundles.
    Returns the number of bundles created. Mutates g."""
    ars = [n for n in g.topo()
           if n.op == "all_reduce" and len(n.inputs) == 1]
    if len(ars) < 2:
        return 0
    # independence: no all_reduce may (transitively) feed another one in
    # the same bundle. ancestors() per candidate over the ar set.
    anc: Dict[int, set] = {}
    for n in g.topo():
        s = set()
        for i in g.nodes[n.id].inputs:
            s.add(i)
            s |= anc.get(i, set())
        anc[n.id] = s

    bundles: List[List] = []
    cur, cur_bytes = [], 0.0
    for n in ars:
        nb = g.bytes_of(n)
        dep = any(m.id in anc[n.id] for m in cur)
        if cur and (dep or cur_bytes + nb > max_bytes or
                    len(cur) >= max_count):
            bundles.append(cur)
            cur, cur_bytes = [], 0.0
        cur.append(n)
        cur_bytes += nb
    bundles.append(cur)

    made = 0
    for group in bundles:
        if len(group) < 2:
            continue
        ins = [g.nodes[m.inputs[0]] for m in group]
        total = sum(m.numel for m in group)
        bundle = g.add("all_reduce", ins, (total,), group[0].dtype,
                       {"bundled": True})
        cons = g.consumers()
        for idx, m in enumerate(group):
            proj = g.add("bundle_get", [bundle], m.shape, m.dtype,
                         {"index": idx,
                          "offsets": [int(x.numel) for x in group]})
            for u in cons[m.id]:
                un = g.nodes[u]
                un.inputs = [proj.id if i == m.id else i for i in un.inputs]
            g.outputs = [proj.id if o == m.id else o for o in g.outputs]
            del g.nodes[m.id]
        made += 1
    return made
