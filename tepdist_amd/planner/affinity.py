"""Instruction co-location affinity.

The reference's InstAffinityMap (service/parallel/inst_affinity_map.h:33-60,
applied at cost_spmd_strategy.cc:4336-4344) biases the strategy search so
that aliased pairs (a variable and its updated value) and optimizer
auxiliary variables (AUX_AFFINITY: Adam moments with their parameter) land
on the same devices with the same sharding. Here affinity is applied as a
post-pass over the planned specs: every member of an affinity group adopts
the group leader's DimStrategy (the ZeRO planner and checkpoint manager
then shard optimizer state identically to the parameter)."""

from __future__ import annotations

from typing import Dict, List

from tepdist_amd.config import get_env
from tepdist_amd.ir.graph import Graph
from tepdist_amd.planner.dist_spec import DimStrategy


class InstAffinityMap:
    def __init__(self, graph: Graph, aux_affinity: bool = None):
        self.g = graph
        if aux_affinity is None:
            aux_affinity = get_env().aux_affinity
        self.groups: List[List[int]] = []
        self._of: Dict[int, int] = {}
        if aux_affinity:
            self._build_aux_groups()

    def _build_aux_groups(self):
        """Optimizer aux variables follow their parameter: params named
        `opt.m.X` / `opt.v.X` (the checkpoint naming, runtime/checkpoint.py)
        or with an `aux_of` attr group with param X."""
        by_name = dict(self.g.params)
        for name, nid in self.g.params.items():
            base = None
            if name.startswith(("opt.m.", "opt.v.")):
                base = name.split(".", 2)[2]
            aux_of = self.g.nodes[nid].attrs.get("aux_of")
            if aux_of:
                base = aux_of
            if base and base in by_name:
                self.add_affinity(by_name[base], nid)

    def add_affinity(self, a: int, b: int):
        """Declares a and b co-located (the reference's in/out-alias and
        AUX_AFFINITY rules)."""
        ga, gb = self._of.get(a), self._of.get(b)
        if ga is None and gb is None:
            self.groups.append([a, b])
            self._of[a] = self._of[b] = len(self.groups) - 1
        elif ga is None:
            self.groups[gb].append(a)
            self._of[a] = gb
        elif gb is None:
            self.groups[ga].append(b)
            self._of[b] = ga
        elif ga != gb:
            for m in self.groups[gb]:
                self._of[m] = ga
            self.groups[ga].extend(self.groups[gb])
            self.groups[gb] = []

    def apply(self, specs: Dict[int, DimStrategy]) -> Dict[int, DimStrategy]:
        """Forces every group member to the leader's spec (first member
        with a decided, non-glue strategy)."""
        for grp in self.groups:
            lead = next((specs[m] for m in grp
                         if m in specs and not specs[m].is_glue), None)
            if lead is None:
                continue
            for m in grp:
                specs[m] = lead
        return specs
