"""Per-tensor sharding descriptors.

Mirrors the reference's DimStrategy / DistSpec
(service/parallel/dist_spec.h:36-227, hlo_strategy_spec.h:28-133): a
DimStrategy describes one split round (partition dim, shard count, partial
flag, replicated, or Glue = undecided); a DistSpec stacks one DimStrategy
per device-mesh dim plus the pipeline stage."""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional

GLUE = "glue"          # undecided (planner has not assigned yet)
REPLICATED = "rep"     # full copy on every shard
SPLIT = "split"        # partitioned on partition_dim
PARTIAL = "partial"    # partial sums pending an all-reduce


@dataclass(frozen=True)
class DimStrategy:
    kind: str = GLUE
    partition_dim: int = -1
    num_shards: int = 1

    @staticmethod
    def glue() -> "DimStrategy":
        return DimStrategy(GLUE)

    @staticmethod
    def replicated(n: int = 1) -> "DimStrategy":
        return DimStrategy(REPLICATED, -1, n)

    @staticmethod
    def split(dim: int, n: int) -> "DimStrategy":
        return DimStrategy(SPLIT, dim, n)

    @staticmethod
    def partial(n: int) -> "DimStrategy":
        return DimStrategy(PARTIAL, -1, n)

    @property
    def is_glue(self) -> bool:
        return self.kind == GLUE

    @property
    def is_split(self) -> bool:
        return self.kind == SPLIT

    @property
    def is_partial(self) -> bool:
        return self.kind == PARTIAL

    @property
    def is_replicated(self) -> bool:
        return self.kind == REPLICATED

    def shard_numel_factor(self) -> float:
        return 1.0 / self.num_shards if self.is_split else 1.0

    def __str__(self):
        if self.is_split:
            return f"S{self.partition_dim}x{self.num_shards}"
        if self.is_partial:
            return f"P x{self.num_shards}"
        if self.is_replicated:
            return f"R x{self.num_shards}"
        return "?"


@dataclass
class DistSpec:
    """One DimStrategy per mesh dim (multi-round splits stack), plus the
    pipeline stage this tensor's producer is placed in."""
    dims: List[DimStrategy] = field(default_factory=list)
    stage: int = 0

    def round(self, i: int) -> DimStrategy:
        return self.dims[i] if i < len(self.dims) else DimStrategy.glue()

    def set_round(self, i: int, s: DimStrategy):
        while len(self.dims) <= i:
            self.dims.append(DimStrategy.glue())
        self.dims[i] = s

    def total_shards(self) -> int:
        n = 1
        for d in self.dims:
            if d.is_split:
                n *= d.num_shards
        return n

    def __str__(self):
        return "/".join(str(d) for d in self.dims) + f"@st{self.stage}"


# reshard edge classification (what collective moves a tensor from spec a
# to spec b; lowered by the runtime's reshard layer — the reference's
# CustomCollectiveExpander)

def reshard_collective(src: DimStrategy, dst: DimStrategy) -> Optional[str]:
    """Returns the collective needed on an edge whose producer has spec src
    and whose consumer wants dst (None = no-op)."""
    if src == dst or dst.is_glue:
        return None
    if src.is_partial:
        if dst.is_replicated:
            return "all_reduce"
        if dst.is_split:
            return "reduce_scatter"
    if src.is_split:
        if dst.is_replicated:
            return "all_gather"
        if dst.is_split and src.partition_dim != dst.partition_dim:
            return "all_to_all"
        if dst.is_split:
            return None
    if src.is_replicated and dst.is_split:
        return "dynamic_slice"
    if src.is_replicated and dst.is_replicated:
        return None
    if src.is_glue:
        return None
    return "all_gather"  # conservative default
