"""DistTensor: the distributed-tensor type (DAPPLEBuffer equivalent).

The reference's DAPPLEBuffer (pjrt/dapple_buffer.h:27-192) pairs a host
view of the GLOBAL tensor with per-device shards and knows how to gather
itself (`ToLiteral`). Here a DistTensor pairs this rank's LOCAL shard
with the slicing metadata the multi-round transform produced — the
per-round (round, dim, nshards) narrows plus the rank's mesh coordinates
— and can

  - `to_full()`: reassemble the global tensor (one all-gather per
    splitting round over that round's CommDevManager group, innermost
    round first),
  - `from_full(...)`: take this rank's nested shard of a global tensor,
  - `slice_meta()`: express the shard as checkpoint SliceMeta
    (single-round shards map directly; nested shards gather their inner
    rounds first so the saved slice stays one-dimensional).

PlannedModule exposes its parameters as DistTensors (`dist_param`), which
is what variable fetch and sharded checkpointing consume.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Tuple

import torch

from tepdist_amd.parallel.mappings import gather_from_group
from tepdist_amd.runtime.checkpoint import SliceMeta


@dataclass
class DistTensor:
    local: torch.Tensor
    # (round, dim, nshards) narrows in application order (outermost first)
    rounds: List[Tuple[int, int, int]]
    comm: object                      # CommDevManager (None = unsharded)
    name: str = ""

    @property
    def global_shape(self) -> Tuple[int, ...]:
        sh = list(self.local.shape)
        for (_, dim, n) in reversed(self.rounds):
            sh[dim] *= n
        return tuple(sh)

    @property
    def is_sharded(self) -> bool:
        return any(n > 1 for (_, _, n) in self.rounds)

    def to_full(self) -> torch.Tensor:
        """Gathers the global tensor on every rank of the involved groups
        (reference DAPPLEBuffer::ToLiteral). Innermost round first: each
        gather undoes the latest narrow."""
        t = self.local
        for (r, dim, n) in reversed(self.rounds):
            if n <= 1:
                continue
            grp = self.comm.mesh_group(r) if self.comm is not None else None
            t = gather_from_group(t.contiguous(), grp, dim=dim)
        return t

    @staticmethod
    def from_full(full: torch.Tensor, rounds, comm,
                  name: str = "") -> "DistTensor":
        coords = comm.coords()[1] if comm is not None else []
        t = full
        for (r, dim, n) in rounds:
            if n <= 1:
                continue
            idx = coords[r] if r < len(coords) else 0
            sz = t.shape[dim] // n
            t = t.narrow(dim, idx * sz, sz)
        return DistTensor(t.contiguous(), list(rounds), comm, name)

    def slice_meta(self) -> Tuple[torch.Tensor, SliceMeta]:
        """(tensor-to-save, SliceMeta) for the sharded checkpoint writer.
        One splitting round maps directly; nested shards gather the INNER
        rounds so the saved slice stays single-dimensional (checkpoint
        layout independence comes from the restore-side reslicing)."""
        splits = [s for s in self.rounds if s[2] > 1]
        if not splits:
            return self.local, SliceMeta(tuple(self.local.shape))
        if len(splits) == 1:
            (r, dim, n) = splits[0]
            coords = self.comm.coords()[1] if self.comm is not None else [0]
            idx = coords[r] if r < len(coords) else 0
            return self.local, SliceMeta(self.global_shape, dim, idx, n)
        # nested: gather all but the OUTERMOST round
        t = self.local
        for (r, dim, n) in reversed(splits[1:]):
            grp = self.comm.mesh_group(r) if self.comm is not None else None
            t = gather_from_group(t.contiguous(), grp, dim=dim)
        (r0, dim0, n0) = splits[0]
        coords = self.comm.coords()[1] if self.comm is not None else [0]
        idx0 = coords[r0] if r0 < len(coords) else 0
        return t, SliceMeta(self.global_shape, dim0, idx0, n0)
