"""CommDevManager: the SplitId -> process-group registry.

Re-implements the reference's CommDevManager / NcclContext pairing
(pjrt/dev_id_util.h:192-330 builds per-split-ordinal DevGroupArrays;
pjrt/nccl_context.h:34-74 maps sorted-device-group keys to communicators;
service_rt.cc:310-334 bootstraps them over RPC). On MI355X the
communicator layer is torch.distributed process groups over RCCL
(backend "nccl" IS RCCL on ROCm; gloo for CPU tests) — one subgroup per
mesh-round coordinate slice, plus per-edge 2-rank groups for pipeline
send/recv.

Rank layout (one process per GPU): stage-major, then mesh rounds with the
LAST round fastest-varying, so the innermost (usually tensor-parallel)
round lands on consecutive ranks — consecutive GPUs share the node's xGMI
mesh, which is where the chattiest collectives belong.

    rank = ((stage * mesh[0] + c0) * mesh[1] + c1) * ... + c_last

Every rank must construct every group (torch.distributed.new_group is
collective over the WORLD), so the registry is built identically on all
ranks from the plan alone.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Tuple

import torch.distributed as dist


class CommDevManager:
    def __init__(self, mesh: Sequence[int], pp: int = 1,
                 rank: Optional[int] = None,
                 world: Optional[int] = None):
        """mesh[r] = shard count of mesh round r (the per-stage SPMD mesh);
        pp = number of pipeline stages. world must equal pp * prod(mesh)."""
        self.mesh = [int(n) for n in mesh if int(n) > 1] or [1]
        self.pp = max(int(pp), 1)
        size = self.pp
        for n in self.mesh:
            size *= n
        if dist.is_initialized():
            self.world = world if world is not None else dist.get_world_size()
            self.rank = rank if rank is not None else dist.get_rank()
        else:
            self.world = world if world is not None else 1
            self.rank = rank if rank is not None else 0
        if size != self.world:
            raise ValueError(f"plan needs {size} ranks "
                             f"(pp={self.pp} x mesh={self.mesh}), "
                             f"world is {self.world}")
        # round ordinal -> this rank's group; plus every group cached by its
        # sorted rank tuple so equal groups are shared
        self._by_ranks: Dict[Tuple[int, ...], object] = {}
        self.round_groups: Dict[int, object] = {}
        self.stage_group = None          # same-stage ranks (full mesh)
        self._build()

    # -- layout ----------------------------------------------------------

    def coords(self, rank: Optional[int] = None) -> Tuple[int, List[int]]:
        """(stage, [c0, c1, ...]) of `rank` under the stage-major layout."""
        r = self.rank if rank is None else rank
        cs = []
        for n in reversed(self.mesh):
            cs.append(r % n)
            r //= n
        return r, list(reversed(cs))

    def rank_of(self, stage: int, coords: Sequence[int]) -> int:
        r = stage
        for n, c in zip(self.mesh, coords):
            r = r * n + c
        return r

    @property
    def stage(self) -> int:
        return self.coords()[0]

    @property
    def mesh_size(self) -> int:
        s = 1
        for n in self.mesh:
            s *= n
        return s

    # -- groups ----------------------------------------------------------

    def _group(self, ranks: List[int]):
        key = tuple(sorted(ranks))
        if key not in self._by_ranks:
            if dist.is_initialized() and len(key) > 1:
                self._by_ranks[key] = dist.new_group(list(key))
            else:
                self._by_ranks[key] = None
        return self._by_ranks[key]

    def _build(self):
        """Creates all round groups (every rank executes the same sequence
        of new_group calls, as torch.distributed requires)."""
        import itertools
        mesh = self.mesh
        for ri in range(len(mesh)):
            others = [range(n) for i, n in enumerate(mesh) if i != ri]
            for stage in range(self.pp):
                for combo in itertools.product(*others):
                    ranks = []
                    for c in range(mesh[ri]):
                        coords = list(combo[:ri]) + [c] + list(combo[ri:])
                        ranks.append(self.rank_of(stage, coords))
                    g = self._group(ranks)
                    if self.rank in ranks:
                        self.round_groups[ri] = g
        for stage in range(self.pp):
            ranks = [self.rank_of(stage, self._unflatten(i))
                     for i in range(self.mesh_size)]
            g = self._group(ranks)
            if self.rank in ranks:
                self.stage_group = g

    def _unflatten(self, i: int) -> List[int]:
        cs = []
        for n in reversed(self.mesh):
            cs.append(i % n)
            i //= n
        return list(reversed(cs))

    def mesh_group(self, round_i: int):
        """This rank's process group for mesh round `round_i` (None when
        that round has a single shard or world==1)."""
        return self.round_groups.get(round_i)

    def rounds_group(self, rounds):
        """This rank's group spanning the given SET of mesh rounds (ranks
        varying in those rounds' coordinates, all else fixed) — e.g. a
        gradient summed over both rounds of a dp x dp mesh needs ONE
        all-reduce over their combined group. Built collectively: every
        rank derives the same round-set sequence from the same plan."""
        import itertools
        rset = sorted(set(int(r) for r in rounds))
        if not rset:
            return None
        if len(rset) == 1:
            return self.mesh_group(rset[0])
        if len(rset) == len(self.mesh):
            return self.stage_group
        mesh = self.mesh
        fixed = [i for i in range(len(mesh)) if i not in rset]
        mine = None
        for stage in range(self.pp):
            for fixed_vals in itertools.product(
                    *[range(mesh[i]) for i in fixed]):
                ranks = []
                for var_vals in itertools.product(
                        *[range(mesh[i]) for i in rset]):
                    coords = [0] * len(mesh)
                    for i, v in zip(fixed, fixed_vals):
                        coords[i] = v
                    for i, v in zip(rset, var_vals):
                        coords[i] = v
                    ranks.append(self.rank_of(stage, coords))
                grp = self._group(ranks)
                if self.rank in ranks:
                    mine = grp
        return mine

    def groups_dict(self) -> Dict[int, object]:
        return dict(self.round_groups)

    # -- pipeline edges ---------------------------------------------------

    def pipeline_peer(self, direction: int) -> Optional[int]:
        """Global rank of the same-coordinate rank one stage over
        (direction +1 = next stage, -1 = previous); None at the boundary."""
        stage, coords = self.coords()
        t = stage + direction
        if t < 0 or t >= self.pp:
            return None
        return self.rank_of(t, coords)

    def pipeline_pair_group(self, direction: int):
        """2-rank group for this rank's pipeline edge (the reference builds
        one comm per cross-stage edge, virtual_client.cc:2161-2192).
        All edge groups are created on every rank."""
        out = None
        for stage in range(self.pp - 1):
            for i in range(self.mesh_size):
                coords = self._unflatten(i)
                a = self.rank_of(stage, coords)
                b = self.rank_of(stage + 1, coords)
                g = self._group([a, b])
                if self.rank == a and direction > 0:
                    out = g
                if self.rank == b and direction < 0:
                    out = g
        return out

    def pipeline_column_group(self):
        """Group of ALL pipeline stages at this rank's mesh coordinate
        (the per-dp-slice pipeline column) — for whole-pipeline
        collectives like the final loss broadcast. A pair group is NOT a
        substitute at pp>=3: broadcast from the last stage needs every
        stage in one group."""
        out = None
        for i in range(self.mesh_size):
            coords = self._unflatten(i)
            ranks = [self.rank_of(s, coords) for s in range(self.pp)]
            g = self._group(ranks)
            if self.rank in ranks:
                out = g
        return out

    def describe(self) -> str:
        stage, coords = self.coords()
        return (f"CommDevManager[world={self.world} pp={self.pp} "
                f"mesh={self.mesh} rank={self.rank} -> stage={stage} "
                f"coords={coords}]")
