"""Distributed sharded checkpoint save/restore.

Re-implements the reference CheckpointUtil design
(pjrt/distributed_checkpoint_utils.h:31-57): every rank
writes its variable SHARDS with slice metadata (full shape, shard dim,
shard index/count — the VariableSpecsMgr information), a master index maps
variables to shards, restore reads each rank's slices back (with automatic
resharding when the restore-time layout differs from save-time), and a
`max_to_keep` rotation queue is persisted. Storage: safetensors files +
a JSON index per step (the tensor-bundle role)."""

from __future__ import annotations

import json
import os
import shutil
from dataclasses import asdict, dataclass
from typing import Dict, Optional, Tuple

import torch
from safetensors.torch import load_file, save_file


@dataclass
class SliceMeta:
    """Slice semantics of one saved shard (reference VariableSpec,
    variable_specs.h:27-76)."""
    full_shape: Tuple[int, ...]
    shard_dim: int = -1          # -1 = full tensor
    shard_index: int = 0
    num_shards: int = 1


class CheckpointManager:
    def __init__(self, directory: str, max_to_keep: int = 5):
        self.dir = directory
        self.max_to_keep = max_to_keep
        os.makedirs(directory, exist_ok=True)

    # -- paths -------------------------------------------------------------

    def _step_dir(self, step: int) -> str:
        return os.path.join(self.dir, f"step-{step}")

    def _queue_path(self) -> str:
        return os.path.join(self.dir, "checkpoint")

    # -- save ---------------------------------------------------------------

    def save(self, step: int, shards: Dict[str, Tuple[torch.Tensor,
                                                      SliceMeta]],
             rank: int = 0, world: int = 1):
        """Each rank calls with ITS shards; rank 0 additionally writes the
        index and rotates old checkpoints. Caller synchronizes ranks
        (barrier) around this."""
        d = self._step_dir(step)
        os.makedirs(d, exist_ok=True)
        tensors = {}
        meta = {}
        for name, (t, sm) in shards.items():
            key = f"{name}@{sm.shard_dim}.{sm.shard_index}.{sm.num_shards}"
            tensors[key] = t.detach().contiguous().cpu()
            meta[key] = {"name": name, **asdict(sm)}
        save_file(tensors, os.path.join(d, f"shards-{rank}.safetensors"))
        with open(os.path.join(d, f"meta-{rank}.json"), "w") as f:
            json.dump(meta, f)
        if rank == 0:
            with open(os.path.join(d, "index.json"), "w") as f:
                json.dump({"step": step, "world": world}, f)
            self._rotate(step)

    def _rotate(self, new_step: int):
        steps = []
        if os.path.exists(self._queue_path()):
            steps = json.load(open(self._queue_path()))["steps"]
        if new_step not in steps:
            steps.append(new_step)
        while len(steps) > self.max_to_keep:
            old = steps.pop(0)
            shutil.rmtree(self._step_dir(old), ignore_errors=True)
        with open(self._queue_path(), "w") as f:
            json.dump({"steps": steps, "latest": steps[-1]}, f)

    # -- restore -------------------------------------------------------------

    def latest_step(self) -> Optional[int]:
        if not os.path.exists(self._queue_path()):
            return None
        return json.load(open(self._queue_path())).get("latest")

    def _load_all(self, step: int):
        d = self._step_dir(step)
        merged: Dict[str, list] = {}
        for fn in sorted(os.listdir(d)):
            if not fn.startswith("meta-"):
                continue
            r = fn[len("meta-"):-len(".json")]
            meta = json.load(open(os.path.join(d, fn)))
            data = load_file(os.path.join(d, f"shards-{r}.safetensors"))
            for key, m in meta.items():
                merged.setdefault(m["name"], []).append((m, data[key]))
        return merged

    def restore(self, step: int,
                want: Dict[str, SliceMeta]) -> Dict[str, torch.Tensor]:
        """Returns each requested variable in the requested slice layout,
        assembling saved shards and re-slicing as needed."""
        saved = self._load_all(step)
        out = {}
        for name, sm in want.items():
            if name not in saved:
                raise KeyError(f"{name} not in checkpoint step {step}")
            full = self._assemble(saved[name])
            assert tuple(full.shape) == tuple(sm.full_shape), \
                (name, full.shape, sm.full_shape)
            if sm.shard_dim < 0 or sm.num_shards == 1:
                out[name] = full
            else:
                if full.shape[sm.shard_dim] % sm.num_shards != 0:
                    raise ValueError(
                        f"{name}: dim {sm.shard_dim} of {tuple(full.shape)} "
                        f"not divisible into {sm.num_shards} shards")
                n = full.shape[sm.shard_dim] // sm.num_shards
                out[name] = full.narrow(sm.shard_dim, sm.shard_index * n,
                                        n).contiguous()
        return out

    @staticmethod
    def _assemble(pieces) -> torch.Tensor:
        metas = [m for m, _ in pieces]
        if len(pieces) == 1 and metas[0]["shard_dim"] < 0:
            return pieces[0][1]
        dim = metas[0]["shard_dim"]
        if dim < 0:  # replicated saved by several ranks: verify they agree
            first = pieces[0][1]
            for m, t in pieces[1:]:
                if not torch.equal(t, first):
                    raise ValueError(
                        f"{metas[0]['name']}: replicated copies from "
                        f"different ranks disagree")
            return first
        uniq = {}
        for m, t in pieces:
            prev = uniq.get(m["shard_index"])
            if prev is not None and not torch.equal(prev[1], t):
                raise ValueError(
                    f"{m['name']}: duplicate shard {m['shard_index']} "
                    f"copies disagree")
            uniq[m["shard_index"]] = (m, t)
        if sorted(uniq) != list(range(len(uniq))):
            raise ValueError(
                f"{metas[0]['name']}: missing shards "
                f"(have {sorted(uniq)})")
        ordered = [uniq[i][1] for i in sorted(uniq)]
        return torch.cat(ordered, dim=dim)
