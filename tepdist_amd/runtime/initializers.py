"""Server-side sharded variable initialization.

Re-implements the reference DistributedRandomInitializer
(pjrt/initializers.h:124-242): variables are initialized on
the server from init specs (random_uniform / random_normal /
truncated_normal / constant) with a counter-based generator that is
SHARD-AWARE: each shard generates exactly its slice, keyed by
(global_seed, var_name) and the element's GLOBAL linear index, so any
sharding layout produces bit-identical values to slicing the unsharded
tensor (tested in tests/test_state_cpu.py)."""

from __future__ import annotations

import hashlib
import math
from dataclasses import dataclass
from typing import Tuple

import torch


@dataclass
class InitSpec:
    kind: str = "random_normal"  # random_uniform|random_normal|truncated_normal|constant|zeros|ones
    mean: float = 0.0
    std: float = 0.02
    low: float = -0.05
    high: float = 0.05
    value: float = 0.0


def default_init_spec(name: str, shape) -> InitSpec:
    """Init-spec convention shared by the service's server-side variable
    creation and the planned-graph executor: biases/norm-offsets zero, norm
    gains one, matrices normal(0.02)."""
    if name.endswith(("_b", ".bias")) or "b_" in name.split(".")[-1]:
        return InitSpec("zeros")
    if name.endswith("_g") or "ln" in name:
        return InitSpec("ones") if name.endswith("_g") else InitSpec("zeros")
    if len(shape) >= 2:
        return InitSpec("random_normal", std=0.02)
    return InitSpec("zeros")


def _var_seed(global_seed: int, name: str) -> int:
    h = hashlib.sha256(f"{global_seed}:{name}".encode()).digest()
    return int.from_bytes(h[:8], "little") & 0x7FFFFFFFFFFFFFFF


def _uniform_at(seed: int, idx: torch.Tensor) -> torch.Tensor:
    """Deterministic uniform [0,1) at arbitrary global linear indices
    (counter-based splitmix64 keyed by (seed, index)). Computed in numpy
    uint64 — torch int64 is signed and its arithmetic shifts would break
    the mix."""
    import numpy as np
    z = idx.numpy().astype(np.uint64)
    with np.errstate(over="ignore"):
        z = z * np.uint64(0x9E3779B97F4A7C15) + np.uint64(seed)
        z = (z ^ (z >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)
        z = (z ^ (z >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)
        z = z ^ (z >> np.uint64(31))
    u = (z >> np.uint64(11)).astype(np.float64) / float(1 << 53)
    return torch.from_numpy(u.astype(np.float32))


def _global_indices(full_shape: Tuple[int, ...], shard_dim: int,
                    shard_index: int, num_shards: int) -> torch.Tensor:
    shape = list(full_shape)
    if shard_dim < 0 or num_shards == 1:
        n = 1
        for s in shape:
            n *= s
        return torch.arange(n, dtype=torch.int64)
    assert shape[shard_dim] % num_shards == 0
    inner = 1
    for s in full_shape[shard_dim + 1:]:
        inner *= s
    outer = 1
    for s in full_shape[:shard_dim]:
        outer *= s
    run = (full_shape[shard_dim] // num_shards) * inner
    stride = full_shape[shard_dim] * inner
    base = torch.arange(outer, dtype=torch.int64) * stride + \
        shard_index * run
    return (base.unsqueeze(1) + torch.arange(run)).reshape(-1)


def init_shard(name: str, full_shape: Tuple[int, ...], spec: InitSpec,
               global_seed: int = 1234, shard_dim: int = -1,
               shard_index: int = 0, num_shards: int = 1,
               dtype=torch.bfloat16) -> torch.Tensor:
    """Generates THIS shard's slice of the variable (shard_dim=-1 =
    replicated/full)."""
    shape = list(full_shape)
    if shard_dim >= 0 and num_shards > 1:
        shape[shard_dim] //= num_shards

    if spec.kind == "zeros" or (spec.kind == "constant" and spec.value == 0):
        return torch.zeros(shape, dtype=dtype)
    if spec.kind == "ones":
        return torch.ones(shape, dtype=dtype)
    if spec.kind == "constant":
        return torch.full(shape, spec.value, dtype=dtype)

    seed = _var_seed(global_seed, name)
    idx = _global_indices(tuple(full_shape), shard_dim, shard_index,
                          num_shards)
    # one counter stream, two values per element (keeps shard consistency
    # AND independence of the Box-Muller pair)
    u = _uniform_at(seed, idx * 2)
    if spec.kind == "random_uniform":
        out = spec.low + (spec.high - spec.low) * u
    elif spec.kind in ("random_normal", "truncated_normal"):
        u2 = _uniform_at(seed, idx * 2 + 1)
        r = torch.sqrt(-2.0 * torch.log(u.clamp_min(1e-12)))
        out = spec.mean + spec.std * r * torch.cos(2 * math.pi * u2)
        if spec.kind == "truncated_normal":
            out = out.clamp(spec.mean - 2 * spec.std,
                            spec.mean + 2 * spec.std)
    else:
        raise ValueError(spec.kind)
    return out.reshape(shape).to(dtype)


def init_shard_multi(name: str, full_shape: Tuple[int, ...], spec: InitSpec,
                     splits, global_seed: int = 1234,
                     dtype=torch.bfloat16) -> torch.Tensor:
    """Multi-round shard init: `splits` is a list of (dim, shard_index,
    num_shards) narrows applied IN ORDER, each on the previous round's local
    shape (the multi_round_transform's param_rounds convention, with this
    rank's mesh coordinate as shard_index). Values are bit-identical to
    slicing the unsharded tensor the same way."""
    splits = [s for s in splits if s[2] > 1]
    if not splits:
        return init_shard(name, full_shape, spec, global_seed, dtype=dtype)
    if len(splits) == 1:
        d, i, n = splits[0]
        return init_shard(name, full_shape, spec, global_seed, shard_dim=d,
                          shard_index=i, num_shards=n, dtype=dtype)
    shape = list(full_shape)
    if spec.kind in ("zeros", "ones", "constant") or \
            (spec.kind == "constant" and spec.value == 0):
        for d, i, n in splits:
            assert shape[d] % n == 0, (name, shape, d, n)
            shape[d] //= n
        if spec.kind == "zeros":
            return torch.zeros(shape, dtype=dtype)
        if spec.kind == "ones":
            return torch.ones(shape, dtype=dtype)
        return torch.full(shape, spec.value, dtype=dtype)
    numel = 1
    for s in full_shape:
        numel *= s
    idx = torch.arange(numel, dtype=torch.int64).reshape(full_shape)
    for d, i, n in splits:
        assert idx.shape[d] % n == 0, (name, tuple(idx.shape), d, n)
        sz = idx.shape[d] // n
        idx = idx.narrow(d, i * sz, sz)
    shape = list(idx.shape)
    idx = idx.reshape(-1).contiguous()
    seed = _var_seed(global_seed, name)
    u = _uniform_at(seed, idx * 2)
    if spec.kind == "random_uniform":
        out = spec.low + (spec.high - spec.low) * u
    elif spec.kind in ("random_normal", "truncated_normal"):
        u2 = _uniform_at(seed, idx * 2 + 1)
        r = torch.sqrt(-2.0 * torch.log(u.clamp_min(1e-12)))
        out = spec.mean + spec.std * r * torch.cos(2 * math.pi * u2)
        if spec.kind == "truncated_normal":
            out = out.clamp(spec.mean - 2 * spec.std,
                            spec.mean + 2 * spec.std)
    else:
        raise ValueError(spec.kind)
    return out.reshape(shape).to(dtype)
