"""State-management tests: shard-aware initializers (bit-identical to the
unsharded tensor) and sharded checkpoint save/restore with resharding."""

import torch

from tepdist_amd.runtime.checkpoint import CheckpointManager, SliceMeta
from tepdist_amd.runtime.initializers import InitSpec, init_shard


def test_init_shard_consistency_dim0():
    spec = InitSpec("random_normal", std=0.02)
    full = init_shard("w", (16, 8), spec, dtype=torch.float32)
    parts = [init_shard("w", (16, 8), spec, shard_dim=0, shard_index=i,
                        num_shards=4, dtype=torch.float32) for i in range(4)]
    torch.testing.assert_close(torch.cat(parts, 0), full)


def test_init_shard_consistency_dim1():
    spec = InitSpec("random_uniform", low=-1, high=1)
    full = init_shard("v", (6, 12), spec, dtype=torch.float32)
    parts = [init_shard("v", (6, 12), spec, shard_dim=1, shard_index=i,
                        num_shards=3, dtype=torch.float32) for i in range(3)]
    torch.testing.assert_close(torch.cat(parts, 1), full)


def test_init_distribution():
    spec = InitSpec("random_normal", std=1.0)
    t = init_shard("big", (1000, 100), spec, dtype=torch.float32)
    assert abs(t.mean().item()) < 0.02
    assert abs(t.std().item() - 1.0) < 0.02
    # different names decorrelate
    t2 = init_shard("big2", (1000, 100), spec, dtype=torch.float32)
    assert (t - t2).abs().mean() > 0.5


def test_checkpoint_roundtrip_and_reshard(tmp_path):
    mgr = CheckpointManager(str(tmp_path), max_to_keep=2)
    w = torch.randn(8, 6)
    # save as 2 dim-0 shards from 2 "ranks"
    mgr.save(1, {"w": (w[:4], SliceMeta((8, 6), 0, 0, 2))}, rank=0, world=2)
    mgr.save(1, {"w": (w[4:], SliceMeta((8, 6), 0, 1, 2))}, rank=1, world=2)
    # restore full
    out = mgr.restore(1, {"w": SliceMeta((8, 6))})
    torch.testing.assert_close(out["w"], w)
    # restore resharded on dim 1 (3 shards)
    out = mgr.restore(1, {"w": SliceMeta((8, 6), 1, 2, 3)})
    torch.testing.assert_close(out["w"], w[:, 4:6])


def test_checkpoint_rotation(tmp_path):
    mgr = CheckpointManager(str(tmp_path), max_to_keep=2)
    for step in (1, 2, 3):
        mgr.save(step, {"x": (torch.ones(2), SliceMeta((2,)))})
    assert mgr.latest_step() == 3
    import os
    assert not os.path.exists(str(tmp_path / "step-1"))
    assert os.path.exists(str(tmp_path / "step-3"))
    out = mgr.restore(3, {"x": SliceMeta((2,))})
    assert torch.equal(out["x"], torch.ones(2))
