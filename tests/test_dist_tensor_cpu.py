"""DistTensor (runtime/dist_tensor.py): the DAPPLEBuffer-equivalent
distributed tensor — gather-to-full over the mesh groups, nested-shard
slicing, checkpoint SliceMeta export, and a sharded save through the
PlannedModule (reference pjrt/dapple_buffer.h + CheckpointUtil)."""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from tests.conftest import free_port


def _worker(rank, world, port, tmpdir):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from tests.test_planned_cpu import (_build_case, _hand_hybrid_plan,
                                            _single_reference)
        from tepdist_amd.runtime.checkpoint import CheckpointManager, \
            SliceMeta
        from tepdist_amd.runtime.dist_tensor import DistTensor
        from tepdist_amd.runtime.planned import PlannedModule
        g, cfg = _build_case()
        plan = _hand_hybrid_plan(g, world)     # dp2 x tp2
        m = PlannedModule(g, plan)
        _, ref_vars, _ = _single_reference(g)

        # gather-to-full reproduces the unsharded weights exactly
        for name in ("h0.w_fc", "wte", "h0.w_out"):
            dt = m.dist_param(name)
            full = dt.to_full()
            assert tuple(full.shape) == tuple(ref_vars[name].shape)
            assert torch.equal(full, ref_vars[name]), name

        # from_full round-trips to the local shard
        dt = m.dist_param("h0.w_fc")
        back = DistTensor.from_full(ref_vars["h0.w_fc"], dt.rounds, m.comm)
        assert torch.equal(back.local, dt.local)

        # sharded checkpoint: every rank writes its slice; restore on the
        # FULL layout reassembles the global tensor
        ck = CheckpointManager(tmpdir)
        m.save_checkpoint(ck, step=1)
        dist.barrier()
        want = {"h0.w_fc": SliceMeta(tuple(ref_vars["h0.w_fc"].shape))}
        got = ck.restore(1, want)["h0.w_fc"]
        assert torch.equal(got, ref_vars["h0.w_fc"])
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_dist_tensor_gather_and_checkpoint(tmp_path):
    port = free_port()
    mp.spawn(_worker, args=(4, port, str(tmp_path)), nprocs=4, join=True)
