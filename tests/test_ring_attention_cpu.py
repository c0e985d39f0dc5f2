"""Ring (context-parallel) attention (parallel/ring_attention.py):
sequence-sharded attention over ring P2P must match full-sequence
attention EXACTLY (fp32) in loss and q/k/v gradients — the beyond-parity
SP mode SURVEY §5.7 flags as the natural xGMI extension."""

import math
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from tests.conftest import free_port


def _full_ref(q, k, v, causal=True):
    q2, k2, v2 = (t.detach().clone().requires_grad_() for t in (q, k, v))
    out = torch.nn.functional.scaled_dot_product_attention(
        q2, k2, v2, is_causal=causal)
    return out, (q2, k2, v2)


def _worker(rank, world, port, causal):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from tepdist_amd.parallel.ring_attention import ring_attention
        torch.manual_seed(0)
        B, H, S, D = 2, 3, 8 * world, 16
        q = torch.randn(B, H, S, D)
        k = torch.randn(B, H, S, D)
        v = torch.randn(B, H, S, D)
        dout = torch.randn(B, H, S, D)
        sl = S // world
        qs = q[:, :, rank * sl:(rank + 1) * sl].clone().requires_grad_()
        ks = k[:, :, rank * sl:(rank + 1) * sl].clone().requires_grad_()
        vs = v[:, :, rank * sl:(rank + 1) * sl].clone().requires_grad_()

        out = ring_attention(qs, ks, vs, causal=causal)
        out.backward(dout[:, :, rank * sl:(rank + 1) * sl])

        ref, (q2, k2, v2) = _full_ref(q, k, v, causal)
        ref.backward(dout)
        blk = slice(rank * sl, (rank + 1) * sl)
        torch.testing.assert_close(out, ref[:, :, blk], rtol=1e-5,
                                   atol=1e-5)
        torch.testing.assert_close(qs.grad, q2.grad[:, :, blk], rtol=1e-4,
                                   atol=1e-5)
        torch.testing.assert_close(ks.grad, k2.grad[:, :, blk], rtol=1e-4,
                                   atol=1e-5)
        torch.testing.assert_close(vs.grad, v2.grad[:, :, blk], rtol=1e-4,
                                   atol=1e-5)
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("world,causal", [(2, True), (2, False),
                                          (4, True)])
@pytest.mark.timeout(600)
def test_ring_attention_matches_full(world, causal):
    port = free_port()
    mp.spawn(_worker, args=(world, port, causal), nprocs=world, join=True)


def test_ring_world1_matches_full():
    from tepdist_amd.parallel.ring_attention import ring_attention
    torch.manual_seed(0)
    B, H, S, D = 2, 3, 32, 16
    q = torch.randn(B, H, S, D, requires_grad=True)
    k = torch.randn(B, H, S, D, requires_grad=True)
    v = torch.randn(B, H, S, D, requires_grad=True)
    out = ring_attention(q, k, v, causal=True)
    out.sum().backward()
    ref, (q2, k2, v2) = _full_ref(q, k, v, True)
    ref.sum().backward()
    torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-5)
    torch.testing.assert_close(q.grad, q2.grad, rtol=1e-4, atol=1e-5)


def _zz_worker(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from tepdist_amd.parallel.ring_attention import (ring_attention,
                                                         zigzag_shard)
        torch.manual_seed(0)
        B, H, S, D = 2, 3, 16 * world, 16
        q = torch.randn(B, H, S, D)
        k = torch.randn(B, H, S, D)
        v = torch.randn(B, H, S, D)
        dout = torch.randn(B, H, S, D)
        qs = zigzag_shard(q, world)[rank].requires_grad_()
        ks = zigzag_shard(k, world)[rank].requires_grad_()
        vs = zigzag_shard(v, world)[rank].requires_grad_()
        out = ring_attention(qs, ks, vs, causal=True, zigzag=True)
        out.backward(zigzag_shard(dout, world)[rank])

        ref, (q2, k2, v2) = _full_ref(q, k, v, True)
        ref.backward(dout)
        torch.testing.assert_close(out, zigzag_shard(ref, world)[rank],
                                   rtol=1e-5, atol=1e-5)
        torch.testing.assert_close(qs.grad,
                                   zigzag_shard(q2.grad, world)[rank],
                                   rtol=1e-4, atol=1e-5)
        torch.testing.assert_close(ks.grad,
                                   zigzag_shard(k2.grad, world)[rank],
                                   rtol=1e-4, atol=1e-5)
        torch.testing.assert_close(vs.grad,
                                   zigzag_shard(v2.grad, world)[rank],
                                   rtol=1e-4, atol=1e-5)
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("world", [2, 4])
@pytest.mark.timeout(600)
def test_zigzag_ring_matches_full(world):
    port = free_port()
    mp.spawn(_zz_worker, args=(world, port), nprocs=world, join=True)


def test_zigzag_shard_roundtrip():
    from tepdist_amd.parallel.ring_attention import (zigzag_shard,
                                                     zigzag_unshard)
    x = torch.arange(2 * 1 * 16 * 2, dtype=torch.float32).reshape(
        2, 1, 16, 2)
    shards = zigzag_shard(x, 4)
    torch.testing.assert_close(zigzag_unshard(shards), x)
