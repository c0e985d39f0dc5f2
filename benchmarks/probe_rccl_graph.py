"""Probe: can a hipGraph capture RCCL collectives? (world 1, backend nccl)

The trainer's multi-rank default is eager because captured-RCCL was
unverified (docs/ROUND2.md item 10). This probe exercises the capture
mechanics torch uses for collectives in graphs — ProcessGroupNCCL's
capture-safe work handling — at world 1 on one GPU: capture an
all_reduce + compute into a hipGraph, replay it, and check numerics.
Run under torchrun --nproc-per-node 1 (needs a process group).
"""

import os

import torch
import torch.distributed as dist


def main():
    dist.init_process_group("nccl")
    torch.cuda.set_device(0)
    x = torch.ones(1 << 20, device="cuda")
    y = torch.zeros_like(x)

    # eager warmup on a side stream (the capture recipe the trainer uses)
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(3):
            y.copy_(x * 2)
            dist.all_reduce(y)
    torch.cuda.current_stream().wait_stream(s)
    torch.cuda.synchronize()

    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        y.copy_(x * 2)
        dist.all_reduce(y)
        y.add_(1.0)
    for i in range(5):
        x.fill_(float(i))
        g.replay()
        torch.cuda.synchronize()
        expect = 2.0 * i + 1.0
        assert torch.allclose(y, torch.full_like(y, expect)), \
            (i, y[0].item(), expect)
    print("RCCL_GRAPH_CAPTURE_OK")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
