"""Flagship benchmark: GPT-2 auto-parallel training throughput on MI355X.

Contract (driver-facing):
  python bench.py --gpus N --steps K --warmup W
N>1 is launched by the driver via torch.distributed.run with one rank per
GPU over RCCL; this script reads RANK/LOCAL_RANK/WORLD_SIZE from the env.
W untimed warmup steps, then exactly K timed steps bracketed by a barrier +
torch.cuda.synchronize on both sides; elapsed time is MAX over ranks; rank 0
prints one JSON line with the whole-node aggregate tokens/sec.

Metric/config per BASELINE.json: tokens/sec (whole node), GPT-2
auto-parallel, synthetic data, random-init weights, bf16.
"""

from __future__ import annotations

import argparse
import json
import time

import torch

from tepdist_amd.models import GPT2, GPT2_CONFIGS
from tepdist_amd.parallel import GradReducer, init_distributed
from tepdist_amd.train import AdamW, Trainer


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--model", type=str, default="gpt2-345m")
    ap.add_argument("--micro-batch", type=int, default=8)
    ap.add_argument("--grad-accum", type=int, default=1)
    ap.add_argument("--seq", type=int, default=1024)
    ap.add_argument("--device", type=str, default=None,
                    help="override device (cpu for plumbing tests)")
    args = ap.parse_args()

    rank, world, local_rank = init_distributed()
    if args.device is not None:
        device = torch.device(args.device)
    else:
        device = torch.device("cuda", local_rank) if torch.cuda.is_available() \
            else torch.device("cpu")

    cfg = GPT2_CONFIGS[args.model]
    seq = min(args.seq, cfg.n_ctx)
    dtype = torch.bfloat16
    torch.manual_seed(1234)
    model = GPT2(cfg, dtype=dtype).to(device)
    opt = AdamW(model.parameters(), lr=1e-4)
    reducer = None
    if world > 1:
        reducer = GradReducer(model.parameters())
    trainer = Trainer(model, opt, grad_accum_steps=args.grad_accum,
                      reducer=reducer)

    # synthetic data of the benchmark shape (no network for datasets)
    g = torch.Generator().manual_seed(4321 + rank)
    def make_batch(_i):
        ids = torch.randint(0, cfg.vocab_size, (args.micro_batch, seq + 1),
                            generator=g)
        x = ids[:, :-1].to(device)
        y = ids[:, 1:].to(device)
        return x, y

    import torch.distributed as dist

    def barrier_sync():
        if device.type == "cuda":
            torch.cuda.synchronize(device)
        if world > 1:
            dist.barrier()
        if device.type == "cuda":
            torch.cuda.synchronize(device)

    for _ in range(args.warmup):
        trainer.train_step(make_batch)

    barrier_sync()
    t0 = time.perf_counter()
    loss = 0.0
    for _ in range(args.steps):
        loss = trainer.train_step(make_batch)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    if world > 1:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if device.type == "cuda" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    tokens_per_step = args.micro_batch * args.grad_accum * seq * world
    ms_per_step = elapsed / args.steps * 1000.0
    tokens_per_sec = tokens_per_step * args.steps / elapsed

    if rank == 0:
        print(json.dumps({
            "metric": "tokens/sec",
            "value": round(tokens_per_sec, 2),
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(ms_per_step, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16",
            "data": "synthetic",
            "final_loss": round(loss, 4),
            "config": {
                "model": args.model,
                "global_batch": args.micro_batch * args.grad_accum * world,
                "seq_len": seq,
                "parallelism": f"dp{world}",
            },
        }))

    if world > 1:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
