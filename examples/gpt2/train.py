"""GPT-2 auto-parallel training (the reference's examples/GPT2 counterpart).
Single node, one process per GPU:

  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 examples/gpt2/train.py --model gpt2-345m
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import torch

from tepdist_amd.ir import gpt2_ir
from tepdist_amd.models import GPT2, GPT2_CONFIGS
from tepdist_amd.parallel import GradReducer, init_distributed
from tepdist_amd.parallel.tp import ParallelEnv
from tepdist_amd.planner import AutoParallel
from tepdist_amd.train import AdamW, Trainer


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="gpt2-117m")
    ap.add_argument("--micro-batch", type=int, default=8)
    ap.add_argument("--seq", type=int, default=1024)
    ap.add_argument("--stop-at-step", type=int, default=10)
    ap.add_argument("--log-every-step", type=int, default=1)
    ap.add_argument("--ckpt-dir", default="")
    args = ap.parse_args()

    rank, world, local = init_distributed()
    dev = torch.device("cuda", local) if torch.cuda.is_available() else "cpu"
    cfg = GPT2_CONFIGS[args.model]
    seq = min(args.seq, cfg.n_ctx)
    plan = AutoParallel(gpt2_ir(cfg, args.micro_batch * world, seq),
                        world).run()
    if rank == 0:
        print(plan.summary())
    env = ParallelEnv.create(plan.tp) if world > 1 else ParallelEnv.single()
    dtype = torch.bfloat16 if dev != "cpu" else torch.float32
    model = GPT2(cfg, dtype=dtype, env=env).to(dev)
    opt = AdamW(model.parameters(), lr=1e-4)
    reducer = GradReducer(model.parameters(), env.dp_group) \
        if plan.dp > 1 else None
    trainer = Trainer(model, opt, reducer=reducer)

    g = torch.Generator().manual_seed(1234 + rank)
    for step in range(args.stop_at_step):
        ids = torch.randint(0, cfg.vocab_size,
                            (args.micro_batch, seq + 1), generator=g)
        t0 = time.time()
        loss = trainer.train_step(
            lambda i: (ids[:, :-1].to(dev), ids[:, 1:].to(dev)))
        if rank == 0 and step % args.log_every_step == 0:
            print(f"step {step} loss {loss:.4f} "
                  f"({(time.time() - t0) * 1e3:.1f} ms)")
        if args.ckpt_dir and step == args.stop_at_step - 1:
            from tepdist_amd.runtime.checkpoint import (CheckpointManager,
                                                        SliceMeta)
            mgr = CheckpointManager(args.ckpt_dir)
            shards = {n: (p.data, SliceMeta(tuple(p.shape)))
                      for n, p in model.named_parameters()}
            if rank == 0:
                mgr.save(step, shards, rank=0, world=world)


if __name__ == "__main__":
    main()
