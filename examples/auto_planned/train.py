"""The planned-graph workflow end to end (what bench.py's auto path does,
as a minimal user script):

    capture IR -> (optional) sharding annotations -> AutoParallel ->
    PlannedModule (multi-round SpmdTransform + CommDevManager groups) ->
    Trainer

Single process runs as-is; under torchrun (one rank per GPU over RCCL)
the same script executes the plan distributed:

    python -m torch.distributed.run --nproc-per-node 8 \
        --master-addr 127.0.0.1 examples/auto_planned/train.py
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import torch

from tepdist_amd.ir import gpt2_ir
from tepdist_amd.ir.sharding import replicate, split
from tepdist_amd.models import GPT2_CONFIGS
from tepdist_amd.parallel import init_distributed
from tepdist_amd.planner import AutoParallel
from tepdist_amd.runtime.planned import PlannedModule
from tepdist_amd.train import Trainer


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="gpt2-117m")
    ap.add_argument("--micro-batch", type=int, default=8)
    ap.add_argument("--seq", type=int, default=512)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--annotate", action="store_true",
                    help="demonstrate user sharding annotations")
    args = ap.parse_args()

    rank, world, local = init_distributed()
    device = torch.device("cuda", local) if torch.cuda.is_available() \
        else torch.device("cpu")
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    cfg = GPT2_CONFIGS[args.model]
    seq = min(args.seq, cfg.n_ctx)
    batch = args.micro_batch * world

    g = gpt2_ir(cfg, batch=batch, seq=seq)
    if args.annotate:
        # pin the embedding table's sharding and keep the final LN gain
        # replicated (the reference's xla_sharding split/replicate)
        split(g.nodes[g.params["wte"]], 0)
        replicate(g.nodes[g.params["lnf_g"]])

    plan = None
    if rank == 0:
        ap = AutoParallel(g, world)
        plan = ap.run()
        if plan.pp > 1:
            # this minimal script shows the flat-mesh module; bench.py's
            # auto path runs pipeline plans through the stage-decomposed
            # task-list executor
            plan = ap._best_over_rounds(1, world)
        print(plan.summary(), flush=True)
    if world > 1:
        import torch.distributed as dist
        obj = [plan]
        dist.broadcast_object_list(obj, src=0)
        plan = obj[0]

    model = PlannedModule(g, plan, device=device, dtype=dtype)
    opt = model.make_optimizer(lr=1e-4)
    trainer = Trainer(model, opt, grad_accum_steps=1,
                      reducer=model.make_reducer())
    gen = torch.Generator().manual_seed(4321)   # identical on every rank
    for step in range(args.steps):
        ids = torch.randint(0, cfg.vocab_size, (batch, seq + 1),
                            generator=gen)
        loss = trainer.train_step(
            lambda i: (ids[:, :-1].to(device), ids[:, 1:].to(device)))
        if rank == 0:
            print(f"step {step} loss {loss:.4f}", flush=True)


if __name__ == "__main__":
    main()
