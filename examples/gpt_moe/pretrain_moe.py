"""GPT-MoE expert-parallel pretraining (the reference's
examples/gpt_moe/pretrain_gpt_moe.py counterpart: fake input,
stop_at_step, per-step logging; expert parallelism = planner all-to-all
over xGMI)."""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import torch

from tepdist_amd.models.configs import MOE_CONFIGS
from tepdist_amd.models.moe import GPTMoE
from tepdist_amd.parallel import GradReducer, init_distributed
from tepdist_amd.train import AdamW, Trainer


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", default="gpt-moe-base")
    ap.add_argument("--micro-batch", type=int, default=8)
    ap.add_argument("--seq", type=int, default=1024)
    ap.add_argument("--stop-at-step", type=int, default=10)
    ap.add_argument("--fake-input", action="store_true", default=True)
    args = ap.parse_args()

    rank, world, local = init_distributed()
    dev = torch.device("cuda", local) if torch.cuda.is_available() else "cpu"
    cfg = MOE_CONFIGS[args.config]
    seq = min(args.seq, cfg.n_ctx)
    dtype = torch.bfloat16 if dev != "cpu" else torch.float32
    model = GPTMoE(cfg, dtype=dtype, ep_group=None, ep_size=world,
                   ep_rank=rank).to(dev)
    opt = AdamW(model.parameters(), lr=1e-4)
    # DP over the same ranks as EP: gate/attention grads all-reduce;
    # expert weights are rank-private (EP), excluded from reduction
    dp_params = [p for n, p in model.named_parameters()
                 if ".moe.w1" not in n and ".moe.w2" not in n
                 and ".moe.b1" not in n and ".moe.b2" not in n]
    reducer = GradReducer(dp_params) if world > 1 else None
    trainer = Trainer(model, opt, reducer=reducer)

    g = torch.Generator().manual_seed(7 + rank)
    for step in range(args.stop_at_step):
        ids = torch.randint(0, cfg.vocab_size, (args.micro_batch, seq + 1),
                            generator=g)
        t0 = time.time()
        loss = trainer.train_step(
            lambda i: (ids[:, :-1].to(dev), ids[:, 1:].to(dev)))
        if rank == 0:
            tok = args.micro_batch * seq * world / (time.time() - t0)
            print(f"step {step} loss {loss:.4f} tokens/s {tok:.0f}")


if __name__ == "__main__":
    main()
