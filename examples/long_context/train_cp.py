"""Long-context training with context parallelism (ring attention).

Shards the SEQUENCE across ranks: each GPU holds S/world tokens, K/V
blocks rotate between xGMI ring neighbors inside attention, and every
other op runs token-local. Activation memory falls by the CP degree, so
context length scales with the number of GPUs:

    python -m torch.distributed.run --nproc-per-node 8 \
        --master-addr 127.0.0.1 examples/long_context/train_cp.py \
        --model gpt2-345m --seq 65536 --zigzag

Single process runs as-is (world 1 = plain flash attention).
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import torch

from tepdist_amd.models import GPT2_CONFIGS
from tepdist_amd.parallel import init_distributed
from tepdist_amd.parallel.cp import ContextParallelGPT2
from tepdist_amd.parallel.dp import GradReducer
from tepdist_amd.train.optim import AdamW


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="gpt2-345m")
    ap.add_argument("--batch", type=int, default=1)
    ap.add_argument("--seq", type=int, default=8192)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--zigzag", action="store_true",
                    help="causal load-balanced chunk layout")
    args = ap.parse_args()

    rank, world, local = init_distributed()
    dev = f"cuda:{local}" if torch.cuda.is_available() else "cpu"
    dtype = torch.bfloat16 if dev.startswith("cuda") else torch.float32

    import dataclasses
    cfg = dataclasses.replace(GPT2_CONFIGS[args.model], n_ctx=args.seq)
    torch.manual_seed(0)  # replicated params: same init on every rank
    model = ContextParallelGPT2(cfg, zigzag=args.zigzag,
                                dtype=dtype).to(dev)
    model.reset_parameters()
    reducer = GradReducer(model.parameters(), average=False) \
        if world > 1 else None
    opt = AdamW(model.parameters(), lr=1e-4)

    g = torch.Generator().manual_seed(1)  # identical batch on all ranks;
    # shard_inputs takes this rank's sequence slice
    for step in range(args.steps):
        ids = torch.randint(0, cfg.vocab_size,
                            (args.batch, args.seq + 1), generator=g)
        inp = ids[:, :-1].to(dev)
        lab = ids[:, 1:].contiguous().to(dev)
        li, ll, pos = model.shard_inputs(inp, lab)
        t0 = time.perf_counter()
        opt.zero_grad()
        if reducer:
            reducer.reset()
            reducer.arm()
        loss = model(li, ll, pos=pos)
        loss.backward()
        if reducer:
            reducer.finalize()
        opt.step()
        if dev.startswith("cuda"):
            torch.cuda.synchronize()
        if rank == 0:
            dt = time.perf_counter() - t0
            tok = args.batch * args.seq
            print(f"step {step} loss {loss.item():.4f} "
                  f"{tok / dt:,.0f} tokens/s ({dt * 1e3:.1f} ms)",
                  flush=True)


if __name__ == "__main__":
    main()
