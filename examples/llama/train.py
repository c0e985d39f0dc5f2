"""Llama-family training example (beyond the reference's model set):
RMSNorm + rotary + SwiGLU decoder with the wide-head (D=128) flash
attention path. Synthetic data, bf16 on GPU."""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import torch

from tepdist_amd.data import SyntheticTokens
from tepdist_amd.models.llama import LLAMA_CONFIGS, Llama
from tepdist_amd.train import AdamW, Trainer


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--config", default="llama-1b")
    ap.add_argument("--micro-batch", type=int, default=8)
    ap.add_argument("--seq", type=int, default=2048)
    ap.add_argument("--stop-at-step", type=int, default=10)
    args = ap.parse_args()

    cfg = LLAMA_CONFIGS[args.config]
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    dtype = torch.bfloat16 if dev == "cuda" else torch.float32
    torch.manual_seed(0)
    model = Llama(cfg, dtype=dtype).to(dev)
    n_params = sum(p.numel() for p in model.parameters())
    print(f"{cfg.name}: {n_params/1e6:.0f}M params, "
          f"head_dim={cfg.n_embd // cfg.n_head}", flush=True)
    opt = AdamW(model.parameters(), lr=3e-4)
    trainer = Trainer(model, opt, grad_accum_steps=1)
    seq = min(args.seq, cfg.n_ctx)
    stream = iter(SyntheticTokens(cfg.vocab_size, args.micro_batch, seq,
                                  seed=1))
    batches = [next(stream) for _ in range(2)]
    toks = args.micro_batch * seq
    for step in range(args.stop_at_step):
        t0 = time.perf_counter()
        x, y = batches[step % 2]
        loss = trainer.train_step(lambda i: (x.to(dev), y.to(dev)))
        if dev == "cuda":
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        print(f"step {step} loss {loss:.4f} tokens/s {toks/dt:,.0f}",
              flush=True)


if __name__ == "__main__":
    main()
