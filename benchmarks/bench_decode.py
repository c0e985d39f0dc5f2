"""Serving-path benchmark: prefill + decode throughput of the KV-cache
generation engine (inference/engine.py) on one GPU. Prints one JSON line
per config: prefill tokens/s and steady-state decode tokens/s."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from tepdist_amd.inference.engine import Generator
from tepdist_amd.models import GPT2, GPT2_CONFIGS


def main(name="gpt2-345m", batch=32, prompt=512, new=128):
    cfg = GPT2_CONFIGS[name]
    torch.manual_seed(0)
    m = GPT2(cfg, dtype=torch.bfloat16).cuda()
    m.reset_parameters()
    m.eval()
    g = Generator(m)
    ids = torch.randint(0, cfg.vocab_size, (batch, prompt)).cuda()
    with torch.no_grad():
        g.generate(ids, 8)                      # warmup (autotune etc.)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        out = g.generate(ids, new)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        # split prefill vs decode: time a prefill-only call (1 new token)
        torch.cuda.synchronize()
        t1 = time.perf_counter()
        g.generate(ids, 1)
        torch.cuda.synchronize()
        pre = time.perf_counter() - t1
    n_new = int(out.shape[1] - prompt)
    dec = (batch * (n_new - 1)) / max(dt - pre, 1e-9)
    print(json.dumps({
        "bench": "decode", "config": name, "batch": batch,
        "prompt": prompt, "new_tokens": n_new, "dtype": "bf16",
        "prefill_tokens_per_s": round(batch * prompt / pre, 1),
        "decode_tokens_per_s": round(dec, 1),
        "total_s": round(dt, 3)}), flush=True)


if __name__ == "__main__":
    main(*(sys.argv[1:2] or ["gpt2-345m"]))
