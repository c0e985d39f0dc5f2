"""Secondary model-family benchmarks on 1 GPU (BASELINE.md target table):
Wide-ResNet bf16 images/s and GPT-MoE tokens/s (the reference's
wide_resnet / gpt_moe examples, fake data, stop_at_step protocol)."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from tepdist_amd.train import AdamW, Trainer


def measure(model, make_batch, steps=8, warmup=3, accum=1):
    opt = AdamW(model.parameters(), lr=1e-4)
    tr = Trainer(model, opt, grad_accum_steps=accum)
    for _ in range(warmup):
        tr.train_step(make_batch)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        loss = tr.train_step(make_batch)
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / steps, loss


def bench_wrn(name="wrn-250m", batch=64, res=224):
    import tepdist_amd.models.configs as C
    from tepdist_amd.models import wide_resnet as W
    cfg = C.WIDE_RESNET_CONFIGS[name]
    m = W.WideResNet(cfg, dtype=torch.bfloat16).cuda()
    x = torch.randn(batch, 3, res, res).bfloat16().cuda()
    y = torch.randint(0, cfg.num_classes, (batch,)).cuda()
    dt, loss = measure(m, lambda i: (x, y))
    print(json.dumps({"bench": "wide_resnet", "config": name,
                      "batch": batch, "res": res, "dtype": "bf16",
                      "ms_per_step": round(dt * 1e3, 2),
                      "images_per_s": round(batch / dt, 1),
                      "final_loss": round(float(loss), 4)}), flush=True)


def bench_moe(batch=8, seq=1024):
    import tepdist_amd.models.configs as C
    from tepdist_amd.models.moe import GPTMoE
    cfg = list(C.MOE_CONFIGS.values())[0]
    m = GPTMoE(cfg, dtype=torch.bfloat16).cuda()
    seq = min(seq, cfg.n_ctx)
    ids = torch.randint(0, cfg.vocab_size, (batch, seq + 1)).cuda()
    dt, loss = measure(m, lambda i: (ids[:, :-1], ids[:, 1:]))
    print(json.dumps({"bench": "gpt_moe", "config": cfg.name,
                      "batch": batch, "seq": seq, "dtype": "bf16",
                      "experts": cfg.num_experts,
                      "ms_per_step": round(dt * 1e3, 2),
                      "tokens_per_s": round(batch * seq / dt, 1),
                      "final_loss": round(float(loss), 4)}), flush=True)


def bench_llama(name="llama-1b", batch=8, seq=2048):
    from tepdist_amd.models.llama import LLAMA_CONFIGS, Llama
    cfg = LLAMA_CONFIGS[name]
    m = Llama(cfg, dtype=torch.bfloat16).cuda()
    seq = min(seq, cfg.n_ctx)
    ids = torch.randint(0, cfg.vocab_size, (batch, seq + 1)).cuda()
    dt, loss = measure(m, lambda i: (ids[:, :-1], ids[:, 1:]))
    print(json.dumps({"bench": "llama", "config": cfg.name,
                      "batch": batch, "seq": seq, "dtype": "bf16",
                      "ms_per_step": round(dt * 1e3, 2),
                      "tokens_per_s": round(batch * seq / dt, 1),
                      "final_loss": round(float(loss), 4)}), flush=True)


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    if which in ("all", "wrn"):
        bench_wrn()
    if which in ("all", "moe"):
        bench_moe()
    if which in ("all", "llama"):
        bench_llama()
