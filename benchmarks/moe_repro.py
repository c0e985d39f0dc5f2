import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import dataclasses
import torch
from tepdist_amd.models.configs import MOE_CONFIGS
from tepdist_amd.models.moe import GPTMoE
from tepdist_amd import ops
from tepdist_amd.train.optim import AdamW

cfg = dataclasses.replace(MOE_CONFIGS["gpt-moe-base"], n_layer=2)
torch.manual_seed(0)
model = GPTMoE(cfg, dtype=torch.bfloat16).cuda()
opt = AdamW(model.parameters(), lr=1e-4)
ids = torch.randint(0, cfg.vocab_size, (4, 513), device="cuda")
loss = model(ids[:, :-1], labels=ids[:, 1:])
loss.backward()
torch.cuda.synchronize()
print("bwd ok", flush=True)
named = dict(model.named_parameters())
for n, p in named.items():
    if p.grad is None:
        continue
    st = opt.state[p]
    print(f"{n} shape={tuple(p.shape)} pdtype={p.dtype} gdtype={p.grad.dtype} "
          f"gcontig={p.grad.is_contiguous()}", flush=True)
    ops.adamw_step(p.data, st["master"], p.grad, st["exp_avg"],
                   st["exp_avg_sq"], lr=1e-4, step=1)
    torch.cuda.synchronize()
print("ALL OK", flush=True)
