"""Bisect hipGraph capture of the MoE dispatch: capture increasing
prefixes of the forward and report the last stage that instantiates."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from tepdist_amd import ops

torch.manual_seed(0)
dev = "cuda"
T, d, E, k, C = 2048, 256, 8, 2, 640
xt = torch.randn(T, d, device=dev).bfloat16()
w_gate = torch.randn(E, d, device=dev).bfloat16() * 0.02
w1 = torch.randn(E, 4 * d, d, device=dev).bfloat16() * 0.02
b1 = torch.zeros(E, 4 * d, device=dev).bfloat16()
w2 = torch.randn(E, d, 4 * d, device=dev).bfloat16() * 0.02
b2 = torch.zeros(E, d, device=dev).bfloat16()


def stage(n):
    logits = ops.linear(xt, w_gate)
    gates = ops.softmax(logits.unsqueeze(0)).squeeze(0)
    topv, topi = torch.topk(gates.float(), k, dim=-1)
    topv = topv / topv.sum(-1, keepdim=True).clamp_min(1e-9)
    if n == 1:
        return topv.sum()
    flat_e = topi.reshape(-1)
    flat_w = topv.reshape(-1)
    flat_t = torch.arange(T, device=dev).repeat_interleave(k)
    order = torch.argsort(flat_e, stable=True)
    counts = torch.zeros(E, dtype=torch.long, device=dev).scatter_add_(
        0, flat_e, torch.ones_like(flat_e))
    offs = torch.cumsum(counts, 0) - counts
    r = torch.arange(flat_e.numel(), device=dev)
    pos = torch.empty_like(r)
    pos[order] = r - offs[flat_e[order]]
    keep = pos < C
    slot = flat_e * C + pos
    slot_safe = torch.where(keep, slot, torch.zeros_like(slot))
    if n == 2:
        return slot_safe.float().sum() + flat_w.sum() + flat_t.float().sum()
    contrib = xt[flat_t] * keep.unsqueeze(-1).to(xt.dtype)
    D = torch.zeros(E * C, d, dtype=xt.dtype, device=dev)
    D = D.index_put((slot_safe,), contrib, accumulate=True)
    if n == 3:
        return D.float().sum()
    recv = D.reshape(1, E, C, d)
    ys = []
    for e in range(E):
        xe = recv[:, e].reshape(-1, d).contiguous()
        h = ops.linear(xe, w1[e], b1[e], act="gelu")
        ys.append(ops.linear(h, w2[e], b2[e]))
    Y = torch.stack(ys, dim=0)
    back = Y.reshape(E * C, d)
    if n == 4:
        return back.float().sum()
    gathered = back[slot_safe] * \
        (flat_w * keep.to(flat_w.dtype)).unsqueeze(-1).to(back.dtype)
    out = torch.zeros_like(xt).index_add(0, flat_t, gathered.to(xt.dtype))
    return out.float().sum()


for n in range(1, 6):
    for _ in range(2):
        stage(n)   # warm eager
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    try:
        with torch.cuda.graph(g):
            val = stage(n)
        g.replay()
        torch.cuda.synchronize()
        print(f"stage {n}: captured + replayed ok", flush=True)
    except Exception as e:
        print(f"stage {n}: capture FAILED: {e}", flush=True)
        break
print("done", flush=True)


# ---- backward + full-module stages ----------------------------------------
def run_case(name, fn):
    for _ in range(2):
        fn()
    torch.cuda.synchronize()
    g = torch.cuda.CUDAGraph()
    try:
        with torch.cuda.graph(g):
            fn()
        g.replay()
        torch.cuda.synchronize()
        print(f"{name}: captured + replayed ok", flush=True)
    except Exception as e:
        print(f"{name}: capture FAILED: {type(e).__name__} {e}", flush=True)


for t in (w_gate, w1, b1, w2, b2):
    t.requires_grad_()


def fwd_bwd():
    for t in (w_gate, w1, b1, w2, b2):
        t.grad = None
    loss = stage(5)
    loss.backward()


run_case("stage6 fwd+bwd dispatch", fwd_bwd)

from tepdist_amd.models.configs import MOE_CONFIGS
from tepdist_amd.models.moe import GPTMoE, MoELayer

layer = MoELayer(256, 8, 2, 1.25, dtype=torch.bfloat16).cuda()
xin = torch.randn(2, 128, 256, device=dev).bfloat16()


def layer_fb():
    for p in layer.parameters():
        p.grad = None
    y = layer(xin)
    (y.float().sum() + layer.aux_loss).backward()
    layer.aux_loss = layer.aux_loss.detach()   # else the stale graph's
    # AccumulateGrad nodes poison the NEXT iteration's capture


run_case("stage7 MoELayer fwd+bwd", layer_fb)

cfg = list(MOE_CONFIGS.values())[0]
model = GPTMoE(cfg, dtype=torch.bfloat16).cuda()
ids = torch.randint(0, cfg.vocab_size, (2, 129), device=dev)


def model_fb():
    for p in model.parameters():
        p.grad = None
    loss = model(ids[:, :-1], labels=ids[:, 1:])
    loss.backward()


run_case("stage8 GPTMoE fwd+bwd", model_fb)

from tepdist_amd.train.optim import AdamW
from tepdist_amd.train.trainer import Trainer

opt = AdamW(model.parameters(), lr=1e-4)
tr = Trainer(model, opt, grad_accum_steps=1)


def bi(i):
    return ids[:, :-1], ids[:, 1:]


print("stage9 trainer:", flush=True)
for s in range(3):
    loss = tr.train_step(bi)
    print(f"  step {s} loss {loss:.4f} graph={'yes' if tr._graph else 'no'}",
          flush=True)
print("done2", flush=True)
