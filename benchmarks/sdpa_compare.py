"""Times torch's scaled_dot_product_attention (fwd and fwd+bwd) against
our flash kernels on the bench shape."""
import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torch.nn.functional as F
from tepdist_amd.ops import hip

B, H, S, D = 16, 16, 1024, 64

def bench(fn, iters=20):
    for _ in range(5): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters

q = torch.randn(B, H, S, D, dtype=torch.bfloat16, device="cuda")
k, v = torch.randn_like(q), torch.randn_like(q)
dout = torch.randn_like(q)

print(f"sdpa fwd      : {bench(lambda: F.scaled_dot_product_attention(q, k, v, is_causal=True))*1e6:8.1f} us")
qg = q.clone().requires_grad_(); kg = k.clone().requires_grad_(); vg = v.clone().requires_grad_()
def fb():
    o = F.scaled_dot_product_attention(qg, kg, vg, is_causal=True)
    o.backward(dout)
    qg.grad = kg.grad = vg.grad = None
print(f"sdpa fwd+bwd  : {bench(fb)*1e6:8.1f} us")

out, res = hip.attention_fwd(q, k, v, True)
print(f"ours fwd      : {bench(lambda: hip.attention_fwd(q, k, v, True))*1e6:8.1f} us")
def ob():
    hip.attention_bwd(dout, q, k, v, res, True)
print(f"ours bwd only : {bench(ob)*1e6:8.1f} us")
