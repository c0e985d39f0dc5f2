import sys, os
sys.path.insert(0, "/root/repo")
import torch
from tepdist_amd.ops import hip
torch.manual_seed(0)
B,H,S,D = 4,8,1024,128
q = torch.randn(B,H,S,D).bfloat16().cuda()
k = torch.randn(B,H,S,D).bfloat16().cuda()
v = torch.randn(B,H,S,D).bfloat16().cuda()
out,_ = hip.attention_fwd(q,k,v,causal=True)
ref = torch.nn.functional.scaled_dot_product_attention(q.float(),k.float(),v.float(),is_causal=True)
err = (out.float()-ref).abs()
print("fwd max err", err.max().item())
# backward
dout = torch.randn_like(q)
out2, res = hip.attention_fwd(q,k,v,causal=True)
dq,dk,dv = hip.attention_bwd(dout,q,k,v,res,causal=True)
q2,k2,v2 = (t.float().requires_grad_() for t in (q,k,v))
r2 = torch.nn.functional.scaled_dot_product_attention(q2,k2,v2,is_causal=True)
r2.backward(dout.float())
for name, a, b in (("dq",dq,q2.grad),("dk",dk,k2.grad),("dv",dv,v2.grad)):
    e = (a.float()-b).abs()
    print(name, "max err", e.max().item(), "rel", (e.max()/b.abs().max()).item())
err = err[0,0]
# error by q-row block of 16 and d block of 16
eb = err.reshape(16,64,8,16).amax(dim=(1,3))
for r in eb: print(" ".join(f"{x:.3f}" for x in r))
