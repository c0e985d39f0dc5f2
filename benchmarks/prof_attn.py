import os, sys, time, json
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from tepdist_amd.ops import hip
which = sys.argv[1] if len(sys.argv) > 1 else "fwd"
B, H, S, D = 16, 16, 1024, 64
q = torch.randn(B, H, S, D).bfloat16().cuda()
k = torch.randn(B, H, S, D).bfloat16().cuda()
v = torch.randn(B, H, S, D).bfloat16().cuda()
out, res = hip.attention_fwd(q, k, v, True)
dout = torch.randn(B, H, S, D).bfloat16().cuda()
for _ in range(10):
    if which == "fwd":
        hip.attention_fwd(q, k, v, True)
    else:
        hip.attention_bwd(dout, q, k, v, res, True)
torch.cuda.synchronize()
print("done", which)
