import os, sys
sys.path.insert(0, "/root/repo")
import torch, time
from tepdist_amd.ops import hip

def t(fn, iters=20):
    for _ in range(4): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/iters

M = 32768
for name, (m,n,k) in {
    "qkv_fwd": (M,3072,1024), "proj_fwd": (M,1024,1024),
    "fc_fwd": (M,4096,1024), "out_fwd": (M,1024,4096),
    "logits": (M,50432,1024),
    "wgrad_qkv": (3072,1024,M), "wgrad_fc": (4096,1024,M),
    "dgrad_fc": (M,1024,4096),
}.items():
    a = torch.randn(m,k).bfloat16().cuda()
    b = torch.randn(n,k).bfloat16().cuda()
    sec = t(lambda: hip.matmul(a, b.t()))
    print(f"{name:10s} [{m},{n},{k}] {sec*1e6:7.1f}us {2*m*n*k/sec/1e12:7.1f} TF")

# gelu-epilogue shape (fc forward: dual output C + Cpre)
a = torch.randn(M, 1024).bfloat16().cuda()
w = torch.randn(4096, 1024).bfloat16().cuda()
b = torch.zeros(4096).bfloat16().cuda()
sec = t(lambda: hip.linear_fwd(a, w, b, "gelu"))
print(f"fc_gelu    [{M},4096,1024] {sec*1e6:7.1f}us {2*M*4096*1024/sec/1e12:7.1f} TF")
