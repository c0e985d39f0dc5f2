import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from tepdist_amd.ops import hip
torch.manual_seed(0)
for (M, N, K) in [(768, 768, 2048), (2304, 768, 2048), (768, 2304, 2048)]:
    a = torch.randn(M, K).bfloat16().cuda()
    b = torch.randn(N, K).bfloat16().cuda()
    y = hip.matmul(a, b.t())
    torch.cuda.synchronize()
    ref = a.float() @ b.float().t()
    err = (y.float() - ref).abs().max().item()
    print(f"[{M},{N},{K}] max abs err {err:.4f}", flush=True)
print("OK", flush=True)
