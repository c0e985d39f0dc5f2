"""Focused single-shape GEMM runner for rocprof PMC passes."""
import os, sys, time, json
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from tepdist_amd.ops import hip

def main():
    case = sys.argv[1] if len(sys.argv) > 1 else "wgrad"
    iters = int(sys.argv[2]) if len(sys.argv) > 2 else 30
    torch.manual_seed(0)
    if case == "wgrad":            # A KO, B KO-ish mixed (dy^T @ x)
        dy = torch.randn(8192, 4096).bfloat16().cuda()
        x = torch.randn(8192, 1024).bfloat16().cuda()
        fn = lambda: hip.matmul(dy.t(), x)   # [4096,1024], K=8192: A KO, B KO
        flops = 2 * 4096 * 1024 * 8192
    elif case == "dgrad":
        dy = torch.randn(8192, 4096).bfloat16().cuda()
        w = torch.randn(4096, 1024).bfloat16().cuda()
        fn = lambda: hip.matmul(dy, w)       # A KC, B KO
        flops = 2 * 8192 * 4096 * 1024
    elif case == "nt":
        a = torch.randn(8192, 1024).bfloat16().cuda()
        w = torch.randn(4096, 1024).bfloat16().cuda()
        fn = lambda: hip.linear_fwd(a, w, None, "none")
        flops = 2 * 8192 * 4096 * 1024
    elif case == "sq8k":
        a = torch.randn(8192, 8192).bfloat16().cuda()
        b = torch.randn(8192, 8192).bfloat16().cuda()
        fn = lambda: hip.matmul(a, b.t())
        flops = 2 * 8192**3
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / iters
    print(json.dumps({"case": case, "ms": round(dt*1e3, 3),
                      "tflops": round(flops/dt/1e12, 1)}))

main()
