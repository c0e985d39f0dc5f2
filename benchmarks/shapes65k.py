"""GEMM + flash shapes of the flagship step (gpt2-345m, batch 64, seq
1024 -> M = 65536) vs torch/rocBLAS: locates per-shape headroom.

Run on the GPU box: python benchmarks/shapes65k.py
"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from tepdist_amd.ops import hip

BF16 = torch.bfloat16


def timeit(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    M = 65536
    # (name, N, K): fwd y[M,N] = x[M,K] @ w[N,K]^T
    fwd = [("qkv", 3072, 1024), ("proj", 1024, 1024),
           ("fc", 4096, 1024), ("out", 1024, 4096),
           ("logits", 50432, 1024)]
    for name, N, K in fwd:
        x = torch.randn(M, K).to(BF16).cuda()
        w = torch.randn(N, K).to(BF16).cuda()
        t = timeit(lambda: hip.linear_fwd(x, w, None, "none"))
        tf = 2.0 * M * N * K / t / 1e12
        t2 = timeit(lambda: x @ w.t())
        print(json.dumps({"op": f"fwd_{name}", "ms": round(t * 1e3, 3),
                          "tflops": round(tf, 1),
                          "rocblas_tflops":
                          round(2.0 * M * N * K / t2 / 1e12, 1)}), flush=True)
        del x, w
        # dgrad: dx[M,K] = dy[M,N] @ w[N,K];  wgrad: dw[N,K] = dy^T @ x
        dy = torch.randn(M, N).to(BF16).cuda()
        x = torch.randn(M, K).to(BF16).cuda()
        w = torch.randn(N, K).to(BF16).cuda()
        t = timeit(lambda: hip.linear_bwd(dy, x, w, False, "none", None))
        tf = 2.0 * 2 * M * N * K / t / 1e12   # two GEMMs
        t2 = timeit(lambda: (dy @ w, dy.t() @ x))
        print(json.dumps({"op": f"bwd_{name}", "ms": round(t * 1e3, 3),
                          "tflops_2gemm": round(tf, 1),
                          "rocblas_tflops":
                          round(2.0 * 2 * M * N * K / t2 / 1e12, 1)}),
              flush=True)
        del x, w, dy
        torch.cuda.empty_cache()

    # flash attention at bench shape
    B, H, S, D = 64, 16, 1024, 64
    qkv = torch.randn(B, S, 3 * H * D).to(BF16).cuda()
    from tepdist_amd.ops import attention_qkv
    t = timeit(lambda: attention_qkv(qkv, H, causal=True))
    fl = 2.0 * B * H * S * S * D * 2 / 2   # causal halves
    print(json.dumps({"op": "flash_fwd", "ms": round(t * 1e3, 3),
                      "tflops": round(fl / t / 1e12, 1)}), flush=True)
    qkv.requires_grad_()
    y = attention_qkv(qkv, H, causal=True)
    g = torch.randn_like(y)
    t = timeit(lambda: torch.autograd.grad(y, qkv, g, retain_graph=True),
               iters=5)
    print(json.dumps({"op": "flash_bwd", "ms": round(t * 1e3, 3),
                      "tflops": round(2.5 * fl / t / 1e12, 1)}), flush=True)


if __name__ == "__main__":
    main()
