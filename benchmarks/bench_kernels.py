"""Kernel microbenchmarks on MI355X: GEMM TFLOP/s per training shape,
memory-bound kernel bandwidths. Prints one JSON line per measurement.

Run on the GPU box:  python benchmarks/bench_kernels.py
"""

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from tepdist_amd.ops import hip

BF16 = torch.bfloat16


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def bench_gemm():
    shapes = [
        ("square4k", 4096, 4096, 4096, 1),
        ("square8k", 8192, 8192, 8192, 1),
        ("qkv345m", 8192, 3072, 1024, 1),
        ("fc345m", 8192, 4096, 1024, 1),
        ("out345m", 8192, 1024, 4096, 1),
        ("vocab345m", 8192, 50304, 1024, 1),
        ("attn_scores", 1024, 1024, 64, 128),   # B*H=128 heads of S=1024
        ("attn_pv", 1024, 64, 1024, 128),
    ]
    for name, M, N, K, B in shapes:
        a = torch.randn(B, M, K).to(BF16).cuda() if B > 1 else \
            torch.randn(M, K).to(BF16).cuda()
        b = torch.randn(B, K, N).to(BF16).cuda() if B > 1 else \
            torch.randn(K, N).to(BF16).cuda()
        t = timeit(lambda: hip.matmul(a, b))
        tf = 2.0 * B * M * N * K / t / 1e12
        # A/B: rocBLAS via torch for comparison
        t2 = timeit(lambda: a @ b)
        tf2 = 2.0 * B * M * N * K / t2 / 1e12
        print(json.dumps({"kernel": "gemm", "shape": name,
                          "ours_tflops": round(tf, 1),
                          "rocblas_tflops": round(tf2, 1),
                          "ms": round(t * 1e3, 3)}), flush=True)

    # transposed operand cases (dgrad NN / wgrad TN)
    M, N, K = 8192, 1024, 4096
    dy = torch.randn(M, K).to(BF16).cuda()     # acts as [M,K_contr]
    w = torch.randn(K, N).to(BF16).cuda()      # stored [K,N] -> NN
    t = timeit(lambda: hip.matmul(dy, w))
    print(json.dumps({"kernel": "gemm_nn_dgrad", "tflops":
                      round(2.0 * M * N * K / t / 1e12, 1)}), flush=True)
    x = torch.randn(K, M).to(BF16).cuda()
    t = timeit(lambda: hip.matmul(dy.t(), x.t()))  # wgrad-like TN
    print(json.dumps({"kernel": "gemm_tn_wgrad_like", "tflops":
                      round(2.0 * M * N * K / t / 1e12, 1)}), flush=True)


def bench_membound():
    rows, cols = 8192, 1024
    x = torch.randn(rows, cols).to(BF16).cuda()
    g = torch.randn(cols).to(BF16).cuda()
    b = torch.randn(cols).to(BF16).cuda()
    t = timeit(lambda: hip.layernorm_fwd(x, g, b, 1e-5))
    bw = 2.0 * rows * cols * 2 / t / 1e12  # read+write bf16
    print(json.dumps({"kernel": "layernorm_fwd", "tb_s": round(bw, 2),
                      "us": round(t * 1e6, 1)}), flush=True)

    y, mean, rstd = hip.layernorm_fwd(x, g, b, 1e-5)
    dy = torch.randn(rows, cols).to(BF16).cuda()
    t = timeit(lambda: hip.layernorm_bwd(dy, x, g, mean, rstd))
    bw = 5.0 * rows * cols * 2 / t / 1e12  # 2 reads x2 passes + 1 write
    print(json.dumps({"kernel": "layernorm_bwd", "tb_s": round(bw, 2),
                      "us": round(t * 1e6, 1)}), flush=True)

    s = torch.randn(128, 1024, 1024).to(BF16).cuda()
    t = timeit(lambda: hip.softmax_fwd(s, scale=0.125, causal=True))
    bw = 2.0 * s.numel() * 2 / t / 1e12
    print(json.dumps({"kernel": "softmax_causal_fwd", "tb_s": round(bw, 2),
                      "us": round(t * 1e6, 1)}), flush=True)

    n = 1 << 26
    p = torch.randn(n).to(BF16).cuda()
    master = torch.randn(n).cuda()
    gr = torch.randn(n).to(BF16).cuda()
    m = torch.zeros(n).cuda()
    v = torch.zeros(n).cuda()
    t = timeit(lambda: hip.adamw_step(p, master, gr, m, v, lr=1e-3, beta1=0.9,
                                      beta2=0.999, eps=1e-8, weight_decay=0.01,
                                      step=1))
    bw = n * (2 + 4 * 3 + 4 * 3 + 2) / t / 1e12  # rough rw bytes
    print(json.dumps({"kernel": "adamw", "tb_s": round(bw, 2),
                      "us": round(t * 1e6, 1)}), flush=True)

    V = 50304
    logits = torch.randn(8192, V).to(BF16).cuda()
    tg = torch.randint(0, 50257, (8192,)).cuda()
    t = timeit(lambda: hip.cross_entropy_fwd(logits, tg, -1), iters=10)
    bw = 2.0 * 8192 * V * 2 / t / 1e12
    print(json.dumps({"kernel": "cross_entropy_fwd", "tb_s": round(bw, 2),
                      "us": round(t * 1e6, 1)}), flush=True)


def bench_attention():
    B, H, S, D = 16, 16, 1024, 64
    q = torch.randn(B, H, S, D).to(BF16).cuda()
    k = torch.randn(B, H, S, D).to(BF16).cuda()
    v = torch.randn(B, H, S, D).to(BF16).cuda()
    t = timeit(lambda: hip.attention_fwd(q, k, v, True))
    fl = 4.0 * B * H * S * S * D / 2  # causal
    print(json.dumps({"kernel": "flash_fwd", "tflops": round(fl / t / 1e12, 1),
                      "us": round(t * 1e6, 1)}), flush=True)
    out, res = hip.attention_fwd(q, k, v, True)
    dout = torch.randn(B, H, S, D).to(BF16).cuda()
    t = timeit(lambda: hip.attention_bwd(dout, q, k, v, res, True), iters=10)
    print(json.dumps({"kernel": "flash_bwd",
                      "tflops": round(2.5 * fl / t / 1e12, 1),
                      "us": round(t * 1e6, 1)}), flush=True)
    rows, cols = 32768, 4096
    dy = torch.randn(rows, cols).to(BF16).cuda()
    t = timeit(lambda: hip.linear_bwd(dy[:, :1024], torch.randn(rows, 1024).to(BF16).cuda(), torch.randn(1024, 1024).to(BF16).cuda(), True, "none", None), iters=3) if False else None
    db = torch.empty(cols, dtype=BF16, device="cuda")
    ws = torch.zeros(cols, dtype=torch.float32, device="cuda")
    from tepdist_amd.ops import _tepdist_hip as ext
    import torch as _t
    def run_bias():
        ws.zero_()
        ext.bias_sum(dy.data_ptr(), db.data_ptr(), ws.data_ptr(), rows, cols,
                     _t.cuda.current_stream().cuda_stream)
    t = timeit(run_bias)
    print(json.dumps({"kernel": "bias_sum",
                      "tb_s": round(rows * cols * 2 / t / 1e12, 2),
                      "us": round(t * 1e6, 1)}), flush=True)


if __name__ == "__main__":
    torch.manual_seed(0)
    bench_gemm()
    bench_membound()
    bench_attention()
