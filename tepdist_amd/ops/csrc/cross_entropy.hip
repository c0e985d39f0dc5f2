// Fused cross entropy (log-softmax + NLL) over large vocab rows for gfx950.
// fwd: one 256-thread block per row computes logsumexp in a SINGLE online
// pass (flash-style running max/sum rescale — the two-pass version
// re-read each 100 KB row and only partly hit L2), writes nll and lse.
// bwd: elementwise d_logits = (softmax - onehot) * dloss/n, vectorized;
// nontemporal streams (nothing is re-read).

#include <algorithm>
#include <stdexcept>

#include "common.h"
#include "kernels.h"

namespace tepdist {

namespace {

constexpr int NT = 256;

__launch_bounds__(NT) __global__
void ce_fwd_kernel(const bf16_t* __restrict__ logits,
                   const int64_t* __restrict__ targets,
                   float* __restrict__ nll, float* __restrict__ lse_out,
                   int64_t rows, int cols, int ignore_index) {
  __shared__ float scratch[NT / WAVE];
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const bf16_t* lr = logits + row * cols;
    // online logsumexp: one pass, per-thread running (max, sum)
    float m = -3.0e38f;
    float s = 0.f;
    for (int c0 = threadIdx.x * 8; c0 < cols; c0 += NT * 8) {
      bf16x8 v;
      if (c0 + 8 <= cols) {
        v = __builtin_nontemporal_load(
            reinterpret_cast<const bf16x8*>(lr + c0));
      } else {
        for (int e = 0; e < 8 && c0 + e < cols; ++e) v[e] = lr[c0 + e];
        for (int e = cols - c0; e < 8; ++e) v[e] = f2bf(-3.0e38f);
      }
      float cm = -3.0e38f;
#pragma unroll
      for (int e = 0; e < 8; ++e) cm = fmaxf(cm, bf2f(v[e]));
      const float mn = fmaxf(m, cm);
      float cs = 0.f;
#pragma unroll
      for (int e = 0; e < 8; ++e) cs += __expf(bf2f(v[e]) - mn);
      s = s * __expf(m - mn) + cs;
      m = mn;
    }
    // combine per-thread (m, s) pairs: block max, then rescaled sums
    const float bm = block_allreduce(m, scratch, NT / WAVE,
                                     [](float a, float b) {
                                       return fmaxf(a, b);
                                     }, -3.0e38f);
    __syncthreads();
    s = (m > -3.0e38f) ? s * __expf(m - bm) : 0.f;
    s = block_allreduce(s, scratch, NT / WAVE,
                        [](float a, float b) { return a + b; }, 0.f);
    const float m2 = bm;
    if (threadIdx.x == 0) {
      const float lse = m2 + __logf(s);
      lse_out[row] = lse;
      const int64_t t = targets[row];
      // t < 0: either the ignore index or a "no target in this vocab
      // shard" sentinel (vocab-parallel CE) -> no target-logit read
      nll[row] = (t < 0) ? 0.f : lse - bf2f(lr[t]);
    }
    __syncthreads();
  }
}

__launch_bounds__(NT) __global__
void ce_bwd_kernel(const bf16_t* __restrict__ logits,
                   const int64_t* __restrict__ targets,
                   const float* __restrict__ lse,
                   const float* __restrict__ dscale,  // device: dloss / n
                   bf16_t* __restrict__ dlogits, int64_t rows, int cols,
                   int ignore_index) {
  // read from device memory so the launch is hipGraph-capturable (a host
  // scalar argument would force a sync to compute it)
  const float dloss_over_n = *dscale;
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const bf16_t* lr = logits + row * cols;
    bf16_t* dr = dlogits + row * cols;
    const int64_t t = targets[row];  // t==ignore: zero row; other t<0
    // (vocab-parallel "target not in shard"): softmax grad, no onehot
    if (t == ignore_index) {
      for (int c0 = threadIdx.x * 8; c0 < cols; c0 += NT * 8) {
        bf16x8 z = {};
        if (c0 + 8 <= cols)
          *reinterpret_cast<bf16x8*>(dr + c0) = z;
        else
          for (int e = 0; e < 8 && c0 + e < cols; ++e) dr[c0 + e] = z[e];
      }
      continue;
    }
    const float l = lse[row];
    for (int c0 = threadIdx.x * 8; c0 < cols; c0 += NT * 8) {
      bf16x8 v = {}, o;
      if (c0 + 8 <= cols) {
        v = __builtin_nontemporal_load(
            reinterpret_cast<const bf16x8*>(lr + c0));
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          float p = __expf(bf2f(v[e]) - l);
          if (c0 + e == t) p -= 1.f;
          o[e] = f2bf(p * dloss_over_n);
        }
        __builtin_nontemporal_store(o, reinterpret_cast<bf16x8*>(dr + c0));
      } else {
        for (int e = 0; e < 8 && c0 + e < cols; ++e) {
          float p = __expf(bf2f(lr[c0 + e]) - l);
          if (c0 + e == t) p -= 1.f;
          dr[c0 + e] = f2bf(p * dloss_over_n);
        }
      }
    }
  }
}

}  // namespace

void cross_entropy_fwd_bf16(const void* logits, const int64_t* targets,
                            float* nll, float* lse, int64_t rows, int cols,
                            int ignore_index, hipStream_t stream) {
  const int blocks = (int)std::min<int64_t>(rows, 2048);
  hipLaunchKernelGGL(ce_fwd_kernel, dim3(blocks), dim3(NT), 0, stream,
                     static_cast<const bf16_t*>(logits), targets, nll, lse,
                     rows, cols, ignore_index);
}

void cross_entropy_bwd_bf16(const void* logits, const int64_t* targets,
                            const float* lse, const float* dscale,
                            void* dlogits, int64_t rows, int cols,
                            int ignore_index, hipStream_t stream) {
  const int blocks = (int)std::min<int64_t>(rows, 2048);
  hipLaunchKernelGGL(ce_bwd_kernel, dim3(blocks), dim3(NT), 0, stream,
                     static_cast<const bf16_t*>(logits), targets, lse,
                     dscale, static_cast<bf16_t*>(dlogits), rows, cols,
                     ignore_index);
}

}  // namespace tepdist
