// Fused scale + causal-mask + row softmax (fwd) and softmax backward for
// gfx950. One wave per row, bf16x8 loads, fp32 math, online max/sum in
// registers. Rows up to 2048 cols keep values in registers (NV dispatch);
// longer rows re-read from L2.

#include <algorithm>
#include <stdexcept>

#include "common.h"
#include "kernels.h"

namespace tepdist {

namespace {

constexpr int NT = 256;

template <int NV>
__launch_bounds__(NT) __global__
void softmax_fwd_kernel(const bf16_t* __restrict__ x, bf16_t* __restrict__ p,
                        int64_t rows, int cols, int sq, float scale,
                        bool causal) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int64_t nwaves = (int64_t)gridDim.x * (NT / WAVE);

  for (int64_t row = (int64_t)blockIdx.x * (NT / WAVE) + wid; row < rows;
       row += nwaves) {
    const bf16_t* xr = x + row * cols;
    // causal: query position q allows keys <= q + (cols - sq)
    const int limit = causal ? (int)(row % sq) + (cols - sq) : cols - 1;
    float vals[NV * 8];
    float m = -3.0e38f;
#pragma unroll
    for (int v = 0; v < NV; ++v) {
      const int c0 = v * WAVE * 8 + lane * 8;
      bf16x8 xv = {};
      if (c0 + 8 <= cols) {
        xv = *reinterpret_cast<const bf16x8*>(xr + c0);
      } else {
        for (int e = 0; e < 8 && c0 + e < cols; ++e) xv[e] = xr[c0 + e];
      }
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int c = c0 + e;
        const float f =
            (c <= limit && c < cols) ? bf2f(xv[e]) * scale : -3.0e38f;
        vals[v * 8 + e] = f;
        m = fmaxf(m, f);
      }
    }
    m = wave_allreduce_max(m);
    float s = 0.f;
#pragma unroll
    for (int v = 0; v < NV; ++v)
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float ex =
            (vals[v * 8 + e] > -1.0e38f) ? __expf(vals[v * 8 + e] - m) : 0.f;
        vals[v * 8 + e] = ex;
        s += ex;
      }
    s = wave_allreduce_sum(s);
    const float inv = 1.0f / s;
    bf16_t* pr = p + row * cols;
#pragma unroll
    for (int v = 0; v < NV; ++v) {
      const int c0 = v * WAVE * 8 + lane * 8;
      if (c0 >= cols) continue;
      bf16x8 pv;
#pragma unroll
      for (int e = 0; e < 8; ++e) pv[e] = f2bf(vals[v * 8 + e] * inv);
      if (c0 + 8 <= cols) {
        *reinterpret_cast<bf16x8*>(pr + c0) = pv;
      } else {
        for (int e = 0; e < 8 && c0 + e < cols; ++e) pr[c0 + e] = pv[e];
      }
    }
  }
}

template <int NV>
__launch_bounds__(NT) __global__
void softmax_bwd_kernel(const bf16_t* __restrict__ dp,
                        const bf16_t* __restrict__ p, bf16_t* __restrict__ ds,
                        int64_t rows, int cols, float scale) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int64_t nwaves = (int64_t)gridDim.x * (NT / WAVE);

  for (int64_t row = (int64_t)blockIdx.x * (NT / WAVE) + wid; row < rows;
       row += nwaves) {
    const bf16_t* dpr = dp + row * cols;
    const bf16_t* pr = p + row * cols;
    float pv[NV * 8], dv[NV * 8];
    float dot = 0.f;
#pragma unroll
    for (int v = 0; v < NV; ++v) {
      const int c0 = v * WAVE * 8 + lane * 8;
      bf16x8 dpv = {}, ppv = {};
      if (c0 + 8 <= cols) {
        dpv = *reinterpret_cast<const bf16x8*>(dpr + c0);
        ppv = *reinterpret_cast<const bf16x8*>(pr + c0);
      } else {
        for (int e = 0; e < 8 && c0 + e < cols; ++e) {
          dpv[e] = dpr[c0 + e];
          ppv[e] = pr[c0 + e];
        }
      }
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int c = c0 + e;
        const float pe = (c < cols) ? bf2f(ppv[e]) : 0.f;
        const float de = (c < cols) ? bf2f(dpv[e]) : 0.f;
        pv[v * 8 + e] = pe;
        dv[v * 8 + e] = de;
        dot += pe * de;
      }
    }
    dot = wave_allreduce_sum(dot);
    bf16_t* dsr = ds + row * cols;
#pragma unroll
    for (int v = 0; v < NV; ++v) {
      const int c0 = v * WAVE * 8 + lane * 8;
      if (c0 >= cols) continue;
      bf16x8 out;
#pragma unroll
      for (int e = 0; e < 8; ++e)
        out[e] = f2bf(scale * pv[v * 8 + e] * (dv[v * 8 + e] - dot));
      if (c0 + 8 <= cols) {
        *reinterpret_cast<bf16x8*>(dsr + c0) = out;
      } else {
        for (int e = 0; e < 8 && c0 + e < cols; ++e) dsr[c0 + e] = out[e];
      }
    }
  }
}

int nv_for(int cols) { return (cols + WAVE * 8 - 1) / (WAVE * 8); }

}  // namespace

void softmax_fwd_bf16(const void* x, void* p, int64_t rows, int cols, int sq,
                      float scale, bool causal, hipStream_t stream) {
  const int blocks =
      (int)std::min<int64_t>((rows + 3) / 4, 2048);
  const dim3 g(blocks), blk(NT);
  const bf16_t* xp = static_cast<const bf16_t*>(x);
  bf16_t* pp = static_cast<bf16_t*>(p);
#define SM_FWD(NV)                                                        \
  hipLaunchKernelGGL(softmax_fwd_kernel<NV>, g, blk, 0, stream, xp, pp,  \
                     rows, cols, sq, scale, causal)
  switch (nv_for(cols)) {
    case 1: SM_FWD(1); break;
    case 2: SM_FWD(2); break;
    case 3: SM_FWD(3); break;
    case 4: SM_FWD(4); break;
    default: throw std::runtime_error("softmax: cols > 2048 unsupported");
  }
#undef SM_FWD
}

void softmax_bwd_bf16(const void* dp, const void* p, void* ds, int64_t rows,
                      int cols, float scale, hipStream_t stream) {
  const int blocks =
      (int)std::min<int64_t>((rows + 3) / 4, 2048);
  const dim3 g(blocks), blk(NT);
  const bf16_t* dpp = static_cast<const bf16_t*>(dp);
  const bf16_t* pp = static_cast<const bf16_t*>(p);
  bf16_t* dsp = static_cast<bf16_t*>(ds);
#define SM_BWD(NV)                                                         \
  hipLaunchKernelGGL(softmax_bwd_kernel<NV>, g, blk, 0, stream, dpp, pp,  \
                     dsp, rows, cols, scale)
  switch (nv_for(cols)) {
    case 1: SM_BWD(1); break;
    case 2: SM_BWD(2); break;
    case 3: SM_BWD(3); break;
    case 4: SM_BWD(4); break;
    default: throw std::runtime_error("softmax: cols > 2048 unsupported");
  }
#undef SM_BWD
}

}  // namespace tepdist
