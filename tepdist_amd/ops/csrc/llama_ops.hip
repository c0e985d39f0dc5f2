// Llama-family ops for gfx950: RMSNorm (fwd/bwd), rotary position
// embedding (fwd/bwd, rotate-half, angles computed in-kernel), fused
// SwiGLU (fwd/bwd). All memory-bound elementwise/row kernels following
// the LayerNorm kernel's wave-per-row NV-template pattern.

#include <algorithm>
#include <stdexcept>

#include "common.h"
#include "kernels.h"

namespace tepdist {

namespace {

constexpr int NT = 256;

inline int nv_for(int cols) { return (cols + WAVE * 8 - 1) / (WAVE * 8); }

// ---------------------------------------------------------------------------
// RMSNorm
// ---------------------------------------------------------------------------

template <int NV>
__launch_bounds__(NT) __global__
void rms_fwd_kernel(const bf16_t* __restrict__ x,
                    const bf16_t* __restrict__ g, bf16_t* __restrict__ y,
                    float* __restrict__ rstd, int64_t rows, int cols,
                    float eps) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int64_t nw = (int64_t)gridDim.x * (NT / WAVE);

  bf16x8 gv[NV];
#pragma unroll
  for (int v = 0; v < NV; ++v) {
    const int c0 = v * WAVE * 8 + lane * 8;
    bf16x8 t = {};
    if (c0 + 8 <= cols) {
      t = *reinterpret_cast<const bf16x8*>(g + c0);
    } else {
      for (int e = 0; e < 8 && c0 + e < cols; ++e) t[e] = g[c0 + e];
    }
    gv[v] = t;
  }

  for (int64_t row = (int64_t)blockIdx.x * (NT / WAVE) + wid; row < rows;
       row += nw) {
    const bf16_t* xr = x + row * cols;
    bf16x8 xv[NV];
    float ss = 0.f;
#pragma unroll
    for (int v = 0; v < NV; ++v) {
      const int c0 = v * WAVE * 8 + lane * 8;
      bf16x8 t = {};
      if (c0 + 8 <= cols) {
        t = *reinterpret_cast<const bf16x8*>(xr + c0);
      } else {
        for (int e = 0; e < 8 && c0 + e < cols; ++e) t[e] = xr[c0 + e];
      }
      xv[v] = t;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float f = bf2f(t[e]);
        ss += f * f;
      }
    }
    ss = wave_allreduce_sum(ss);
    const float rs = rsqrtf(ss / cols + eps);
    if (lane == 0) rstd[row] = rs;
    bf16_t* yr = y + row * cols;
#pragma unroll
    for (int v = 0; v < NV; ++v) {
      const int c0 = v * WAVE * 8 + lane * 8;
      if (c0 >= cols) continue;
      bf16x8 o;
#pragma unroll
      for (int e = 0; e < 8; ++e)
        o[e] = f2bf(bf2f(xv[v][e]) * rs * bf2f(gv[v][e]));
      if (c0 + 8 <= cols) {
        *reinterpret_cast<bf16x8*>(yr + c0) = o;
      } else {
        for (int e = 0; e < 8 && c0 + e < cols; ++e) yr[c0 + e] = o[e];
      }
    }
  }
}

template <int NV>
__launch_bounds__(NT) __global__
void rms_bwd_kernel(const bf16_t* __restrict__ dy,
                    const bf16_t* __restrict__ x,
                    const bf16_t* __restrict__ g,
                    const float* __restrict__ rstd, bf16_t* __restrict__ dx,
                    float* __restrict__ dg_part, int64_t rows, int cols) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int gwave = blockIdx.x * (NT / WAVE) + wid;
  const int64_t nw = (int64_t)gridDim.x * (NT / WAVE);

  float dg_acc[NV * 8] = {};
  bf16x8 gv[NV];
#pragma unroll
  for (int v = 0; v < NV; ++v) {
    const int c0 = v * WAVE * 8 + lane * 8;
    bf16x8 t = {};
    if (c0 + 8 <= cols) {
      t = *reinterpret_cast<const bf16x8*>(g + c0);
    } else {
      for (int e = 0; e < 8 && c0 + e < cols; ++e) t[e] = g[c0 + e];
    }
    gv[v] = t;
  }

  for (int64_t row = gwave; row < rows; row += nw) {
    const bf16_t* dyr = dy + row * cols;
    const bf16_t* xr = x + row * cols;
    const float rs = rstd[row];
    bf16x8 dyv[NV], xv[NV];
    float c1 = 0.f;
#pragma unroll
    for (int v = 0; v < NV; ++v) {
      const int c0 = v * WAVE * 8 + lane * 8;
      bf16x8 a = {}, b = {};
      if (c0 + 8 <= cols) {
        a = *reinterpret_cast<const bf16x8*>(dyr + c0);
        b = *reinterpret_cast<const bf16x8*>(xr + c0);
      } else {
        for (int e = 0; e < 8 && c0 + e < cols; ++e) {
          a[e] = dyr[c0 + e];
          b[e] = xr[c0 + e];
        }
      }
      dyv[v] = a;
      xv[v] = b;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int c = c0 + e;
        const float xh = (c < cols) ? bf2f(b[e]) * rs : 0.f;
        const float t = (c < cols) ? bf2f(a[e]) * bf2f(gv[v][e]) : 0.f;
        dg_acc[v * 8 + e] += bf2f(a[e]) * xh;
        c1 += t * xh;
      }
    }
    c1 = wave_allreduce_sum(c1) / cols;
    bf16_t* dxr = dx + row * cols;
#pragma unroll
    for (int v = 0; v < NV; ++v) {
      const int c0 = v * WAVE * 8 + lane * 8;
      if (c0 >= cols) continue;
      // dx = rs * (dy*g - xh * mean(dy*g*xh))
      bf16x8 o;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float xh = bf2f(xv[v][e]) * rs;
        const float t = bf2f(dyv[v][e]) * bf2f(gv[v][e]);
        o[e] = f2bf(rs * (t - xh * c1));
      }
      if (c0 + 8 <= cols) {
        *reinterpret_cast<bf16x8*>(dxr + c0) = o;
      } else {
        for (int e = 0; e < 8 && c0 + e < cols; ++e) dxr[c0 + e] = o[e];
      }
    }
  }

  float* dgp = dg_part + (int64_t)gwave * cols;
#pragma unroll
  for (int v = 0; v < NV; ++v)
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const int c = v * WAVE * 8 + lane * 8 + e;
      if (c < cols) dgp[c] = dg_acc[v * 8 + e];
    }
}

__global__ void rms_reduce_kernel(const float* __restrict__ dg_part,
                                  bf16_t* __restrict__ dgamma,
                                  int part_rows, int cols) {
  const int lane = threadIdx.x & 63;
  const int c = blockIdx.x * (NT / WAVE) + (threadIdx.x >> 6);
  if (c >= cols) return;
  float s = 0.f;
  for (int r = lane; r < part_rows; r += WAVE)
    s += dg_part[(int64_t)r * cols + c];
  s = wave_allreduce_sum(s);
  if (lane == 0) dgamma[c] = f2bf(s);
}

// ---------------------------------------------------------------------------
// RoPE (rotate-half; angles from __sincosf in-kernel, no tables)
// ---------------------------------------------------------------------------

template <bool BWD>
__launch_bounds__(NT) __global__
void rope_kernel(const bf16_t* __restrict__ x, bf16_t* __restrict__ y,
                 int64_t tokens, int heads, int D, int seq_len,
                 float neg2_over_d, float ltheta) {
  // one thread handles 2 paired elements (d, d + D/2) of one (token, head)
  const int64_t half = (int64_t)tokens * heads * (D / 2);
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < half;
       i += stride) {
    const int d = (int)(i % (D / 2));
    const int64_t th = i / (D / 2);
    const int64_t tok = th / heads;
    const int pos = (int)(tok % seq_len);
    const float freq = __expf(ltheta * neg2_over_d * d);
    float sn, cs;
    __sincosf(pos * freq, &sn, &cs);
    if (BWD) sn = -sn;
    const int64_t base = th * D + d;
    const float x1 = bf2f(x[base]);
    const float x2 = bf2f(x[base + D / 2]);
    y[base] = f2bf(x1 * cs - x2 * sn);
    y[base + D / 2] = f2bf(x2 * cs + x1 * sn);
  }
}

// ---------------------------------------------------------------------------
// SwiGLU
// ---------------------------------------------------------------------------

DEV_INLINE float sigf(float x) { return 1.0f / (1.0f + __expf(-x)); }

__global__ void swiglu_fwd_kernel(const bf16_t* __restrict__ a,
                                  const bf16_t* __restrict__ b,
                                  bf16_t* __restrict__ y, int64_t n) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  for (int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
       i < n; i += stride) {
    if (i + 8 <= n) {
      const bf16x8 av = *reinterpret_cast<const bf16x8*>(a + i);
      const bf16x8 bv = *reinterpret_cast<const bf16x8*>(b + i);
      bf16x8 o;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float af = bf2f(av[e]);
        o[e] = f2bf(af * sigf(af) * bf2f(bv[e]));
      }
      *reinterpret_cast<bf16x8*>(y + i) = o;
    } else {
      for (int64_t j = i; j < n; ++j) {
        const float af = bf2f(a[j]);
        y[j] = f2bf(af * sigf(af) * bf2f(b[j]));
      }
    }
  }
}

__global__ void swiglu_bwd_kernel(const bf16_t* __restrict__ dy,
                                  const bf16_t* __restrict__ a,
                                  const bf16_t* __restrict__ b,
                                  bf16_t* __restrict__ da,
                                  bf16_t* __restrict__ db, int64_t n) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  for (int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
       i < n; i += stride) {
    if (i + 8 <= n) {
      const bf16x8 dv = *reinterpret_cast<const bf16x8*>(dy + i);
      const bf16x8 av = *reinterpret_cast<const bf16x8*>(a + i);
      const bf16x8 bv = *reinterpret_cast<const bf16x8*>(b + i);
      bf16x8 oa, ob;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float af = bf2f(av[e]);
        const float s = sigf(af);
        const float silu = af * s;
        const float d = bf2f(dv[e]);
        oa[e] = f2bf(d * bf2f(bv[e]) * (s + silu * (1.0f - s)));
        ob[e] = f2bf(d * silu);
      }
      *reinterpret_cast<bf16x8*>(da + i) = oa;
      *reinterpret_cast<bf16x8*>(db + i) = ob;
    } else {
      for (int64_t j = i; j < n; ++j) {
        const float af = bf2f(a[j]);
        const float s = sigf(af);
        const float silu = af * s;
        const float d = bf2f(dy[j]);
        da[j] = f2bf(d * bf2f(b[j]) * (s + silu * (1.0f - s)));
        db[j] = f2bf(d * silu);
      }
    }
  }
}

}  // namespace

void rmsnorm_fwd_bf16(const void* x, const void* g, void* y, float* rstd,
                      int64_t rows, int cols, float eps, hipStream_t stream) {
  const int blocks =
      (int)std::min<int64_t>((rows + NT / WAVE - 1) / (NT / WAVE), 2048);
#define RF(NVV)                                                              \
  hipLaunchKernelGGL(rms_fwd_kernel<NVV>, dim3(std::max(blocks, 1)),         \
                     dim3(NT), 0, stream, static_cast<const bf16_t*>(x),     \
                     static_cast<const bf16_t*>(g), static_cast<bf16_t*>(y), \
                     rstd, rows, cols, eps)
  switch (nv_for(cols)) {
    case 1: RF(1); break;
    case 2: RF(2); break;
    case 3: RF(3); break;
    case 4: RF(4); break;
    case 5: RF(5); break;
    case 6: RF(6); break;
    default: throw std::runtime_error("rmsnorm: cols too large");
  }
#undef RF
}

void rmsnorm_bwd_bf16(const void* dy, const void* x, const void* g,
                      const float* rstd, void* dx, void* dgamma,
                      float* dg_part, int part_rows, int64_t rows, int cols,
                      hipStream_t stream) {
  const int blocks = (part_rows + NT / WAVE - 1) / (NT / WAVE);
#define RB(NVV)                                                            \
  hipLaunchKernelGGL(rms_bwd_kernel<NVV>, dim3(std::max(blocks, 1)),       \
                     dim3(NT), 0, stream, static_cast<const bf16_t*>(dy),  \
                     static_cast<const bf16_t*>(x),                        \
                     static_cast<const bf16_t*>(g), rstd,                  \
                     static_cast<bf16_t*>(dx), dg_part, rows, cols)
  switch (nv_for(cols)) {
    case 1: RB(1); break;
    case 2: RB(2); break;
    case 3: RB(3); break;
    case 4: RB(4); break;
    case 5: RB(5); break;
    case 6: RB(6); break;
    default: throw std::runtime_error("rmsnorm: cols too large");
  }
#undef RB
  const int rblocks = (cols + NT / WAVE - 1) / (NT / WAVE);
  hipLaunchKernelGGL(rms_reduce_kernel, dim3(rblocks), dim3(NT), 0, stream,
                     dg_part, static_cast<bf16_t*>(dgamma), part_rows, cols);
}

void rope_bf16(const void* x, void* y, int64_t tokens, int heads, int D,
               int seq_len, float theta, bool backward, hipStream_t stream) {
  const int64_t half = tokens * heads * (D / 2);
  const int blocks = (int)std::min<int64_t>((half + NT - 1) / NT, 4096);
  const float ltheta = logf(theta);
  const float neg2 = -2.0f / D;
  if (backward)
    hipLaunchKernelGGL(rope_kernel<true>, dim3(std::max(blocks, 1)),
                       dim3(NT), 0, stream, static_cast<const bf16_t*>(x),
                       static_cast<bf16_t*>(y), tokens, heads, D, seq_len,
                       neg2, ltheta);
  else
    hipLaunchKernelGGL(rope_kernel<false>, dim3(std::max(blocks, 1)),
                       dim3(NT), 0, stream, static_cast<const bf16_t*>(x),
                       static_cast<bf16_t*>(y), tokens, heads, D, seq_len,
                       neg2, ltheta);
}

void swiglu_fwd_bf16(const void* a, const void* b, void* y, int64_t n,
                     hipStream_t stream) {
  const int blocks = (int)std::min<int64_t>((n / 8 + NT - 1) / NT, 2048);
  hipLaunchKernelGGL(swiglu_fwd_kernel, dim3(std::max(blocks, 1)), dim3(NT),
                     0, stream, static_cast<const bf16_t*>(a),
                     static_cast<const bf16_t*>(b), static_cast<bf16_t*>(y),
                     n);
}

void swiglu_bwd_bf16(const void* dy, const void* a, const void* b, void* da,
                     void* db, int64_t n, hipStream_t stream) {
  const int blocks = (int)std::min<int64_t>((n / 8 + NT - 1) / NT, 2048);
  hipLaunchKernelGGL(swiglu_bwd_kernel, dim3(std::max(blocks, 1)), dim3(NT),
                     0, stream, static_cast<const bf16_t*>(dy),
                     static_cast<const bf16_t*>(a),
                     static_cast<const bf16_t*>(b), static_cast<bf16_t*>(da),
                     static_cast<bf16_t*>(db), n);
}

}  // namespace tepdist
