// LayerNorm forward/backward for gfx950. Memory-bound: one wave per row,
// bf16x8-vectorized loads (guide G13), fp32 statistics in registers.
// Register arrays are statically indexed via a compile-time NV (vectors per
// lane) template dispatch — runtime-indexed ext_vector arrays would spill to
// scratch. dgamma/dbeta use per-wave fp32 partials + a reduce kernel.

#include <algorithm>
#include <stdexcept>

#include "common.h"
#include "kernels.h"

namespace tepdist {

namespace {

constexpr int NT = 256;  // 4 waves per block

template <int NV, bool ADD = false>
__launch_bounds__(NT) __global__
void ln_fwd_kernel(const bf16_t* __restrict__ x, const bf16_t* __restrict__ g,
                   const bf16_t* __restrict__ b, bf16_t* __restrict__ y,
                   float* __restrict__ mean_out, float* __restrict__ rstd_out,
                   int rows, int cols, float eps,
                   const bf16_t* __restrict__ res = nullptr,
                   bf16_t* __restrict__ sum_out = nullptr) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int nwaves = gridDim.x * (NT / WAVE);

  for (int row = blockIdx.x * (NT / WAVE) + wid; row < rows; row += nwaves) {
    const bf16_t* xr = x + (int64_t)row * cols;
    const bf16_t* rr = ADD ? res + (int64_t)row * cols : nullptr;
    bf16_t* sr = ADD ? sum_out + (int64_t)row * cols : nullptr;
    float vals[NV * 8];
    float s = 0.f;
#pragma unroll
    for (int v = 0; v < NV; ++v) {
      const int c0 = v * WAVE * 8 + lane * 8;
      bf16x8 xv = {};
      bf16x8 rv = {};
      if (c0 + 8 <= cols) {
        xv = *reinterpret_cast<const bf16x8*>(xr + c0);
        if (ADD) rv = *reinterpret_cast<const bf16x8*>(rr + c0);
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e)
          if (c0 + e < cols) {
            xv[e] = xr[c0 + e];
            if (ADD) rv[e] = rr[c0 + e];
          }
      }
      bf16x8 sv;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        // the residual SUM is computed in fp32 then rounded ONCE to bf16;
        // statistics read back the rounded value so y matches a separate
        // add kernel bit-for-bit
        float f = bf2f(xv[e]);
        if (ADD) {
          sv[e] = f2bf(f + bf2f(rv[e]));
          f = bf2f(sv[e]);
        }
        vals[v * 8 + e] = f;
        s += f;
      }
      if (ADD && c0 < cols) {
        if (c0 + 8 <= cols) {
          *reinterpret_cast<bf16x8*>(sr + c0) = sv;
        } else {
          for (int e = 0; e < 8 && c0 + e < cols; ++e) sr[c0 + e] = sv[e];
        }
      }
    }
    s = wave_allreduce_sum(s);
    const float mean = s / cols;
    float ss = 0.f;
#pragma unroll
    for (int v = 0; v < NV; ++v)
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int c = v * WAVE * 8 + lane * 8 + e;
        const float d = (c < cols) ? (vals[v * 8 + e] - mean) : 0.f;
        ss += d * d;
      }
    ss = wave_allreduce_sum(ss);
    const float rstd = rsqrtf(ss / cols + eps);
    if (lane == 0) {
      mean_out[row] = mean;
      rstd_out[row] = rstd;
    }
    bf16_t* yr = y + (int64_t)row * cols;
#pragma unroll
    for (int v = 0; v < NV; ++v) {
      const int c0 = v * WAVE * 8 + lane * 8;
      if (c0 >= cols) continue;
      bf16x8 yv = {};
      if (c0 + 8 <= cols) {
        const bf16x8 gv = *reinterpret_cast<const bf16x8*>(g + c0);
        const bf16x8 bv = *reinterpret_cast<const bf16x8*>(b + c0);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const float xhat = (vals[v * 8 + e] - mean) * rstd;
          yv[e] = f2bf(xhat * bf2f(gv[e]) + bf2f(bv[e]));
        }
        *reinterpret_cast<bf16x8*>(yr + c0) = yv;
      } else {
        for (int e = 0; e < 8 && c0 + e < cols; ++e) {
          const float xhat = (vals[v * 8 + e] - mean) * rstd;
          yr[c0 + e] = f2bf(xhat * bf2f(g[c0 + e]) + bf2f(b[c0 + e]));
        }
      }
    }
  }
}

// Backward: two passes over the row (second pass hits L1/L2), keeping only
// the dgamma/dbeta per-lane accumulators live across rows.
template <int NV, bool DS = false>
__launch_bounds__(NT) __global__
void ln_bwd_kernel(const bf16_t* __restrict__ dy, const bf16_t* __restrict__ x,
                   const bf16_t* __restrict__ g, const float* __restrict__ mean,
                   const float* __restrict__ rstd, bf16_t* __restrict__ dx,
                   float* __restrict__ dg_part, float* __restrict__ db_part,
                   int rows, int cols,
                   const bf16_t* __restrict__ dsum = nullptr) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int gwave = blockIdx.x * (NT / WAVE) + wid;
  const int nwaves = gridDim.x * (NT / WAVE);

  float dg_acc[NV * 8] = {};
  float db_acc[NV * 8] = {};

  // gamma is row-invariant: load once per wave, not per row
  bf16x8 gvv[NV];
#pragma unroll
  for (int v = 0; v < NV; ++v) {
    const int c0 = v * WAVE * 8 + lane * 8;
    bf16x8 gv = {};
    if (c0 + 8 <= cols) {
      gv = *reinterpret_cast<const bf16x8*>(g + c0);
    } else {
      for (int e = 0; e < 8 && c0 + e < cols; ++e) gv[e] = g[c0 + e];
    }
    gvv[v] = gv;
  }

  for (int row = gwave; row < rows; row += nwaves) {
    const bf16_t* dyr = dy + (int64_t)row * cols;
    const bf16_t* xr = x + (int64_t)row * cols;
    const float mu = mean[row], rs = rstd[row];
    float c1 = 0.f, c2 = 0.f;
    // dy/x stay in registers across both passes (no second global read)
    bf16x8 dyvv[NV], xvv[NV];
#pragma unroll
    for (int v = 0; v < NV; ++v) {
      const int c0 = v * WAVE * 8 + lane * 8;
      bf16x8 dyv = {}, xv = {};
      if (c0 + 8 <= cols) {
        dyv = *reinterpret_cast<const bf16x8*>(dyr + c0);
        xv = *reinterpret_cast<const bf16x8*>(xr + c0);
      } else {
        for (int e = 0; e < 8 && c0 + e < cols; ++e) {
          dyv[e] = dyr[c0 + e];
          xv[e] = xr[c0 + e];
        }
      }
      dyvv[v] = dyv;
      xvv[v] = xv;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int c = c0 + e;
        const float d = bf2f(dyv[e]);
        const float xhat = (c < cols) ? (bf2f(xv[e]) - mu) * rs : 0.f;
        const float gg = (c < cols) ? d * bf2f(gvv[v][e]) : 0.f;
        dg_acc[v * 8 + e] += d * xhat;
        db_acc[v * 8 + e] += d;
        c1 += gg;
        c2 += gg * xhat;
      }
    }
    c1 = wave_allreduce_sum(c1) / cols;
    c2 = wave_allreduce_sum(c2) / cols;
    bf16_t* dxr = dx + (int64_t)row * cols;
    const bf16_t* dsr = DS ? dsum + (int64_t)row * cols : nullptr;
#pragma unroll
    for (int v = 0; v < NV; ++v) {
      const int c0 = v * WAVE * 8 + lane * 8;
      if (c0 >= cols) continue;
      bf16x8 dsv = {};
      if (DS) {
        if (c0 + 8 <= cols) {
          dsv = *reinterpret_cast<const bf16x8*>(dsr + c0);
        } else {
          for (int e = 0; e < 8 && c0 + e < cols; ++e) dsv[e] = dsr[c0 + e];
        }
      }
      bf16x8 dxv;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float xhat = (bf2f(xvv[v][e]) - mu) * rs;
        const float gg = bf2f(dyvv[v][e]) * bf2f(gvv[v][e]);
        float d = (gg - c1 - xhat * c2) * rs;
        if (DS) d += bf2f(dsv[e]);
        dxv[e] = f2bf(d);
      }
      if (c0 + 8 <= cols) {
        *reinterpret_cast<bf16x8*>(dxr + c0) = dxv;
      } else {
        for (int e = 0; e < 8 && c0 + e < cols; ++e) dxr[c0 + e] = dxv[e];
      }
    }
  }

  float* dgp = dg_part + (int64_t)gwave * cols;
  float* dbp = db_part + (int64_t)gwave * cols;
#pragma unroll
  for (int v = 0; v < NV; ++v)
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const int c = v * WAVE * 8 + lane * 8 + e;
      if (c < cols) {
        dgp[c] = dg_acc[v * 8 + e];
        dbp[c] = db_acc[v * 8 + e];
      }
    }
}

// one wave per column: lanes stride the partial rows, wave-reduce.
__global__ void ln_bwd_reduce_kernel(const float* __restrict__ dg_part,
                                     const float* __restrict__ db_part,
                                     bf16_t* __restrict__ dgamma,
                                     bf16_t* __restrict__ dbeta,
                                     int part_rows, int cols) {
  const int lane = threadIdx.x & 63;
  const int c = blockIdx.x * (NT / WAVE) + (threadIdx.x >> 6);
  if (c >= cols) return;
  float sg = 0.f, sb = 0.f;
  for (int r = lane; r < part_rows; r += WAVE) {
    sg += dg_part[(int64_t)r * cols + c];
    sb += db_part[(int64_t)r * cols + c];
  }
  sg = wave_allreduce_sum(sg);
  sb = wave_allreduce_sum(sb);
  if (lane == 0) {
    dgamma[c] = f2bf(sg);
    dbeta[c] = f2bf(sb);
  }
}

int nv_for(int cols) { return (cols + WAVE * 8 - 1) / (WAVE * 8); }

}  // namespace

void layernorm_fwd_bf16(const void* x, const void* gamma, const void* beta,
                        void* y, float* mean, float* rstd, int rows, int cols,
                        float eps, hipStream_t stream, const void* res,
                        void* sum_out) {
  const int blocks = std::min((rows + 3) / 4, 2048);
  const dim3 g(blocks), blk(NT);
  const bf16_t* xp = static_cast<const bf16_t*>(x);
  const bf16_t* gp = static_cast<const bf16_t*>(gamma);
  const bf16_t* bp = static_cast<const bf16_t*>(beta);
  bf16_t* yp = static_cast<bf16_t*>(y);
  const bf16_t* rp = static_cast<const bf16_t*>(res);
  bf16_t* sp = static_cast<bf16_t*>(sum_out);
#define LN_FWD(NV)                                                           \
  do {                                                                       \
    if (rp)                                                                  \
      hipLaunchKernelGGL((ln_fwd_kernel<NV, true>), g, blk, 0, stream, xp,  \
                         gp, bp, yp, mean, rstd, rows, cols, eps, rp, sp);  \
    else                                                                     \
      hipLaunchKernelGGL((ln_fwd_kernel<NV, false>), g, blk, 0, stream, xp, \
                         gp, bp, yp, mean, rstd, rows, cols, eps);          \
  } while (0)
  switch (nv_for(cols)) {
    case 1: LN_FWD(1); break;
    case 2: LN_FWD(2); break;
    case 3: LN_FWD(3); break;
    case 4: LN_FWD(4); break;
    case 5: LN_FWD(5); break;
    case 6: LN_FWD(6); break;
    default: throw std::runtime_error("layernorm: cols > 3072 unsupported");
  }
#undef LN_FWD
}

void layernorm_bwd_bf16(const void* dy, const void* x, const void* gamma,
                        const float* mean, const float* rstd, void* dx,
                        float* dgamma_part, float* dbeta_part, int rows,
                        int cols, int part_rows, hipStream_t stream,
                        const void* dsum) {
  const int blocks = part_rows / (NT / WAVE);
  const dim3 g(blocks), blk(NT);
  const bf16_t* dyp = static_cast<const bf16_t*>(dy);
  const bf16_t* xp = static_cast<const bf16_t*>(x);
  const bf16_t* gp = static_cast<const bf16_t*>(gamma);
  bf16_t* dxp = static_cast<bf16_t*>(dx);
  const bf16_t* dsp = static_cast<const bf16_t*>(dsum);
#define LN_BWD(NV)                                                           \
  do {                                                                       \
    if (dsp)                                                                 \
      hipLaunchKernelGGL((ln_bwd_kernel<NV, true>), g, blk, 0, stream, dyp, \
                         xp, gp, mean, rstd, dxp, dgamma_part, dbeta_part,  \
                         rows, cols, dsp);                                  \
    else                                                                     \
      hipLaunchKernelGGL((ln_bwd_kernel<NV, false>), g, blk, 0, stream,     \
                         dyp, xp, gp, mean, rstd, dxp, dgamma_part,         \
                         dbeta_part, rows, cols);                          \
  } while (0)
  switch (nv_for(cols)) {
    case 1: LN_BWD(1); break;
    case 2: LN_BWD(2); break;
    case 3: LN_BWD(3); break;
    case 4: LN_BWD(4); break;
    case 5: LN_BWD(5); break;
    case 6: LN_BWD(6); break;
    default: throw std::runtime_error("layernorm: cols > 3072 unsupported");
  }
#undef LN_BWD
}

void layernorm_bwd_reduce(const float* dgamma_part, const float* dbeta_part,
                          void* dgamma, void* dbeta, int part_rows, int cols,
                          hipStream_t stream) {
  hipLaunchKernelGGL(ln_bwd_reduce_kernel, dim3((cols + 3) / 4), dim3(NT),
                     0, stream, dgamma_part, dbeta_part,
                     static_cast<bf16_t*>(dgamma), static_cast<bf16_t*>(dbeta),
                     part_rows, cols);
}

}  // namespace tepdist
