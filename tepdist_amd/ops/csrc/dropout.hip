// Philox counter-based dropout for gfx950: the mask is a pure function of
// (seed, offset, element index) — same construction as the reference's
// server-side Philox initializers (SURVEY.md §2.6 DistributedRandomInitializer)
// so recompute never needs the mask shipped across ranks.

#include <algorithm>
#include <stdexcept>

#include "common.h"
#include "kernels.h"

namespace tepdist {

namespace {

constexpr int NT = 256;

__launch_bounds__(NT) __global__
void dropout_fwd_kernel(const bf16_t* __restrict__ x, bf16_t* __restrict__ y,
                        uint8_t* __restrict__ mask, int64_t n, float p,
                        float scale, uint64_t seed, uint64_t offset) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * 4;
  for (int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 4;
       i0 < n; i0 += stride) {
    Philox4 rng(seed, i0 >> 2, offset);
    const uint4 r = rng();
    const uint32_t rs[4] = {r.x, r.y, r.z, r.w};
    if (i0 + 4 <= n) {
      const bf16x4 xv = *reinterpret_cast<const bf16x4*>(x + i0);
      bf16x4 yv;
      uchar4 mv;
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        const bool keep = u32_to_uniform(rs[e]) > p;
        reinterpret_cast<uint8_t*>(&mv)[e] = keep;
        yv[e] = keep ? f2bf(bf2f(xv[e]) * scale) : f2bf(0.f);
      }
      *reinterpret_cast<bf16x4*>(y + i0) = yv;
      *reinterpret_cast<uchar4*>(mask + i0) = mv;
    } else {
      for (int e = 0; e < 4 && i0 + e < n; ++e) {
        const bool keep = u32_to_uniform(rs[e]) > p;
        mask[i0 + e] = keep;
        y[i0 + e] = keep ? f2bf(bf2f(x[i0 + e]) * scale) : f2bf(0.f);
      }
    }
  }
}

__launch_bounds__(NT) __global__
void dropout_bwd_kernel(const bf16_t* __restrict__ dy,
                        const uint8_t* __restrict__ mask,
                        bf16_t* __restrict__ dx, int64_t n, float scale) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * 4;
  for (int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 4;
       i0 < n; i0 += stride) {
    if (i0 + 4 <= n) {
      const bf16x4 dv = *reinterpret_cast<const bf16x4*>(dy + i0);
      const uchar4 mv = *reinterpret_cast<const uchar4*>(mask + i0);
      bf16x4 o;
      const uint8_t* mp = reinterpret_cast<const uint8_t*>(&mv);
#pragma unroll
      for (int e = 0; e < 4; ++e)
        o[e] = mp[e] ? f2bf(bf2f(dv[e]) * scale) : f2bf(0.f);
      *reinterpret_cast<bf16x4*>(dx + i0) = o;
    } else {
      for (int e = 0; e < 4 && i0 + e < n; ++e)
        dx[i0 + e] = mask[i0 + e] ? f2bf(bf2f(dy[i0 + e]) * scale) : f2bf(0.f);
    }
  }
}

}  // namespace

void dropout_fwd_bf16(const void* x, void* y, uint8_t* mask, int64_t n,
                      float p, uint64_t seed, uint64_t offset,
                      hipStream_t stream) {
  const int blocks = (int)std::min<int64_t>((n / 4 + NT - 1) / NT, 2048);
  hipLaunchKernelGGL(dropout_fwd_kernel, dim3(std::max(blocks, 1)), dim3(NT),
                     0, stream, static_cast<const bf16_t*>(x),
                     static_cast<bf16_t*>(y), mask, n, p, 1.0f / (1.0f - p),
                     seed, offset);
}

void dropout_bwd_bf16(const void* dy, const uint8_t* mask, void* dx, int64_t n,
                      float p, hipStream_t stream) {
  const int blocks = (int)std::min<int64_t>((n / 4 + NT - 1) / NT, 2048);
  hipLaunchKernelGGL(dropout_bwd_kernel, dim3(std::max(blocks, 1)), dim3(NT),
                     0, stream, static_cast<const bf16_t*>(dy), mask,
                     static_cast<bf16_t*>(dx), n, 1.0f / (1.0f - p));
}

}  // namespace tepdist
