// Elementwise kernels: GELU fwd/bwd, bias-grad column sum. Vectorized bf16x8
// grid-stride loops (guide G13).

#include <algorithm>
#include <stdexcept>

#include "common.h"
#include "kernels.h"

namespace tepdist {

namespace {

constexpr int NT = 256;

__launch_bounds__(NT) __global__
void gelu_fwd_kernel(const bf16_t* __restrict__ x, bf16_t* __restrict__ y,
                     int64_t n) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  for (int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
       i0 < n; i0 += stride) {
    if (i0 + 8 <= n) {
      // streaming, never re-read: bypass L2 (nontemporal)
      const bf16x8 xv = __builtin_nontemporal_load(
          reinterpret_cast<const bf16x8*>(x + i0));
      bf16x8 yv;
#pragma unroll
      for (int e = 0; e < 8; ++e) yv[e] = f2bf(gelu_f(bf2f(xv[e])));
      __builtin_nontemporal_store(yv, reinterpret_cast<bf16x8*>(y + i0));
    } else {
      for (int64_t i = i0; i < n; ++i) y[i] = f2bf(gelu_f(bf2f(x[i])));
    }
  }
}

__launch_bounds__(NT) __global__
void gelu_bwd_kernel(const bf16_t* __restrict__ dy,
                     const bf16_t* __restrict__ x, bf16_t* __restrict__ dx,
                     int64_t n) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  for (int64_t i0 = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
       i0 < n; i0 += stride) {
    if (i0 + 8 <= n) {
      const bf16x8 dv = __builtin_nontemporal_load(
          reinterpret_cast<const bf16x8*>(dy + i0));
      const bf16x8 xv = __builtin_nontemporal_load(
          reinterpret_cast<const bf16x8*>(x + i0));
      bf16x8 o;
#pragma unroll
      for (int e = 0; e < 8; ++e)
        o[e] = f2bf(bf2f(dv[e]) * gelu_grad_f(bf2f(xv[e])));
      __builtin_nontemporal_store(o, reinterpret_cast<bf16x8*>(dx + i0));
    } else {
      for (int64_t i = i0; i < n; ++i)
        dx[i] = f2bf(bf2f(dy[i]) * gelu_grad_f(bf2f(x[i])));
    }
  }
}

// db[c] = sum_r dy[r][c]. Each thread owns an 8-column group (one bf16x8
// load per row -> coalesced) and walks a row slice; fp32 atomics combine
// the row slices in ws.
__launch_bounds__(NT) __global__
void bias_sum_kernel(const bf16_t* __restrict__ dy, float* __restrict__ ws,
                     int64_t rows, int cols) {
  // thread owns an 8-col group; 4-row unroll keeps 4 dwordx4 loads in
  // flight per thread (latency cover for the strided row walk)
  const int ncg = (cols + 7) / 8;
  const int cg = blockIdx.x * NT + threadIdx.x;
  if (cg >= ncg) return;
  const int c0 = cg * 8;
  const int64_t r0 = (rows * blockIdx.y) / gridDim.y;
  const int64_t r1 = (rows * (blockIdx.y + 1)) / gridDim.y;
  float acc[8] = {};
  if (c0 + 8 <= cols) {
    int64_t r = r0;
    for (; r + 4 <= r1; r += 4) {
      bf16x8 v[4];
#pragma unroll
      for (int u = 0; u < 4; ++u)
        v[u] = *reinterpret_cast<const bf16x8*>(dy + (r + u) * cols + c0);
#pragma unroll
      for (int u = 0; u < 4; ++u)
#pragma unroll
        for (int e = 0; e < 8; ++e) acc[e] += bf2f(v[u][e]);
    }
    for (; r < r1; ++r) {
      const bf16x8 v = *reinterpret_cast<const bf16x8*>(dy + r * cols + c0);
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[e] += bf2f(v[e]);
    }
  } else {
    for (int64_t r = r0; r < r1; ++r)
      for (int e = 0; e < 8 && c0 + e < cols; ++e)
        acc[e] += bf2f(dy[r * cols + c0 + e]);
  }
#pragma unroll
  for (int e = 0; e < 8; ++e)
    if (c0 + e < cols) atomicAdd(&ws[c0 + e], acc[e]);
}

// streaming variant for cols % 512 == 0 (NV = cols/512 <= 8): each wave
// reads whole ROWS sequentially (full 1 KiB per instruction, perfect
// channel interleave) holding NV*8 per-lane column accumulators; waves
// combine through LDS atomics, one global atomic pass per block.
template <int NV>
__launch_bounds__(NT) __global__
void bias_sum_stream_kernel(const bf16_t* __restrict__ dy,
                            float* __restrict__ ws, int64_t rows) {
  const int cols = NV * 512;
  __shared__ float acc_s[NV * 512];
  for (int i = threadIdx.x; i < cols; i += NT) acc_s[i] = 0.f;
  __syncthreads();

  const int wid = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int64_t nw = (int64_t)gridDim.x * (NT / 64);
  const int64_t w = (int64_t)blockIdx.x * (NT / 64) + wid;
  const int64_t rb = (rows * w) / nw, re = (rows * (w + 1)) / nw;

  float acc[NV][8] = {};
  // load ALL of a row's vectors before accumulating (separate destination
  // registers -> the loads pipeline instead of serializing on vmcnt(0))
  for (int64_t r = rb; r < re; ++r) {
    const bf16_t* p = dy + r * cols + lane * 8;
    bf16x8 v[NV];
#pragma unroll
    for (int j = 0; j < NV; ++j)
      v[j] = *reinterpret_cast<const bf16x8*>(p + j * 512);
#pragma unroll
    for (int j = 0; j < NV; ++j)
#pragma unroll
      for (int e = 0; e < 8; ++e) acc[j][e] += bf2f(v[j][e]);
  }
#pragma unroll
  for (int j = 0; j < NV; ++j)
#pragma unroll
    for (int e = 0; e < 8; ++e)
      atomicAdd(&acc_s[j * 512 + lane * 8 + e], acc[j][e]);
  __syncthreads();
  for (int i = threadIdx.x; i < cols; i += NT)
    atomicAdd(&ws[i], acc_s[i]);
}

__global__ void cast_ws_kernel(const float* __restrict__ ws,
                               bf16_t* __restrict__ db, int cols) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c < cols) db[c] = f2bf(ws[c]);
}

}  // namespace

void gelu_fwd_bf16(const void* x, void* y, int64_t n, hipStream_t stream) {
  const int blocks = (int)std::min<int64_t>((n / 8 + NT - 1) / NT, 4096);
  hipLaunchKernelGGL(gelu_fwd_kernel, dim3(std::max(blocks, 1)), dim3(NT), 0,
                     stream, static_cast<const bf16_t*>(x),
                     static_cast<bf16_t*>(y), n);
}

void gelu_bwd_bf16(const void* dy, const void* x, void* dx, int64_t n,
                   hipStream_t stream) {
  const int blocks = (int)std::min<int64_t>((n / 8 + NT - 1) / NT, 4096);
  hipLaunchKernelGGL(gelu_bwd_kernel, dim3(std::max(blocks, 1)), dim3(NT), 0,
                     stream, static_cast<const bf16_t*>(dy),
                     static_cast<const bf16_t*>(x), static_cast<bf16_t*>(dx),
                     n);
}

void cast_ws_f32_bf16(const float* ws, void* db_out, int cols,
                      hipStream_t stream) {
  hipLaunchKernelGGL(cast_ws_kernel, dim3((cols + NT - 1) / NT), dim3(NT), 0,
                     stream, ws, static_cast<bf16_t*>(db_out), cols);
}

void bias_sum_bf16(const void* dy, void* db_out, float* ws_zeroed,
                   int64_t rows, int cols, hipStream_t stream) {
  bf16_t* db = static_cast<bf16_t*>(db_out);
  float* ws = ws_zeroed;  // caller-zeroed fp32 workspace of `cols`
  const int nv = cols / 512;
  if (cols % 512 == 0 && nv >= 1 && nv <= 8) {
    const int blocks =
        (int)std::max<int64_t>(1, std::min<int64_t>((rows + 63) / 64, 256));
#define BS(NVV)                                                           \
  hipLaunchKernelGGL(bias_sum_stream_kernel<NVV>, dim3(blocks), dim3(NT), \
                     0, stream, static_cast<const bf16_t*>(dy), ws, rows)
    switch (nv) {
      case 1: BS(1); break;
      case 2: BS(2); break;
      case 3: BS(3); break;
      case 4: BS(4); break;
      case 5: BS(5); break;
      case 6: BS(6); break;
      case 7: BS(7); break;
      case 8: BS(8); break;
    }
#undef BS
  } else {
    const int ysplit = (int)std::min<int64_t>((rows + 31) / 32, 256);
    const int ncg = (cols + 7) / 8;
    dim3 grid((ncg + NT - 1) / NT, std::max(ysplit, 1));
    hipLaunchKernelGGL(bias_sum_kernel, grid, dim3(NT), 0, stream,
                       static_cast<const bf16_t*>(dy), ws, rows, cols);
  }
  hipLaunchKernelGGL(cast_ws_kernel, dim3((cols + NT - 1) / NT), dim3(NT), 0,
                     stream, ws, db, cols);
}

}  // namespace tepdist
