// Embedding gather (fwd) and scatter-add (bwd) for gfx950.
// fwd: one wave copies one token row with bf16x8 vectors.
// bwd: fp32 atomic scatter-add into the table gradient, then cast to bf16.

#include <algorithm>
#include <stdexcept>

#include "common.h"
#include "kernels.h"

namespace tepdist {

namespace {

constexpr int NT = 256;

__global__ void embed_fwd_kernel(const int64_t* __restrict__ ids,
                                 const bf16_t* __restrict__ table,
                                 bf16_t* __restrict__ out, int64_t n_ids,
                                 int dim) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int64_t nwaves = (int64_t)gridDim.x * (NT / WAVE);
  for (int64_t i = (int64_t)blockIdx.x * (NT / WAVE) + wid; i < n_ids;
       i += nwaves) {
    const bf16_t* src = table + ids[i] * dim;
    bf16_t* dst = out + i * dim;
    for (int c = lane * 8; c + 8 <= dim; c += WAVE * 8)
      *reinterpret_cast<bf16x8*>(dst + c) =
          *reinterpret_cast<const bf16x8*>(src + c);
    // ragged tail (dim % 8 elements)
    for (int c = (dim / 8) * 8 + lane; c < dim; c += WAVE) dst[c] = src[c];
  }
}

__global__ void embed_bwd_scatter_kernel(const bf16_t* __restrict__ dy,
                                         const int64_t* __restrict__ ids,
                                         float* __restrict__ grad,
                                         int64_t n_ids, int dim) {
  const int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t total = n_ids * dim;
  for (int64_t i = idx; i < total; i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t tok = i / dim;
    const int c = (int)(i - tok * dim);
    atomicAdd(&grad[ids[tok] * dim + c], bf2f(dy[i]));
  }
}

__global__ void cast_f32_bf16_kernel(const float* __restrict__ src,
                                     bf16_t* __restrict__ dst, int64_t n) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * 4;
  for (int64_t idx = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 4;
       idx < n; idx += stride) {
    if (idx + 4 <= n) {
      const f32x4 v = *reinterpret_cast<const f32x4*>(src + idx);
      bf16x4 o;
#pragma unroll
      for (int e = 0; e < 4; ++e) o[e] = f2bf(v[e]);
      *reinterpret_cast<bf16x4*>(dst + idx) = o;
    } else {
      for (int64_t i = idx; i < n; ++i) dst[i] = f2bf(src[i]);
    }
  }
}

}  // namespace

void embedding_fwd_bf16(const int64_t* ids, const void* table, void* out,
                        int64_t n_ids, int dim, hipStream_t stream) {
  const int blocks = (int)std::min<int64_t>((n_ids + 3) / 4, 2048);
  hipLaunchKernelGGL(embed_fwd_kernel, dim3(blocks), dim3(NT), 0, stream, ids,
                     static_cast<const bf16_t*>(table),
                     static_cast<bf16_t*>(out), n_ids, dim);
}

void embedding_bwd_bf16(const void* dy, const int64_t* ids, float* grad_f32,
                        void* grad_bf16, int64_t n_ids, int vocab, int dim,
                        hipStream_t stream) {
  const int64_t total = n_ids * dim;
  const int blocks = (int)std::min<int64_t>((total + NT - 1) / NT, 2048);
  hipLaunchKernelGGL(embed_bwd_scatter_kernel, dim3(blocks), dim3(NT), 0,
                     stream, static_cast<const bf16_t*>(dy), ids, grad_f32,
                     n_ids, dim);
  const int64_t tn = (int64_t)vocab * dim;
  const int cblocks = (int)std::min<int64_t>((tn / 4 + NT - 1) / NT, 4096);
  hipLaunchKernelGGL(cast_f32_bf16_kernel, dim3(cblocks), dim3(NT), 0, stream,
                     grad_f32, static_cast<bf16_t*>(grad_bf16), tn);
}

}  // namespace tepdist
