// Tiled bf16 2-D transpose for gfx950: 64x64 tiles through LDS, 16B
// coalesced loads and stores (the LDS image is written row-major and read
// column-wise via the v_perm 8x8 register transpose on the way out).
// Used to canonicalize GEMM operands: dgrad/wgrad become k-contiguous x
// k-contiguous so the single tuned GEMM schedule serves every case.

#include <algorithm>
#include <stdexcept>

#include "common.h"
#include "kernels.h"

namespace tepdist {

namespace {

constexpr int NT = 256;
constexpr int TS = 64;  // tile size

// out[c][r] = in[r][c]; batched over blockIdx.z. GELUG additionally
// multiplies by gelu'(pre) on the way through (the fused gelu-backward:
// the wgrad path transposes dy anyway, so the elementwise backward rides
// along instead of a separate full read+write pass) and NAT also writes
// the post-gelu natural-layout image (the dgrad GEMM operand).
template <bool GELUG, bool NAT, bool BIAS = false>
__launch_bounds__(NT) __global__
void transpose_kernel(const bf16_t* __restrict__ in,
                      const bf16_t* __restrict__ pre,
                      bf16_t* __restrict__ out,
                      bf16_t* __restrict__ out_nat,
                      float* __restrict__ bias_ws, int R, int C,
                      int64_t stride_in, int64_t stride_out) {
  __shared__ bf16_t tile[TS * (TS + 8)];  // +16B row pad (b128-aligned)
  __shared__ float colacc[TS];            // BIAS: per-tile column sums
  const bf16_t* src = in + blockIdx.z * stride_in;
  bf16_t* dst = out + blockIdx.z * stride_out;
  const int r0 = blockIdx.y * TS;
  const int c0 = blockIdx.x * TS;
  if (BIAS) {
    if (threadIdx.x < TS) colacc[threadIdx.x] = 0.f;
    __syncthreads();
  }

  // load [64 rows][64 cols] with 16B vectors: 512 loads / 256 threads
#pragma unroll
  for (int u = 0; u < 2; ++u) {
    const int idx = threadIdx.x + u * NT;
    const int r = idx >> 3;
    const int c = (idx & 7) * 8;
    bf16x8 v = {};
    if (r0 + r < R) {
      const bf16_t* p = src + (int64_t)(r0 + r) * C + c0 + c;
      if (c0 + c + 8 <= C) {
        v = *reinterpret_cast<const bf16x8*>(p);
      } else {
        for (int e = 0; e < 8 && c0 + c + e < C; ++e) v[e] = p[e];
      }
      if (GELUG) {
        const bf16_t* q = pre + (int64_t)(r0 + r) * C + c0 + c;
        bf16x8 pv = {};
        if (c0 + c + 8 <= C) {
          pv = *reinterpret_cast<const bf16x8*>(q);
        } else {
          for (int e = 0; e < 8 && c0 + c + e < C; ++e) pv[e] = q[e];
        }
#pragma unroll
        for (int e = 0; e < 8; ++e)
          v[e] = f2bf(bf2f(v[e]) * gelu_grad_f(bf2f(pv[e])));
      }
      if (NAT) {
        bf16_t* pn = out_nat + (int64_t)(r0 + r) * C + c0 + c;
        if (c0 + c + 8 <= C) {
          *reinterpret_cast<bf16x8*>(pn) = v;
        } else {
          for (int e = 0; e < 8 && c0 + c + e < C; ++e) pn[e] = v[e];
        }
      }
      if (BIAS) {
        // column partial sums ride along with the transpose: the
        // separate full-tensor bias-grad pass disappears
#pragma unroll
        for (int e = 0; e < 8; ++e)
          if (c0 + c + e < C) atomicAdd(&colacc[c + e], bf2f(v[e]));
      }
    }
    *reinterpret_cast<bf16x8*>(tile + r * (TS + 8) + c) = v;
  }
  __syncthreads();
  if (BIAS && threadIdx.x < TS && c0 + (int)threadIdx.x < C)
    atomicAdd(&bias_ws[c0 + threadIdx.x], colacc[threadIdx.x]);

  // store transposed: thread reads a column 8-run via 8 scalar LDS reads
  // (padded rows -> conflict-light), writes one 16B row of the output
#pragma unroll
  for (int u = 0; u < 2; ++u) {
    const int idx = threadIdx.x + u * NT;
    const int c = idx >> 3;        // output row = input col
    const int r = (idx & 7) * 8;   // output col run = input rows
    if (c0 + c < C) {
      bf16x8 v;
#pragma unroll
      for (int e = 0; e < 8; ++e) v[e] = tile[(r + e) * (TS + 8) + c];
      bf16_t* p = dst + (int64_t)(c0 + c) * R + r0 + r;
      if (r0 + r + 8 <= R) {
        *reinterpret_cast<bf16x8*>(p) = v;
      } else {
        for (int e = 0; e < 8 && r0 + r + e < R; ++e) p[e] = v[e];
      }
    }
  }
}

// v2: 128x128 blocks, XOR-swizzled LDS image, and a v_perm 8x8 register
// transpose on the way out — every LDS access is a 16-byte vector
// (conflict-free by the swizzle) and every global write a 16-byte chunk
// in 128-byte runs, replacing the 8-scalar-LDS-read column walk of the
// generic kernel. Requires R % 128 == 0 && C % 128 == 0 (all the
// canonicalization transposes); the generic kernel covers the rest.
constexpr int T2 = 128;

DEV_INLINE int t2off(int row, int col_e) {  // loff for a [.][128] image
  const int r1 = row >> 1;
  const int x = ((r1 & 1) << 2) | (r1 & 2) | ((r1 >> 2) & 1);
  return row * T2 + (col_e ^ ((x << 3) & 127));
}

template <bool GELUG, bool NAT>
__launch_bounds__(NT) __global__
void transpose2_kernel(const bf16_t* __restrict__ in,
                       const bf16_t* __restrict__ pre,
                       bf16_t* __restrict__ out,
                       bf16_t* __restrict__ out_nat, int R, int C,
                       int64_t stride_in, int64_t stride_out) {
  __shared__ bf16_t tile[T2 * T2];
  const bf16_t* src = in + blockIdx.z * stride_in;
  bf16_t* dst = out + blockIdx.z * stride_out;
  const int r0 = blockIdx.y * T2;
  const int c0 = blockIdx.x * T2;

  // load [128][128]: 2048 16B vectors / 256 threads
#pragma unroll
  for (int u = 0; u < 8; ++u) {
    const int idx = threadIdx.x + u * NT;
    const int r = idx >> 4;
    const int c = (idx & 15) * 8;
    bf16x8 v =
        *reinterpret_cast<const bf16x8*>(src + (int64_t)(r0 + r) * C +
                                         c0 + c);
    if (GELUG) {
      const bf16x8 pv = *reinterpret_cast<const bf16x8*>(
          pre + (int64_t)(r0 + r) * C + c0 + c);
#pragma unroll
      for (int e = 0; e < 8; ++e)
        v[e] = f2bf(bf2f(v[e]) * gelu_grad_f(bf2f(pv[e])));
    }
    if (NAT)
      *reinterpret_cast<bf16x8*>(out_nat + (int64_t)(r0 + r) * C + c0 + c) =
          v;
    *reinterpret_cast<bf16x8*>(tile + t2off(r, c)) = v;
  }
  __syncthreads();

  // store: thread owns an 8x8 block; cblk is the slow index so that the
  // 8 lanes of each cblk group write 128 contiguous bytes per output row
  const int cblk = threadIdx.x >> 4;       // 0..15 (input col block)
  const int rblk = threadIdx.x & 15;       // 0..15 (input row block)
  uint32_t rows[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
    *reinterpret_cast<bf16x8*>(rows[i]) = *reinterpret_cast<const bf16x8*>(
        tile + t2off(rblk * 8 + i, cblk * 8));
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    uint32_t lo[4], hi[4];
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      lo[p] = __builtin_amdgcn_perm(rows[2 * p + 1][j], rows[2 * p][j],
                                    0x05040100u);
      hi[p] = __builtin_amdgcn_perm(rows[2 * p + 1][j], rows[2 * p][j],
                                    0x07060302u);
    }
    bf16_t* d0 = dst + (int64_t)(c0 + cblk * 8 + 2 * j) * R + r0 + rblk * 8;
    bf16_t* d1 = d0 + R;
    *reinterpret_cast<uint4*>(d0) = make_uint4(lo[0], lo[1], lo[2], lo[3]);
    *reinterpret_cast<uint4*>(d1) = make_uint4(hi[0], hi[1], hi[2], hi[3]);
  }
}

}  // namespace

void transpose_bf16(const void* in, void* out, int R, int C,
                    int64_t stride_in, int64_t stride_out, int batch,
                    hipStream_t stream) {
  if (R % T2 == 0 && C % T2 == 0) {
    dim3 grid(C / T2, R / T2, batch);
    hipLaunchKernelGGL((transpose2_kernel<false, false>), grid, dim3(NT), 0,
                       stream, static_cast<const bf16_t*>(in), nullptr,
                       static_cast<bf16_t*>(out), nullptr, R, C, stride_in,
                       stride_out);
    return;
  }
  dim3 grid((C + TS - 1) / TS, (R + TS - 1) / TS, batch);
  hipLaunchKernelGGL((transpose_kernel<false, false>), grid, dim3(NT), 0,
                     stream, static_cast<const bf16_t*>(in), nullptr,
                     static_cast<bf16_t*>(out), nullptr, nullptr, R, C,
                     stride_in, stride_out);
}

// dy-transpose variants for linear backward: optional fused gelu'(pre)
// multiply (+ natural-layout copy) and fused bias column sums.
void transpose_dy_bf16(const void* dy, const void* pre, void* dy_t,
                       void* dy_nat, float* bias_ws, int R, int C,
                       hipStream_t stream) {
  const bf16_t* d = static_cast<const bf16_t*>(dy);
  const bf16_t* p = static_cast<const bf16_t*>(pre);
  bf16_t* t = static_cast<bf16_t*>(dy_t);
  bf16_t* n = static_cast<bf16_t*>(dy_nat);
  if (bias_ws == nullptr && R % T2 == 0 && C % T2 == 0) {
    dim3 g2(C / T2, R / T2, 1);
    if (pre != nullptr)
      hipLaunchKernelGGL((transpose2_kernel<true, true>), g2, dim3(NT), 0,
                         stream, d, p, t, n, R, C, 0, 0);
    else
      hipLaunchKernelGGL((transpose2_kernel<false, false>), g2, dim3(NT), 0,
                         stream, d, p, t, n, R, C, 0, 0);
    return;
  }
  dim3 grid((C + TS - 1) / TS, (R + TS - 1) / TS, 1);
#define TK(G, NA, B)                                                     \
  hipLaunchKernelGGL((transpose_kernel<G, NA, B>), grid, dim3(NT), 0,    \
                     stream, d, p, t, n, bias_ws, R, C, 0, 0)
  if (pre != nullptr) {
    if (bias_ws != nullptr) TK(true, true, true);
    else TK(true, true, false);
  } else {
    if (bias_ws != nullptr) TK(false, false, true);
    else TK(false, false, false);
  }
#undef TK
}

}  // namespace tepdist
