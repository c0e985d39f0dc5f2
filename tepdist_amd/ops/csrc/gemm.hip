// bf16 MFMA GEMM for gfx950 (MI355X), fp32 accumulation.
//
// Structure (v1, register-staged): 128x128 tile, BK=32, 4 waves per block
// (2x2 wave grid, 64x64 per wave as 4x4 fragments of 16x16), double-buffered
// LDS with the load-early/write-late split (guide T14), mfma_f32_16x16x32_bf16
// inner loop. LDS rows padded +16B so the column fragment reads
// (ds_read_b128, one row per lane in a 16-lane group) are bank-conflict-free
// without an XOR swizzle.
//
// Layout handling: both operands are staged into canonical k-contiguous LDS
// images ([rows][BK]); operands whose storage is k-outer ([K,F]) are
// transposed during staging via coalesced 2B loads + packed b128 LDS writes.
// This serves all three cases training needs (fwd NT, dgrad NN, wgrad TN)
// with one inner loop.
//
// Replaces the reference's XLA-codegen GEMMs (all device code
// in TePDist is XLA-generated PTX; here it is hand-written CDNA4).

#include <stdexcept>
#include <string>

#include "common.h"
#include "kernels.h"

namespace tepdist {

namespace {

constexpr int BM = 128, BN = 128, BK = 32;
constexpr int LDS_ROW = BK + 8;  // +8 bf16 = +16B row pad: conflict-free b128
constexpr int NTHREADS = 256;

enum { EPI_NONE = 0, EPI_BIAS = 1, EPI_BIAS_GELU = 2, EPI_GELU = 3 };

// --- staging: k-contiguous operand (stored [F][ld], k inner) ---------------

DEV_INLINE void stage_kc_load(const bf16_t* __restrict__ src, int F, int K,
                              int ld, int f0, int k0, bool aligned,
                              bf16x8 regs[2]) {
#pragma unroll
  for (int u = 0; u < 2; ++u) {
    int idx = threadIdx.x + u * NTHREADS;  // 0..511
    int row = idx >> 2;                    // 0..127
    int kc = (idx & 3) * 8;
    bf16x8 v = {};
    int f = f0 + row, k = k0 + kc;
    if (f < F && k < K) {
      const bf16_t* p = src + (int64_t)f * ld + k;
      if (aligned && k + 8 <= K) {
        v = *reinterpret_cast<const bf16x8*>(p);
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e)
          if (k + e < K) v[e] = p[e];
      }
    }
    regs[u] = v;
  }
}

// --- staging: k-outer operand (stored [K][ld]) -> transpose in registers ---

DEV_INLINE void stage_ko_load(const bf16_t* __restrict__ src, int F, int K,
                              int ld, int f0, int k0, bf16x8 regs[2]) {
  int f = threadIdx.x & 127;
  int kh = threadIdx.x >> 7;  // 0/1: k halves of 16
  int fg = f0 + f;
#pragma unroll
  for (int h = 0; h < 2; ++h) {
    bf16x8 v = {};
    if (fg < F) {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        int k = k0 + kh * 16 + h * 8 + e;
        if (k < K) v[e] = src[(int64_t)k * ld + fg];
      }
    }
    regs[h] = v;
  }
}

DEV_INLINE void stage_write_kc(bf16_t* dst, const bf16x8 regs[2]) {
#pragma unroll
  for (int u = 0; u < 2; ++u) {
    int idx = threadIdx.x + u * NTHREADS;
    int row = idx >> 2;
    int kc = (idx & 3) * 8;
    *reinterpret_cast<bf16x8*>(dst + row * LDS_ROW + kc) = regs[u];
  }
}

DEV_INLINE void stage_write_ko(bf16_t* dst, const bf16x8 regs[2]) {
  int f = threadIdx.x & 127;
  int kh = threadIdx.x >> 7;
  *reinterpret_cast<bf16x8*>(dst + f * LDS_ROW + kh * 16) = regs[0];
  *reinterpret_cast<bf16x8*>(dst + f * LDS_ROW + kh * 16 + 8) = regs[1];
}

template <bool A_KC, bool B_KC, int EPI>
__launch_bounds__(NTHREADS) __global__
void gemm_kernel(const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
                 bf16_t* __restrict__ C, bf16_t* __restrict__ Cpre,
                 const bf16_t* __restrict__ bias, int M, int N, int K,
                 int lda, int ldb, int ldc, int64_t strideA, int64_t strideB,
                 int64_t strideC) {
  A += blockIdx.z * strideA;
  B += blockIdx.z * strideB;
  C += blockIdx.z * strideC;
  if (EPI >= 2) Cpre += blockIdx.z * strideC;

  const int m0 = blockIdx.y * BM, n0 = blockIdx.x * BN;

  __shared__ bf16_t smem[2][2][BM * LDS_ROW];  // [buf][A=0/B=1]

  const int lane = threadIdx.x & 63;
  const int wm = (threadIdx.x >> 7) * 64;        // wave row (wid>>1)*64
  const int wn = ((threadIdx.x >> 6) & 1) * 64;  // wave col (wid&1)*64

  const bool a_al = A_KC && ((lda & 7) == 0);
  const bool b_al = B_KC && ((ldb & 7) == 0);

  f32x4 acc[4][4] = {};

  const int nk = (K + BK - 1) / BK;
  bf16x8 ra[2], rb[2];

  // prologue: stage tile 0
  if (A_KC) stage_kc_load(A, M, K, lda, m0, 0, a_al, ra);
  else      stage_ko_load(A, M, K, lda, m0, 0, ra);
  if (B_KC) stage_kc_load(B, N, K, ldb, n0, 0, b_al, rb);
  else      stage_ko_load(B, N, K, ldb, n0, 0, rb);
  if (A_KC) stage_write_kc(smem[0][0], ra); else stage_write_ko(smem[0][0], ra);
  if (B_KC) stage_write_kc(smem[0][1], rb); else stage_write_ko(smem[0][1], rb);
  __syncthreads();

  for (int t = 0; t < nk; ++t) {
    const int cur = t & 1;
    // issue next tile's global loads early (latency hides under the MFMAs)
    if (t + 1 < nk) {
      const int k0 = (t + 1) * BK;
      if (A_KC) stage_kc_load(A, M, K, lda, m0, k0, a_al, ra);
      else      stage_ko_load(A, M, K, lda, m0, k0, ra);
      if (B_KC) stage_kc_load(B, N, K, ldb, n0, k0, b_al, rb);
      else      stage_ko_load(B, N, K, ldb, n0, k0, rb);
    }

    const bf16_t* sa = smem[cur][0];
    const bf16_t* sb = smem[cur][1];
    bf16x8 af[4], bfr[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      af[i] = *reinterpret_cast<const bf16x8*>(
          sa + (wm + i * 16 + (lane & 15)) * LDS_ROW + 8 * (lane >> 4));
      bfr[i] = *reinterpret_cast<const bf16x8*>(
          sb + (wn + i * 16 + (lane & 15)) * LDS_ROW + 8 * (lane >> 4));
    }
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[mi], bfr[ni], acc[mi][ni], 0, 0, 0);

    if (t + 1 < nk) {
      const int nxt = cur ^ 1;
      if (A_KC) stage_write_kc(smem[nxt][0], ra);
      else      stage_write_ko(smem[nxt][0], ra);
      if (B_KC) stage_write_kc(smem[nxt][1], rb);
      else      stage_write_ko(smem[nxt][1], rb);
    }
    __syncthreads();
  }

  // epilogue: C/D fragment mapping (16x16x32): col = lane&15,
  // row = (lane>>4)*4 + e
  float bv[4];
  if (EPI == EPI_BIAS || EPI == EPI_BIAS_GELU) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int n = n0 + wn + ni * 16 + (lane & 15);
      bv[ni] = (n < N) ? bf2f(bias[n]) : 0.0f;
    }
  }
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      const int n = n0 + wn + ni * 16 + (lane & 15);
      if (n >= N) continue;
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        const int m = m0 + wm + mi * 16 + (lane >> 4) * 4 + e;
        if (m >= M) continue;
        float v = acc[mi][ni][e];
        if (EPI == EPI_BIAS || EPI == EPI_BIAS_GELU) v += bv[ni];
        const int64_t off = (int64_t)m * ldc + n;
        if (EPI >= 2) {
          // round pre-act to bf16 first (matches ops/reference.py semantics)
          bf16_t pre = f2bf(v);
          Cpre[off] = pre;
          v = gelu_f(bf2f(pre));
        }
        C[off] = f2bf(v);
      }
    }
  }
}

}  // namespace

void gemm_bf16(const void* A, const void* B, void* C, void* c_pre,
               const void* bias, int M, int N, int K, int lda, int ldb,
               int ldc, int64_t stride_a, int64_t stride_b, int64_t stride_c,
               int batch, bool a_kc, bool b_kc, int epi, hipStream_t stream) {
  dim3 grid((N + BN - 1) / BN, (M + BM - 1) / BM, batch);
  dim3 block(NTHREADS);
  const bf16_t* a = static_cast<const bf16_t*>(A);
  const bf16_t* b = static_cast<const bf16_t*>(B);
  bf16_t* c = static_cast<bf16_t*>(C);
  bf16_t* cp = static_cast<bf16_t*>(c_pre);
  const bf16_t* bi = static_cast<const bf16_t*>(bias);

#define GEMM_LAUNCH(AKC, BKC, E)                                            \
  hipLaunchKernelGGL((gemm_kernel<AKC, BKC, E>), grid, block, 0, stream, a, \
                     b, c, cp, bi, M, N, K, lda, ldb, ldc, stride_a,        \
                     stride_b, stride_c)

#define GEMM_EPI(AKC, BKC)                         \
  switch (epi) {                                   \
    case 0: GEMM_LAUNCH(AKC, BKC, 0); break;       \
    case 1: GEMM_LAUNCH(AKC, BKC, 1); break;       \
    case 2: GEMM_LAUNCH(AKC, BKC, 2); break;       \
    case 3: GEMM_LAUNCH(AKC, BKC, 3); break;       \
    default: throw std::runtime_error("bad epi");  \
  }

  if (a_kc && b_kc) { GEMM_EPI(true, true); }
  else if (a_kc && !b_kc) { GEMM_EPI(true, false); }
  else if (!a_kc && b_kc) { GEMM_EPI(false, true); }
  else { GEMM_EPI(false, false); }
#undef GEMM_EPI
#undef GEMM_LAUNCH
}

}  // namespace tepdist
