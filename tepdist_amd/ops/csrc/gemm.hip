// bf16 MFMA GEMM for gfx950 (MI355X), fp32 accumulation.
//
// Structure: 128x128 tile, BK=32, 4 waves per block (2x2 wave grid, 64x64
// per wave as 4x4 fragments of 16x16), double-buffered LDS with the
// load-early/write-late split (guide T14), mfma_f32_16x16x32_bf16 inner loop.
//
// Layout handling without transpose cost: the MFMA contraction is invariant
// under any k-permutation applied consistently to both operands' fragments
// (sum over (g,e) of A[m][pi(g,e)]*B[pi(g,e)][n] == sum over k). So:
//  - "KC" operands (k fastest-varying in memory: A stored [M,K], B stored
//    [N,K]) stage into a row-padded [F][BK] image with ds_read_b128 fragment
//    reads (one row per lane in a 16-lane group; +16B row pad keeps it
//    bank-conflict-free).
//  - "KO" operands (k outermost: stored [K,F], i.e. dgrad's weight and both
//    wgrad operands) stage UNtransposed into a [BK/4][BM/16][4][16]-subtiled
//    image using only 16B loads + 16B LDS writes, and fragments are read with
//    the gfx950 hardware transpose-read ds_read_b64_tr_b16 (guide T10).
//  When layouts mix, the KC operand's staging permutes columns so both sides
//  share the tr-read's k order pi(g,e) = e<4 ? 4g+e : 16+4g+(e-4).
//
// Replaces the reference's XLA-codegen GEMMs (all device code
// in TePDist is XLA-generated PTX; here it is hand-written CDNA4).

#include <algorithm>
#include <stdexcept>
#include <string>

#include "common.h"
#include "kernels.h"

namespace tepdist {

namespace {

constexpr int BM = 128, BN = 128, BK = 32;
constexpr int LDS_ROW = BK + 8;  // +8 bf16 = +16B row pad: conflict-free b128
constexpr int SLOT = BM * LDS_ROW;  // elements per operand slot (KO uses 4K)
constexpr int NTHREADS = 256;

enum { EPI_NONE = 0, EPI_BIAS = 1, EPI_BIAS_GELU = 2, EPI_GELU = 3 };

typedef __attribute__((address_space(3))) bf16x4* lds_tr_ptr;

DEV_INLINE bf16x4 tr_read(const bf16_t* p) {
  return __builtin_amdgcn_ds_read_tr16_b64_v4bf16((lds_tr_ptr)p);
}

// --- staging loads ---------------------------------------------------------

// KC operand (stored [F][ld], k inner): 2 x 16B per thread.
DEV_INLINE void stage_kc_load(const bf16_t* __restrict__ src, int F, int K,
                              int ld, int f0, int k0, bool aligned,
                              bf16x8 regs[2]) {
#pragma unroll
  for (int u = 0; u < 2; ++u) {
    const int idx = threadIdx.x + u * NTHREADS;  // 0..511
    const int row = idx >> 2;                    // 0..127
    const int kc = (idx & 3) * 8;
    bf16x8 v = {};
    const int f = f0 + row, k = k0 + kc;
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

// KO operand (stored [K][ld], f inner): 2 x 16B per thread, no transpose.
DEV_INLINE void stage_ko_load(const bf16_t* __restrict__ src, int F, int K,
                              int ld, int f0, int k0, bool aligned,
                              bf16x8 regs[2]) {
#pragma unroll
  for (int u = 0; u < 2; ++u) {
    const int idx = threadIdx.x + u * NTHREADS;  // 0..511
    const int k = idx >> 4;                      // 0..31
    const int f8 = (idx & 15) * 8;               // 0..120
    bf16x8 v = {};
    const int kg = k0 + k, fg = f0 + f8;
    if (kg < K && fg < F) {
      const bf16_t* p = src + (int64_t)kg * ld + fg;
      if (aligned && fg + 8 <= F) {
        v = *reinterpret_cast<const bf16x8*>(p);
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e)
          if (fg + e < F) v[e] = p[e];
      }
    }
    regs[u] = v;
  }
}

// --- staging writes --------------------------------------------------------

// natural k order (used when both operands are KC)
DEV_INLINE void stage_write_kc_natural(bf16_t* dst, const bf16x8 regs[2]) {
#pragma unroll
  for (int u = 0; u < 2; ++u) {
    const int idx = threadIdx.x + u * NTHREADS;
    const int row = idx >> 2;
    const int kc = (idx & 3) * 8;
    *reinterpret_cast<bf16x8*>(dst + row * LDS_ROW + kc) = regs[u];
  }
}

// pi-permuted columns (KC operand mixed with a tr-read operand): global k
// run [8j,8j+8) splits into two b64 writes at cols 16*(j%2)+4*(j/2) and +8.
DEV_INLINE void stage_write_kc_pi(bf16_t* dst, const bf16x8 regs[2]) {
#pragma unroll
  for (int u = 0; u < 2; ++u) {
    const int idx = threadIdx.x + u * NTHREADS;
    const int row = idx >> 2;
    const int j = idx & 3;
    const int col0 = 16 * (j & 1) + 4 * (j >> 1);
    const bf16x8 v = regs[u];
    bf16x4 lo = {v[0], v[1], v[2], v[3]};
    bf16x4 hi = {v[4], v[5], v[6], v[7]};
    *reinterpret_cast<bf16x4*>(dst + row * LDS_ROW + col0) = lo;
    *reinterpret_cast<bf16x4*>(dst + row * LDS_ROW + col0 + 8) = hi;
  }
}

// KO operand image: [F/4 panels][BK rows][4 f] — element (k,f) at
// panel=f/4, offset (panel*BK + k')*4 + f%4 where k' swizzles 4-row blocks
// (k' = (k/4 ^ (panel&7))*4 + k%4) to spread panels over LDS banks.
// ds_read_b64_tr_b16 semantics (measured, tests/test_kernels_gpu.py
// tr16 probe): each QUAD of lanes reads a 4x4 bf16 block at the quad's
// address with 8-byte row stride; lane i of the quad receives column i.
// A panel's 4-row block is exactly such a block (rows 8B apart), so the
// quad read returns 4 k's at a fixed f with zero staging transpose.
DEV_INLINE int ko_elem_off(int k, int f) {
  const int panel = f >> 2;
  const int kb = ((k >> 2) ^ (panel & 7)) << 2;
  return (panel * BK + kb + (k & 3)) * 4 + (f & 3);
}

DEV_INLINE void stage_write_ko(bf16_t* dst, const bf16x8 regs[2]) {
#pragma unroll
  for (int u = 0; u < 2; ++u) {
    const int idx = threadIdx.x + u * NTHREADS;
    const int k = idx >> 4;
    const int f8 = (idx & 15) * 8;
    const bf16x8 v = regs[u];
    const bf16x4 lo = {v[0], v[1], v[2], v[3]};
    const bf16x4 hi = {v[4], v[5], v[6], v[7]};
    *reinterpret_cast<bf16x4*>(dst + ko_elem_off(k, f8)) = lo;
    *reinterpret_cast<bf16x4*>(dst + ko_elem_off(k, f8 + 4)) = hi;
  }
}

// --- fragment loads --------------------------------------------------------

// KC image (natural or pi — content differs, read pattern identical)
DEV_INLINE bf16x8 frag_kc(const bf16_t* s, int fbase, int lane) {
  return *reinterpret_cast<const bf16x8*>(
      s + (fbase + (lane & 15)) * LDS_ROW + 8 * (lane >> 4));
}

// KO image via two quad transpose-reads; k order = pi(g,e): read 1 gives
// k = 4g+j (logical block g), read 2 gives k = 16+4g+j (block 4+g).
DEV_INLINE bf16x8 frag_ko(const bf16_t* s, int fbase, int lane) {
  const int g = lane >> 4;
  const int i = lane & 3;                    // column within the quad
  const int panel = (fbase >> 2) + ((lane >> 2) & 3);
  const int kb1 = (g ^ (panel & 7)) << 2;
  const int kb2 = ((4 + g) ^ (panel & 7)) << 2;
  const bf16_t* p0 = s + (panel * BK + kb1) * 4 + i;
  const bf16_t* p1 = s + (panel * BK + kb2) * 4 + i;
  const bf16x4 lo = tr_read(p0);
  const bf16x4 hi = tr_read(p1);
  return bf16x8{lo[0], lo[1], lo[2], lo[3], hi[0], hi[1], hi[2], hi[3]};
}

template <bool A_KC, bool B_KC, int EPI>
__launch_bounds__(NTHREADS) __global__
void gemm_kernel(const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
                 bf16_t* __restrict__ C, bf16_t* __restrict__ Cpre,
                 const bf16_t* __restrict__ bias, int M, int N, int K,
                 int lda, int ldb, int ldc, int64_t strideA, int64_t strideB,
                 int64_t strideC) {
  constexpr bool NATURAL = A_KC && B_KC;
  A += blockIdx.z * strideA;
  B += blockIdx.z * strideB;
  C += blockIdx.z * strideC;
  if (EPI >= 2) Cpre += blockIdx.z * strideC;

  const int m0 = blockIdx.y * BM, n0 = blockIdx.x * BN;

  __shared__ bf16_t smem[2][2][SLOT];  // [buf][A=0/B=1]

  const int lane = threadIdx.x & 63;
  const int wm = (threadIdx.x >> 7) * 64;        // wave row (wid>>1)*64
  const int wn = ((threadIdx.x >> 6) & 1) * 64;  // wave col (wid&1)*64

  const bool a_al = (lda & 7) == 0;
  const bool b_al = (ldb & 7) == 0;

  f32x4 acc[4][4] = {};

  const int nk = (K + BK - 1) / BK;
  bf16x8 ra[2], rb[2];

  auto load_tiles = [&](int k0) {
    if (A_KC) stage_kc_load(A, M, K, lda, m0, k0, a_al, ra);
    else      stage_ko_load(A, M, K, lda, m0, k0, a_al, ra);
    if (B_KC) stage_kc_load(B, N, K, ldb, n0, k0, b_al, rb);
    else      stage_ko_load(B, N, K, ldb, n0, k0, b_al, rb);
  };
  auto write_tiles = [&](int buf) {
    if (A_KC) {
      if (NATURAL) stage_write_kc_natural(smem[buf][0], ra);
      else         stage_write_kc_pi(smem[buf][0], ra);
    } else {
      stage_write_ko(smem[buf][0], ra);
    }
    if (B_KC) {
      if (NATURAL) stage_write_kc_natural(smem[buf][1], rb);
      else         stage_write_kc_pi(smem[buf][1], rb);
    } else {
      stage_write_ko(smem[buf][1], rb);
    }
  };

  load_tiles(0);
  write_tiles(0);
  __syncthreads();

  for (int t = 0; t < nk; ++t) {
    const int cur = t & 1;
    if (t + 1 < nk) load_tiles((t + 1) * BK);  // issue early (T14)

    const bf16_t* sa = smem[cur][0];
    const bf16_t* sb = smem[cur][1];
    bf16x8 af[4], bfr[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      af[i] = A_KC ? frag_kc(sa, wm + i * 16, lane) : frag_ko(sa, wm + i * 16, lane);
      bfr[i] = B_KC ? frag_kc(sb, wn + i * 16, lane) : frag_ko(sb, wn + i * 16, lane);
    }
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[mi], bfr[ni], acc[mi][ni], 0, 0, 0);

    if (t + 1 < nk) write_tiles(cur ^ 1);
    __syncthreads();
  }

  // epilogue: C/D fragment mapping (16x16x32): col = lane&15,
  // row = (lane>>4)*4 + e
  float bv[4];
  if (EPI == EPI_BIAS || EPI == EPI_BIAS_GELU) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      const int n = n0 + wn + ni * 16 + (lane & 15);
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
          const bf16_t pre = f2bf(v);
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
