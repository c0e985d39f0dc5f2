// bf16 MFMA GEMM for gfx950 (MI355X), fp32 accumulation.
//
// Structure: 128x128 tile, BK=32, 4 waves per block (2x2 wave grid, 64x64
// per wave as 4x4 fragments of 16x16), double-buffered LDS with the
// load-early/write-late split (guide T14), mfma_f32_16x16x32_bf16 inner loop.
//
// Layout handling: both operands stage into the canonical k-contiguous
// row-padded [F][BK] LDS image (ds_read_b128 fragment reads, one row per
// lane in a 16-lane group; the +16B row pad keeps them bank-conflict-free).
//  - "KC" operands (k fastest-varying in memory: A stored [M,K], B stored
//    [N,K]): straight 16B loads + 16B LDS writes.
//  - "KO" operands (k outermost: stored [K,F] — dgrad's weight and both
//    wgrad operands): each thread loads an 8(k) x 2(f) slab with eight 4B
//    loads (f-coalesced across the wave), transposes it in registers with
//    eight v_perm_b32, and writes two k-contiguous 16B rows. No scalar LDS
//    traffic. (ds_read_b64_tr_b16 was measured — see debug.hip tr16_probe —
//    to deliver only 16 distinct values per 16-lane group: each quad reads
//    the (lane&3)-th bf16 of 4 rows based at the group's quad-LEADER
//    addresses, so it cannot feed a 16x16 MFMA fragment; the register
//    transpose path is the fast one.)
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

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int SLOT = BM * BK;  // elements per operand slot
constexpr int NTHREADS = 256;

// LDS addressing: unpadded 128B rows + an XOR swizzle of the 16B column
// slot by x(row) = bit-reversed (row>>1)&7. The bit reversal puts the
// fragment-read service groups' row bit (row bit 1 under quad-strided
// 16-lane grouping, row bits 1..3 under contiguous grouping) into slot
// bits that the reads' own column spread does not already cover, making
// ds_read_b128 column reads conflict-free; writes stay <=2-way.
DEV_INLINE int lds_off(int row, int col_e) {
  const int r1 = row >> 1;
  const int x = ((r1 & 1) << 2) | (r1 & 2) | ((r1 >> 2) & 1);
  return row * BK + (col_e ^ (x << 3));
}

enum { EPI_NONE = 0, EPI_BIAS = 1, EPI_BIAS_GELU = 2, EPI_GELU = 3 };

// --- staging loads ---------------------------------------------------------

// KC operand (stored [F][ld], k inner): 2 x 16B per thread.
DEV_INLINE void stage_kc_load(const bf16_t* __restrict__ src, int F, int K,
                              int ld, int f0, int k0, bool aligned,
                              bf16x8 regs[4]) {
#pragma unroll
  for (int u = 0; u < 4; ++u) {
    const int idx = threadIdx.x + u * NTHREADS;  // 0..1023
    const int row = idx >> 3;                    // 0..127
    const int kc = (idx & 7) * 8;
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

// KO operand (stored [K][ld], f inner): thread t owns the 8(k) x 2(f) slab
// at f = 2*(t&63), k-block = t>>6; eight bf16x2 loads, f-coalesced.
DEV_INLINE void stage_ko_load(const bf16_t* __restrict__ src, int F, int K,
                              int ld, int f0, int k0, bool aligned,
                              bf16x2 regs[2][8]) {
  const int f = f0 + 2 * (threadIdx.x & 63);
#pragma unroll
  for (int h = 0; h < 2; ++h) {
    const int kt = k0 + 8 * ((threadIdx.x >> 6) + 4 * h);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int k = kt + j;
      bf16x2 v = {};
      if (k < K && f < F) {
        const bf16_t* p = src + (int64_t)k * ld + f;
        if (aligned && f + 2 <= F) {
          v = *reinterpret_cast<const bf16x2*>(p);
        } else {
          v[0] = p[0];
          if (f + 1 < F) v[1] = p[1];
        }
      }
      regs[h][j] = v;
    }
  }
}

// KC operand via global_load_lds (direct HBM->LDS DMA, 16B per lane; the
// guide's step-3 lever). The glds destination is wave-uniform base + lane*16B
// (lane-linear), so the XOR swizzle moves to the per-lane SOURCE address
// (rule 21): LDS element (row, c) receives global k = c ^ (x(row)<<3).
// Only for fully-interior tiles (no bounds handling in the DMA).
typedef __attribute__((address_space(1))) const void* glds_src_t;
typedef __attribute__((address_space(3))) void* glds_dst_t;

DEV_INLINE void stage_kc_glds(const bf16_t* __restrict__ src, int ld, int f0,
                              int k0, bf16_t* dst) {
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
#pragma unroll
  for (int u = 0; u < 4; ++u) {
    const int idx = threadIdx.x + u * NTHREADS;
    const int row = idx >> 3;                 // dest row (lane-linear)
    const int c = (idx & 7) * 8;              // dest column (elements)
    const int r1 = row >> 1;
    const int x = ((r1 & 1) << 2) | (r1 & 2) | ((r1 >> 2) & 1);
    const int ksrc = k0 + (c ^ (x << 3));     // inverse swizzle on the source
    const bf16_t* g = src + (int64_t)(f0 + row) * ld + ksrc;
    bf16_t* l = dst + (8 * wave + 32 * u) * BK;  // wave-uniform base
    __builtin_amdgcn_global_load_lds((glds_src_t)g, (glds_dst_t)l, 16,
                                     /*offset=*/0, /*aux=*/0);
  }
}

// --- staging writes --------------------------------------------------------

// natural k order (used when both operands are KC)
DEV_INLINE void stage_write_kc_natural(bf16_t* dst, const bf16x8 regs[4]) {
#pragma unroll
  for (int u = 0; u < 4; ++u) {
    const int idx = threadIdx.x + u * NTHREADS;
    const int row = idx >> 3;
    const int kc = (idx & 7) * 8;
    *reinterpret_cast<bf16x8*>(dst + lds_off(row, kc)) = regs[u];
  }
}

// KO slab write: transpose 8x2 in registers (v_perm_b32), two b128 writes.
DEV_INLINE void stage_write_ko(bf16_t* dst, const bf16x2 regs[2][8]) {
  const int f = 2 * (threadIdx.x & 63);
#pragma unroll
  for (int h = 0; h < 2; ++h) {
    const int kc = 8 * ((threadIdx.x >> 6) + 4 * h);
    uint32_t r[8];
#pragma unroll
    for (int j = 0; j < 8; ++j)
      r[j] = __builtin_bit_cast(uint32_t, regs[h][j]);
    uint32_t o0[4], o1[4];
#pragma unroll
    for (int d = 0; d < 4; ++d) {
      // result bytes [A0,A1,B0,B1]: A=r[2d] (src1=bytes 0-3), B=r[2d+1]
      o0[d] = __builtin_amdgcn_perm(r[2 * d + 1], r[2 * d], 0x05040100u);
      o1[d] = __builtin_amdgcn_perm(r[2 * d + 1], r[2 * d], 0x07060302u);
    }
    *reinterpret_cast<uint4*>(dst + lds_off(f, kc)) =
        make_uint4(o0[0], o0[1], o0[2], o0[3]);
    *reinterpret_cast<uint4*>(dst + lds_off(f + 1, kc)) =
        make_uint4(o1[0], o1[1], o1[2], o1[3]);
  }
}

// --- fragment loads --------------------------------------------------------

// fragment read from the canonical [F][LDS_ROW] image
DEV_INLINE bf16x8 frag(const bf16_t* s, int fbase, int lane, int kk) {
  return *reinterpret_cast<const bf16x8*>(
      s + lds_off(fbase + (lane & 15), 8 * (lane >> 4) + 32 * kk));
}

// F32OUT: split-K mode — blockIdx.z selects a K-chunk (k_chunk elements)
// of batch-0 operands and the epilogue writes fp32 partials at
// C + z*strideC (reduced by splitk_reduce_kernel).
template <bool A_KC, bool B_KC, int EPI, bool F32OUT = false>
__launch_bounds__(NTHREADS) __global__
void gemm_kernel(const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
                 void* __restrict__ Cv, bf16_t* __restrict__ Cpre,
                 const bf16_t* __restrict__ bias, int M, int N, int K,
                 int lda, int ldb, int ldc, int64_t strideA, int64_t strideB,
                 int64_t strideC, int k_chunk) {
  bf16_t* C = static_cast<bf16_t*>(Cv);
  float* Cf = static_cast<float*>(Cv);
  int kbeg = 0, kend = K;
  if (F32OUT) {
    kbeg = blockIdx.z * k_chunk;
    kend = min(K, kbeg + k_chunk);
    Cf += blockIdx.z * strideC;
  } else {
    A += blockIdx.z * strideA;
    B += blockIdx.z * strideB;
    C += blockIdx.z * strideC;
    if (EPI >= 2) Cpre += blockIdx.z * strideC;
  }

  // XCD-aware bijective block swizzle (guide T1): the dispatcher places
  // block b on XCD b%8; remap so each XCD gets a CONTIGUOUS run of tiles
  // (consecutive n-tiles share the A panel -> L2 hits stay on one XCD).
  const int nbx = (N + BN - 1) / BN;
  const int nwg = gridDim.x;
  const int q = nwg >> 3, r = nwg & 7;
  const int xcd = blockIdx.x & 7, idx = blockIdx.x >> 3;
  const int swz = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  const int m0 = (swz / nbx) * BM, n0 = (swz % nbx) * BN;

  __shared__ bf16_t smem[2][2][SLOT];  // [buf][A=0/B=1]

  const int lane = threadIdx.x & 63;
  const int wm = (threadIdx.x >> 7) * 64;        // wave row (wid>>1)*64
  const int wn = ((threadIdx.x >> 6) & 1) * 64;  // wave col (wid&1)*64

  const bool a_al = (lda & 7) == 0;   // 16B loads (KC staging)
  const bool b_al = (ldb & 7) == 0;
  const bool a_al2 = (lda & 1) == 0;  // 4B loads (KO staging)
  const bool b_al2 = (ldb & 1) == 0;

  f32x4 acc[4][4] = {};

  const int nk = (kend - kbeg + BK - 1) / BK;
  bf16x8 ra_kc[4], rb_kc[4];
  bf16x2 ra_ko[2][8], rb_ko[2][8];

  // interior tiles of aligned KC operands go by global_load_lds DMA
  const bool a_glds = A_KC && a_al && (m0 + BM <= M);
  const bool b_glds = B_KC && b_al && (n0 + BN <= N);

  auto load_tiles = [&](int k0, int buf) {
    const bool kin = k0 + BK <= kend;
    if (A_KC) {
      if (a_glds && kin) stage_kc_glds(A, lda, m0, k0, smem[buf][0]);
      else               stage_kc_load(A, M, kend, lda, m0, k0, a_al, ra_kc);
    } else {
      stage_ko_load(A, M, kend, lda, m0, k0, a_al2, ra_ko);
    }
    if (B_KC) {
      if (b_glds && kin) stage_kc_glds(B, ldb, n0, k0, smem[buf][1]);
      else               stage_kc_load(B, N, kend, ldb, n0, k0, b_al, rb_kc);
    } else {
      stage_ko_load(B, N, kend, ldb, n0, k0, b_al2, rb_ko);
    }
  };
  auto write_tiles = [&](int buf, int k0) {
    const bool kin = k0 + BK <= kend;
    if (A_KC) {
      if (!(a_glds && kin)) stage_write_kc_natural(smem[buf][0], ra_kc);
    } else {
      stage_write_ko(smem[buf][0], ra_ko);
    }
    if (B_KC) {
      if (!(b_glds && kin)) stage_write_kc_natural(smem[buf][1], rb_kc);
    } else {
      stage_write_ko(smem[buf][1], rb_ko);
    }
  };

  load_tiles(kbeg, 0);
  write_tiles(0, kbeg);
  __syncthreads();

  for (int t = 0; t < nk; ++t) {
    const int cur = t & 1;
    if (t + 1 < nk)
      load_tiles(kbeg + (t + 1) * BK, cur ^ 1);  // issue early (T14)

    const bf16_t* sa = smem[cur][0];
    const bf16_t* sb = smem[cur][1];
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 af[4], bfr[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        af[i] = frag(sa, wm + i * 16, lane, kk);
        bfr[i] = frag(sb, wn + i * 16, lane, kk);
      }
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bfr[ni], acc[mi][ni], 0, 0, 0);
    }

    if (t + 1 < nk) write_tiles(cur ^ 1, kbeg + (t + 1) * BK);
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
        if (F32OUT) {
          Cf[off] = v;
          continue;
        }
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

__global__ void splitk_reduce_kernel(const float* __restrict__ parts,
                                     bf16_t* __restrict__ out, int nparts,
                                     int64_t mn) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * 4;
  for (int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 4;
       i < mn; i += stride) {
    if (i + 4 <= mn) {
      f32x4 sv = *reinterpret_cast<const f32x4*>(parts + i);
      for (int p = 1; p < nparts; ++p) {
        const f32x4 v = *reinterpret_cast<const f32x4*>(parts + p * mn + i);
#pragma unroll
        for (int e = 0; e < 4; ++e) sv[e] += v[e];
      }
      bf16x4 o;
#pragma unroll
      for (int e = 0; e < 4; ++e) o[e] = f2bf(sv[e]);
      *reinterpret_cast<bf16x4*>(out + i) = o;
    } else {
      for (int64_t j = i; j < mn; ++j) {
        float acc = 0.f;
        for (int p = 0; p < nparts; ++p) acc += parts[p * mn + j];
        out[j] = f2bf(acc);
      }
    }
  }
}

}  // namespace

void splitk_reduce(const float* parts, void* out, int nparts, int64_t mn,
                   hipStream_t stream) {
  const int blocks = (int)std::min<int64_t>(
      (mn / 4 + NTHREADS - 1) / NTHREADS, 2048);
  hipLaunchKernelGGL(splitk_reduce_kernel, dim3(std::max(blocks, 1)),
                     dim3(NTHREADS), 0, stream, parts,
                     static_cast<bf16_t*>(out), nparts, mn);
}

void gemm_bf16(const void* A, const void* B, void* C, void* c_pre,
               const void* bias, int M, int N, int K, int lda, int ldb,
               int ldc, int64_t stride_a, int64_t stride_b, int64_t stride_c,
               int batch, bool a_kc, bool b_kc, int epi, int split_k,
               hipStream_t stream) {
  const int nbx = (N + BN - 1) / BN, nby = (M + BM - 1) / BM;
  dim3 block(NTHREADS);
  const bf16_t* a = static_cast<const bf16_t*>(A);
  const bf16_t* b = static_cast<const bf16_t*>(B);
  bf16_t* cp = static_cast<bf16_t*>(c_pre);
  const bf16_t* bi = static_cast<const bf16_t*>(bias);

  if (split_k > 1) {
    if (batch != 1 || epi != 0)
      throw std::runtime_error("split_k requires batch=1, epi=0");
    if (gemm256_supported(M, N, K, lda, ldb, a_kc, b_kc, epi, split_k)) {
      gemm256_bf16(A, B, C, c_pre, bias, M, N, K, lda, ldb, ldc, stride_a,
                   stride_b, stride_c, batch, epi, split_k, stream);
      return;
    }
    const int k_chunk = ((K + split_k - 1) / split_k + BK - 1) / BK * BK;
    dim3 gridk(nbx * nby, 1, split_k);
#define GEMM_SK(AKC, BKC)                                                   \
    hipLaunchKernelGGL((gemm_kernel<AKC, BKC, 0, true>), gridk, block, 0,   \
                       stream, a, b, C, cp, bi, M, N, K, lda, ldb, ldc,     \
                       stride_a, stride_b, stride_c, k_chunk)
    if (a_kc && b_kc) { GEMM_SK(true, true); }
    else if (a_kc && !b_kc) { GEMM_SK(true, false); }
    else if (!a_kc && b_kc) { GEMM_SK(false, true); }
    else { GEMM_SK(false, false); }
#undef GEMM_SK
    return;
  }

  if (gemm256_supported(M, N, K, lda, ldb, a_kc, b_kc, epi, 1)) {
    gemm256_bf16(A, B, C, c_pre, bias, M, N, K, lda, ldb, ldc, stride_a,
                 stride_b, stride_c, batch, epi, 1, stream);
    return;
  }

  dim3 grid(nbx * nby, 1, batch);
#define GEMM_LAUNCH(AKC, BKC, E)                                            \
  hipLaunchKernelGGL((gemm_kernel<AKC, BKC, E>), grid, block, 0, stream, a, \
                     b, C, cp, bi, M, N, K, lda, ldb, ldc, stride_a,        \
                     stride_b, stride_c, 0)

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
