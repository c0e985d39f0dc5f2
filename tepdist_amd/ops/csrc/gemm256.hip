// Deep-pipelined 256x256 bf16 GEMM for gfx950 (the guide's 8-phase
// counted-vmcnt template, re-derived for this kernel set).
//
// Geometry: 256x256 tile, BK=64, 8 waves (2M x 4N), per-wave output
// 128(M) x 4x16(N, STRIDED: wave wn owns n = p*64 + wn*16 for phase
// p=0..3). K-loop phase p computes all 8 m-fragments against the phase's
// n-fragment (16 MFMAs); A fragments (16 x bf16x8) load once per K-tile at
// phase 0 and live in registers.
//
// Staging: global_load_lds only, in 8 KiB QUARTER units (64 rows x 64 cols,
// one dwordx4 per thread), 8 units per K-tile, two staged per phase, slots
// cycling mod 8 per operand (2 tiles of LDS per operand = 128 KiB total).
// The schedule guarantees each unit's slot had its last reader one raw
// barrier earlier: A(t) quarters are consumed entirely at t.ph0 (register
// A), B(t,q_p) at t.ph_p; stages run A(t+2,q01)@t.ph1, A(t+2,q23)@t.ph2,
// B(t+2,q01)@t.ph3, B(t+2,q23)@(t+1).ph0. All barriers are RAW s_barrier
// (no vmcnt drain: __syncthreads would wait the in-flight DMAs to 0) and a
// single counted s_waitcnt vmcnt(6) + barrier per K-tile boundary keeps 6
// quarter-units (3 "half-tiles") in flight.
//
// Handles only the canonical case (both operands k-contiguous, M,N % 256
// == 0, K % 64 == 0, 16B-aligned rows); gemm.hip covers the rest and
// linear_bwd canonicalizes dgrad/wgrad into this case via transposes.

#include <algorithm>
#include <stdexcept>

#include "common.h"
#include "kernels.h"

namespace tepdist {

namespace {

constexpr int BM = 256, BN = 256, BK = 64;
constexpr int NTH = 512;  // 8 waves
constexpr int QROWS = 64;          // stage-unit rows
constexpr int SLOT_E = QROWS * BK; // elements per slot (8 KiB)

enum { EPI_NONE = 0, EPI_BIAS = 1, EPI_BIAS_GELU = 2, EPI_GELU = 3,
       // dgelu: C = (A@B) * gelu'(aux) — the Cpre pointer is the INPUT
       // pre-activation saved by the forward (the fusion hipBLASLt has
       // no algorithms for on gfx950, profiles/blaslt_epilogue_probe.md)
       EPI_DGELU = 4 };

typedef __attribute__((address_space(1))) const void* gsrc_t;
typedef __attribute__((address_space(3))) void* gdst_t;

// row swizzle within a [64][64] slot (see gemm.hip lds_off)
DEV_INLINE int qoff(int row, int col_e) {
  const int r1 = row >> 1;
  const int x = ((r1 & 1) << 2) | (r1 & 2) | ((r1 >> 2) & 1);
  return row * BK + (col_e ^ (x << 3));
}

// stage one quarter unit: rows [f0, f0+64) x cols [k0, k0+64) of a
// k-contiguous operand into slot `dst` (one glds dwordx4 per thread).
DEV_INLINE void stage_q(const bf16_t* __restrict__ src, int64_t ld, int f0,
                        int k0, bf16_t* dst) {
  const int wave = threadIdx.x >> 6;
  const int row = threadIdx.x >> 3;          // 0..63 (lane-linear per wave)
  const int c = (threadIdx.x & 7) * 8;
  const int r1 = row >> 1;
  const int x = ((r1 & 1) << 2) | (r1 & 2) | ((r1 >> 2) & 1);
  const int ksrc = k0 + (c ^ (x << 3));      // inverse swizzle on source
  const bf16_t* g = src + (int64_t)(f0 + row) * ld + ksrc;
  bf16_t* l = dst + (8 * wave) * BK;         // wave-uniform base
  __builtin_amdgcn_global_load_lds((gsrc_t)g, (gdst_t)l, 16, 0, 0);
}

// f32 image variant of the swizzle (16B slot = 4 floats)
DEV_INLINE int qoff_f32(int row, int col_e) {
  const int r1 = row >> 1;
  const int x = ((r1 & 1) << 2) | (r1 & 2) | ((r1 >> 2) & 1);
  return row * BK + (col_e ^ ((x << 2) & 63));
}

DEV_INLINE bf16x8 fragq(const bf16_t* slot, int row, int col_e) {
  return *reinterpret_cast<const bf16x8*>(slot + qoff(row, col_e));
}

#define RAW_BAR() __builtin_amdgcn_s_barrier()
#define VMCNT6() asm volatile("s_waitcnt vmcnt(6)" ::: "memory")

template <int EPI, bool F32OUT = false>
__launch_bounds__(NTH, 1) __global__
void gemm256_kernel(const bf16_t* __restrict__ A,
                    const bf16_t* __restrict__ B, void* __restrict__ Cv,
                    bf16_t* __restrict__ Cpre,
                    const bf16_t* __restrict__ bias, int M, int N, int K,
                    int lda, int ldb, int ldc, int64_t strideA,
                    int64_t strideB, int64_t strideC, int k_chunk) {
  bf16_t* C = static_cast<bf16_t*>(Cv);
  float* Cf = static_cast<float*>(Cv);
  int kbeg = 0, kend = K;
  if (F32OUT) {  // split-K: blockIdx.z = K-chunk, fp32 partial output
    kbeg = blockIdx.z * k_chunk;
    kend = min(K, kbeg + k_chunk);
    Cf += blockIdx.z * strideC;
    if (kend <= kbeg) kbeg = kend = 0;  // dead chunk: no loads, zero out
  } else {
    A += blockIdx.z * strideA;
    B += blockIdx.z * strideB;
    C += blockIdx.z * strideC;
    if (EPI >= 2) Cpre += blockIdx.z * strideC;
  }

  // XCD-aware bijective swizzle (guide T1)
  const int nbx = N / BN;
  const int nwg = gridDim.x;
  const int qd = nwg >> 3, r = nwg & 7;
  const int xcd = blockIdx.x & 7, idx = blockIdx.x >> 3;
  const int swz = (xcd < r ? xcd * (qd + 1) : r * (qd + 1) + (xcd - r) * qd)
                  + idx;
  const int m0 = (swz / nbx) * BM, n0 = (swz % nbx) * BN;

  __shared__ bf16_t smem[16 * SLOT_E];  // A slots 0..7, B slots 8..15
  bf16_t* const aslot = smem;
  bf16_t* const bslot = smem + 8 * SLOT_E;

  const int lane = threadIdx.x & 63;
  const int wm = (threadIdx.x >> 8) & 1;       // wave row (0..1)
  const int wn = (threadIdx.x >> 6) & 3;       // wave col (0..3)

  f32x4 acc[8][4] = {};

  const int nk = (kend - kbeg) / BK;

  // A quarter q of tile t covers rows m0 + 64q; B quarter: n0 + 64q.
  auto stage_a = [&](int t, int q2) {  // stage quarters q2 and q2+1
    stage_q(A, lda, m0 + 64 * q2, kbeg + t * BK,
            aslot + ((4 * t + q2) & 7) * SLOT_E);
    stage_q(A, lda, m0 + 64 * (q2 + 1), kbeg + t * BK,
            aslot + ((4 * t + q2 + 1) & 7) * SLOT_E);
  };
  auto stage_b = [&](int t, int q2) {
    stage_q(B, ldb, n0 + 64 * q2, kbeg + t * BK,
            bslot + ((4 * t + q2) & 7) * SLOT_E);
    stage_q(B, ldb, n0 + 64 * (q2 + 1), kbeg + t * BK,
            bslot + ((4 * t + q2 + 1) & 7) * SLOT_E);
  };

  // prologue: tile 0 fully + tile 1's A and B q0,q1 (14 units in flight),
  // then complete tile 0 (vmcnt 6) before its reads.
  if (nk > 0) {
    stage_a(0, 0);
    stage_a(0, 2);
    stage_b(0, 0);
    stage_b(0, 2);
    if (nk > 1) {
      stage_a(1, 0);
      stage_a(1, 2);
      stage_b(1, 0);
    }
    VMCNT6();
  }
  RAW_BAR();

  bf16x8 af[8][2];
  for (int t = 0; t < nk; ++t) {
    if (t + 1 == nk && nk > 1) {
      // last tile: its B q2/q3 units (staged at (t-1).ph0) are the newest
      // in flight and the counted vmcnt(6) protocol never covers them —
      // nothing is staged after them to push them past the watermark.
      // Drain fully once per block before the final tile's reads.
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      RAW_BAR();
    }
    // ---- phase 0: read ALL A fragments + B fragment 0; stage B(t+1,q23)
    {
      const bf16_t* as0 = aslot + ((4 * t + 2 * wm) & 7) * SLOT_E;
      const bf16_t* as1 = aslot + ((4 * t + 2 * wm + 1) & 7) * SLOT_E;
#pragma unroll
      for (int mi = 0; mi < 8; ++mi) {
        const bf16_t* s = (mi < 4) ? as0 : as1;
        const int row = (mi & 3) * 16 + (lane & 15);
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          af[mi][kk] = fragq(s, row, 8 * (lane >> 4) + 32 * kk);
      }
      bf16x8 bf0[2];
      {
        const bf16_t* bs = bslot + ((4 * t + 0) & 7) * SLOT_E;
        const int row = wn * 16 + (lane & 15);
        bf0[0] = fragq(bs, row, 8 * (lane >> 4));
        bf0[1] = fragq(bs, row, 8 * (lane >> 4) + 32);
      }
      if (t + 1 < nk) stage_b(t + 1, 2);
      RAW_BAR();
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mi = 0; mi < 8; ++mi)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          acc[mi][0] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi][kk], bf0[kk], acc[mi][0], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      RAW_BAR();
    }
    // ---- phases 1..3: read B fragment p; stage per schedule
#pragma unroll
    for (int p = 1; p < 4; ++p) {
      bf16x8 bfp[2];
      {
        const bf16_t* bs = bslot + ((4 * t + p) & 7) * SLOT_E;
        const int row = wn * 16 + (lane & 15);
        bfp[0] = fragq(bs, row, 8 * (lane >> 4));
        bfp[1] = fragq(bs, row, 8 * (lane >> 4) + 32);
      }
      if (p == 1 && t + 2 < nk) stage_a(t + 2, 0);
      if (p == 2 && t + 2 < nk) stage_a(t + 2, 2);
      if (p == 3 && t + 2 < nk) stage_b(t + 2, 0);
      RAW_BAR();
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mi = 0; mi < 8; ++mi)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          acc[mi][p] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi][kk], bfp[kk], acc[mi][p], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      if (p == 3) VMCNT6();  // tile boundary: t+1 fully resident
      RAW_BAR();
    }
  }

  // ---- epilogue (phase-p fragment n-columns: n = 64p + wn*16) ----
  float bv[4];
  if (EPI == EPI_BIAS || EPI == EPI_BIAS_GELU) {
#pragma unroll
    for (int p = 0; p < 4; ++p)
      bv[p] = bf2f(bias[n0 + 64 * p + wn * 16 + (lane & 15)]);
  }
  if (!F32OUT) {
    // Bounce the output through LDS so global stores are whole 128-byte
    // cachelines: the C-fragment layout is column-strided (per lane, the
    // four accumulator elements are CONSECUTIVE ROWS of one column), so a
    // direct scatter is 2-byte stores in 32-byte segments — at K ~ 1k the
    // C write-out dominates per-block overhead. One 64-column phase at a
    // time: waves deposit fragments into a swizzled [256][64] image, then
    // all 512 threads store it row-major (8 lanes = one full row).
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    // gelu epilogues stage only the PRE-activation image; gelu is applied
    // at store time on the vector chunks (halves the LDS traffic of the
    // dual-output fc epilogue)
    bf16_t* img = smem;                  // [256][64]
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const int cl = wn * 16 + (lane & 15);
#pragma unroll
      for (int mi = 0; mi < 8; ++mi)
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          const int row = wm * 128 + mi * 16 + (lane >> 4) * 4 + e;
          float v = acc[mi][p][e];
          if (EPI == EPI_BIAS || EPI == EPI_BIAS_GELU) v += bv[p];
          img[qoff(row, cl)] = f2bf(v);
        }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
      const int sr = threadIdx.x >> 3;         // 64 rows per round
      const int c8 = (threadIdx.x & 7) * 8;    // 16B chunk within the row
#pragma unroll
      for (int rr = 0; rr < BM; rr += 64) {
        const int row = rr + sr;
        const int64_t off = (int64_t)(m0 + row) * ldc + n0 + 64 * p + c8;
        bf16x8 v8 = *reinterpret_cast<const bf16x8*>(img + qoff(row, c8));
        if (EPI == EPI_DGELU) {
          const bf16x8 pv = *reinterpret_cast<const bf16x8*>(Cpre + off);
#pragma unroll
          for (int e = 0; e < 8; ++e)
            v8[e] = f2bf(bf2f(v8[e]) * gelu_grad_f(bf2f(pv[e])));
        } else if (EPI >= 2) {
          *reinterpret_cast<bf16x8*>(Cpre + off) = v8;
#pragma unroll
          for (int e = 0; e < 8; ++e) v8[e] = f2bf(gelu_f(bf2f(v8[e])));
        }
        *reinterpret_cast<bf16x8*>(C + off) = v8;
      }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();  // image free for the next phase
    }
    return;
  }
  // F32OUT (split-K partials): same LDS bounce with a [256][64] f32 image
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  {
    float* imgf = reinterpret_cast<float*>(smem);
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const int cl = wn * 16 + (lane & 15);
#pragma unroll
      for (int mi = 0; mi < 8; ++mi)
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          const int row = wm * 128 + mi * 16 + (lane >> 4) * 4 + e;
          imgf[qoff_f32(row, cl)] = acc[mi][p][e];
        }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
      const int sr = threadIdx.x >> 4;         // 32 rows per round
      const int c4 = (threadIdx.x & 15) * 4;   // 16B chunk (4 floats)
#pragma unroll
      for (int rr = 0; rr < BM; rr += 32) {
        const int row = rr + sr;
        const int64_t off = (int64_t)(m0 + row) * ldc + n0 + 64 * p + c4;
        *reinterpret_cast<f32x4*>(Cf + off) =
            *reinterpret_cast<const f32x4*>(imgf + qoff_f32(row, c4));
      }
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
  }
}

}  // namespace

bool gemm256_supported(int M, int N, int K, int lda, int ldb, bool a_kc,
                       bool b_kc, int epi, int split_k) {
  if (!(a_kc && b_kc && M % BM == 0 && N % BN == 0 && K % BK == 0 &&
        (lda & 7) == 0 && (ldb & 7) == 0))
    return false;
  if (split_k <= 1) return K >= 2 * BK;
  const int chunk = (K / split_k + BK - 1) / BK * BK;
  // every chunk must be >= 2 K-tiles (the pipeline's minimum) and the
  // chunk count must be exactly split_k (rounding can otherwise leave
  // trailing chunks with no K range, whose prologue would read past K)
  return epi == 0 && chunk >= 2 * BK && K % chunk != BK &&
         (K + chunk - 1) / chunk == split_k;
}

void gemm256_bf16(const void* A, const void* B, void* C, void* c_pre,
                  const void* bias, int M, int N, int K, int lda, int ldb,
                  int ldc, int64_t stride_a, int64_t stride_b,
                  int64_t stride_c, int batch, int epi, int split_k,
                  hipStream_t stream) {
  dim3 block(NTH);
  const bf16_t* a = static_cast<const bf16_t*>(A);
  const bf16_t* b = static_cast<const bf16_t*>(B);
  bf16_t* cp = static_cast<bf16_t*>(c_pre);
  const bf16_t* bi = static_cast<const bf16_t*>(bias);
  if (split_k > 1) {
    const int chunk = (K / split_k + BK - 1) / BK * BK;
    dim3 gridk((N / BN) * (M / BM), 1, split_k);
    hipLaunchKernelGGL((gemm256_kernel<0, true>), gridk, block, 0, stream,
                       a, b, C, cp, bi, M, N, K, lda, ldb, ldc, stride_a,
                       stride_b, stride_c, chunk);
    return;
  }
  dim3 grid((N / BN) * (M / BM), 1, batch);
#define G256(E)                                                             \
  hipLaunchKernelGGL((gemm256_kernel<E>), grid, block, 0, stream, a, b, C, \
                     cp, bi, M, N, K, lda, ldb, ldc, stride_a, stride_b,   \
                     stride_c, 0)
  switch (epi) {
    case 0: G256(0); break;
    case 1: G256(1); break;
    case 2: G256(2); break;
    case 3: G256(3); break;
    case 4: G256(4); break;
    default: throw std::runtime_error("bad epi");
  }
#undef G256
}

}  // namespace tepdist
