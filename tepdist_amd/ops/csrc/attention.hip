// Fused flash-style causal attention (forward + backward) for gfx950.
//
// Forward: one block = 128 q-rows of one (batch, head); 4 waves x 32 rows.
// Per KV tile of 64: K staged k-contiguous in LDS (ds_read_b128 B-fragments
// like the GEMM), V staged transposed via the v_perm register transpose,
// QK^T on mfma_f32_16x16x32_bf16 with Q fragments held in registers across
// the whole row, online softmax (running max/sum) in fp32 registers with
// 16-lane shfl_xor row reductions, P routed through a wave-private LDS tile
// to become the PV A-operand. Saves per-row logsumexp for backward; the
// S x S score matrix is never materialized in HBM.
//
// Backward (flash2-style): one block = 64 kv-rows of one (b,h); recomputes
// P^T = exp(K Q^T * scale - lse) per q-tile (both operands natural-layout
// MFMAs), accumulates dV += P^T dO and dK += dS^T Q in registers,
// dS^T = P^T * (dP^T - delta_q) * scale with dP^T = V dO^T, and scatters
// dQ partials with fp32 atomics (delta = rowsum(dO*O) precomputed).
//
// Replaces the composed scores-GEMM + softmax + PV path (SURVEY.md §2.9:
// the reference leaves attention to XLA codegen; this is the MI355X-native
// fused form).

#include <algorithm>
#include <stdexcept>

#include "common.h"
#include "kernels.h"

namespace tepdist {

namespace {

constexpr int NT = 256;
constexpr int QB = 128;   // q rows per block (fwd)
constexpr int KB = 64;    // kv rows per tile

// LDS addressing for a [rows][C] bf16 image, C in {64,128}: XOR swizzle of
// the 16B column slot by bit-reversed (row>>1) (see gemm.hip).
template <int C>
DEV_INLINE int loff(int row, int col_e) {
  const int r1 = row >> 1;
  const int x = ((r1 & 1) << 2) | (r1 & 2) | ((r1 >> 2) & 1);
  return row * C + (col_e ^ ((x << 3) & (C - 1)));
}

DEV_INLINE float warp16_max(float v) {
#pragma unroll
  for (int d = 1; d < 16; d <<= 1) v = fmaxf(v, __shfl_xor(v, d, 64));
  return v;
}

DEV_INLINE float warp16_sum(float v) {
#pragma unroll
  for (int d = 1; d < 16; d <<= 1) v += __shfl_xor(v, d, 64);
  return v;
}

// ---------------------------------------------------------------------------
// forward
// ---------------------------------------------------------------------------

template <int D>
__launch_bounds__(NT) __global__
void flash_fwd_kernel(const bf16_t* __restrict__ Q,
                      const bf16_t* __restrict__ K,
                      const bf16_t* __restrict__ V, bf16_t* __restrict__ O,
                      float* __restrict__ LSE, int S, int H, float scale,
                      bool causal, int64_t q_bs, int64_t q_hs, int64_t q_rs,
                      int64_t o_bs, int64_t o_hs, int64_t o_rs) {
  constexpr int DK = D / 32;   // k-chunks per fragment row
  constexpr int DF = D / 16;   // output column fragments
  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int q0 = blockIdx.x * QB;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wq0 = q0 + wave * 32;

  // q/k/v share the (bs, hs, rs) layout (a packed qkv differs only in the
  // base pointer); D stays contiguous
  const bf16_t* qp = Q + (int64_t)b * q_bs + (int64_t)h * q_hs;
  const bf16_t* kp = K + (int64_t)b * q_bs + (int64_t)h * q_hs;
  const bf16_t* vp = V + (int64_t)b * q_bs + (int64_t)h * q_hs;
  bf16_t* op = O + (int64_t)b * o_bs + (int64_t)h * o_hs;

  __shared__ bf16_t smem[KB * D + D * KB + 4 * 32 * KB];
  bf16_t* sK = smem;                    // [KB][D] k(=D)-contiguous
  bf16_t* sVT = smem + KB * D;          // [D][KB] k(=kv)-contiguous
  bf16_t* sP = smem + KB * D + D * KB + wave * 32 * KB;  // [32][KB]

  // Q fragments for this wave's 32 rows, kept in registers for the whole row
  bf16x8 qf[2][DK];
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int kk = 0; kk < DK; ++kk) {
      const int row = wq0 + mi * 16 + (lane & 15);
      bf16x8 v8 = {};
      if (row < S)
        v8 = *reinterpret_cast<const bf16x8*>(
            qp + (int64_t)row * q_rs + kk * 32 + 8 * (lane >> 4));
      qf[mi][kk] = v8;
    }

  f32x4 acc_o[2][DF] = {};
  float m_r[2][4], l_r[2][4];
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      m_r[mi][e] = -3.0e38f;
      l_r[mi][e] = 0.f;
    }

  // staging registers (issue-early / write-late split, guide T14):
  // K: 2 x 16B per thread; V: 8 x 4B per thread (v_perm transpose slabs)
  constexpr int KUN = KB * D / 8 / NT;
  bf16x8 krg[KUN];
  uint32_t vrg[8];

  auto stage_load = [&](int kv0) {
#pragma unroll
    for (int u = 0; u < KUN; ++u) {
      const int idx = threadIdx.x + u * NT;
      const int row = idx / (D / 8);
      const int c = (idx % (D / 8)) * 8;
      bf16x8 v8 = {};
      if (kv0 + row < S)
        v8 = *reinterpret_cast<const bf16x8*>(
            kp + (int64_t)(kv0 + row) * q_rs + c);
      krg[u] = v8;
    }
    const int f = 2 * (threadIdx.x % (D / 2));
    const int kb = threadIdx.x / (D / 2);
    if (kb < KB / 8) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int kv = kv0 + kb * 8 + j;
        bf16x2 v2 = {};
        if (kv < S)
          v2 = *reinterpret_cast<const bf16x2*>(vp + (int64_t)kv * q_rs + f);
        vrg[j] = __builtin_bit_cast(uint32_t, v2);
      }
    }
  };
  auto stage_write = [&]() {
#pragma unroll
    for (int u = 0; u < KUN; ++u) {
      const int idx = threadIdx.x + u * NT;
      const int row = idx / (D / 8);
      const int c = (idx % (D / 8)) * 8;
      *reinterpret_cast<bf16x8*>(sK + loff<D>(row, c)) = krg[u];
    }
    const int f = 2 * (threadIdx.x % (D / 2));
    const int kb = threadIdx.x / (D / 2);
    if (kb < KB / 8) {
      uint32_t o0[4], o1[4];
#pragma unroll
      for (int d2 = 0; d2 < 4; ++d2) {
        o0[d2] = __builtin_amdgcn_perm(vrg[2 * d2 + 1], vrg[2 * d2],
                                       0x05040100u);
        o1[d2] = __builtin_amdgcn_perm(vrg[2 * d2 + 1], vrg[2 * d2],
                                       0x07060302u);
      }
      *reinterpret_cast<uint4*>(sVT + loff<KB>(f, kb * 8)) =
          make_uint4(o0[0], o0[1], o0[2], o0[3]);
      *reinterpret_cast<uint4*>(sVT + loff<KB>(f + 1, kb * 8)) =
          make_uint4(o1[0], o1[1], o1[2], o1[3]);
    }
  };

  const int kv_end = causal ? min(S, q0 + QB) : S;
  stage_load(0);
  stage_write();
  __syncthreads();
  for (int kv0 = 0; kv0 < kv_end; kv0 += KB) {
    if (kv0 + KB < kv_end) stage_load(kv0 + KB);  // overlap with compute

    if (!causal || kv0 <= wq0 + 31) {  // wave has unmasked work
      // --- S = Q K^T fragments ---
      f32x4 sf[2][4] = {};
#pragma unroll
      for (int kk = 0; kk < DK; ++kk)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf) {
          const bf16x8 kf = *reinterpret_cast<const bf16x8*>(
              sK + loff<D>((lane & 15) + 16 * nf,
                           8 * (lane >> 4) + 32 * kk));
#pragma unroll
          for (int mi = 0; mi < 2; ++mi)
            sf[mi][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                qf[mi][kk], kf, sf[mi][nf], 0, 0, 0);
        }
      // --- scale + mask; online softmax ---
#pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          const int qg = wq0 + mi * 16 + 4 * (lane >> 4) + e;
          float tmax = -3.0e38f;
#pragma unroll
          for (int nf = 0; nf < 4; ++nf) {
            const int kg = kv0 + nf * 16 + (lane & 15);
            float sv = sf[mi][nf][e] * scale;
            if ((causal && kg > qg) || kg >= S || qg >= S) sv = -3.0e38f;
            sf[mi][nf][e] = sv;
            tmax = fmaxf(tmax, sv);
          }
          tmax = warp16_max(tmax);
          const float mn = fmaxf(m_r[mi][e], tmax);
          const float alpha = (mn <= -1.0e38f) ? 1.f
                                               : __expf(m_r[mi][e] - mn);
          m_r[mi][e] = mn;
          float rs = 0.f;
#pragma unroll
          for (int nf = 0; nf < 4; ++nf) {
            const float sv = sf[mi][nf][e];
            const float p = (sv <= -1.0e38f) ? 0.f : __expf(sv - mn);
            sf[mi][nf][e] = p;
            rs += p;
          }
          rs = warp16_sum(rs);
          l_r[mi][e] = l_r[mi][e] * alpha + rs;
#pragma unroll
          for (int df = 0; df < DF; ++df) acc_o[mi][df][e] *= alpha;
          // write P (bf16) to the wave-private LDS tile in C layout
          const int prow = mi * 16 + 4 * (lane >> 4) + e;
#pragma unroll
          for (int nf = 0; nf < 4; ++nf)
            sP[loff<KB>(prow, nf * 16 + (lane & 15))] =
                f2bf(sf[mi][nf][e]);
        }
      }
      // ensure the wave's sP writes are visible to its own reads
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      // --- O += P V ---
#pragma unroll
      for (int kk2 = 0; kk2 < KB / 32; ++kk2) {
#pragma unroll
        for (int mi = 0; mi < 2; ++mi) {
          const bf16x8 pf = *reinterpret_cast<const bf16x8*>(
              sP + loff<KB>(mi * 16 + (lane & 15),
                            8 * (lane >> 4) + 32 * kk2));
#pragma unroll
          for (int df = 0; df < DF; ++df) {
            const bf16x8 vf = *reinterpret_cast<const bf16x8*>(
                sVT + loff<KB>((lane & 15) + 16 * df,
                               8 * (lane >> 4) + 32 * kk2));
            acc_o[mi][df] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                pf, vf, acc_o[mi][df], 0, 0, 0);
          }
        }
      }
    }
    __syncthreads();  // all waves done reading sK/sVT
    if (kv0 + KB < kv_end) {
      stage_write();  // overwrite with the pre-loaded next tile
      __syncthreads();
    }
  }

  // --- epilogue: O /= l, write O and lse ---
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      const int qg = wq0 + mi * 16 + 4 * (lane >> 4) + e;
      if (qg >= S) continue;
      const float lv = l_r[mi][e];
      const float inv = (lv > 0.f) ? 1.0f / lv : 0.f;
      if ((lane & 15) == 0)
        LSE[(int64_t)bh * S + qg] =
            (lv > 0.f) ? m_r[mi][e] + __logf(lv) : -3.0e38f;
#pragma unroll
      for (int df = 0; df < DF; ++df)
        op[(int64_t)qg * o_rs + df * 16 + (lane & 15)] =
            f2bf(acc_o[mi][df][e] * inv);
    }
}

}  // namespace

void attention_fwd_bf16(const void* q, const void* k, const void* v, void* o,
                        float* lse, int B, int H, int S, int D, float scale,
                        bool causal, int64_t q_bs, int64_t q_hs, int64_t q_rs,
                        int64_t o_bs, int64_t o_hs, int64_t o_rs,
                        hipStream_t stream) {
  dim3 grid((S + QB - 1) / QB, B * H);
  dim3 block(NT);
#define FWD_D(DD)                                                       \
  hipLaunchKernelGGL(flash_fwd_kernel<DD>, grid, block, 0, stream,      \
                     static_cast<const bf16_t*>(q),                     \
                     static_cast<const bf16_t*>(k),                     \
                     static_cast<const bf16_t*>(v),                     \
                     static_cast<bf16_t*>(o), lse, S, H, scale, causal, \
                     q_bs, q_hs, q_rs, o_bs, o_hs, o_rs)
  if (D == 64) FWD_D(64);
  else if (D == 128) FWD_D(128);
  else throw std::runtime_error("flash fwd: head dim must be 64 or 128");
#undef FWD_D
}

namespace {

// ---------------------------------------------------------------------------
// backward
// ---------------------------------------------------------------------------

// delta[bh][s] = rowsum(dO * O) (flash2 preprocess)
template <int D>
__launch_bounds__(NT) __global__
void flash_bwd_delta_kernel(const bf16_t* __restrict__ dO,
                            const bf16_t* __restrict__ O,
                            float* __restrict__ delta, int64_t rows) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int64_t nw = (int64_t)gridDim.x * (NT / WAVE);
  for (int64_t r = (int64_t)blockIdx.x * (NT / WAVE) + wid; r < rows;
       r += nw) {
    float s = 0.f;
    for (int c = lane * 8; c < D; c += WAVE * 8) {
      const bf16x8 a = *reinterpret_cast<const bf16x8*>(dO + r * D + c);
      const bf16x8 b = *reinterpret_cast<const bf16x8*>(O + r * D + c);
#pragma unroll
      for (int e = 0; e < 8; ++e) s += bf2f(a[e]) * bf2f(b[e]);
    }
    s = wave_allreduce_sum(s);
    if (lane == 0) delta[r] = s;
  }
}

// cooperative stage of a [T][D] tile as BOTH natural [T][D] and transposed
// [D][T] images (v_perm 8x2 slabs for the transpose)
template <int T, int D>
DEV_INLINE void stage_nat_t(const bf16_t* __restrict__ src, int row0, int S,
                            int64_t rs, bf16_t* nat, bf16_t* tr) {
  constexpr int UN = T * D / 8 / NT;
#pragma unroll
  for (int u = 0; u < UN; ++u) {
    const int idx = threadIdx.x + u * NT;
    const int row = idx / (D / 8);
    const int c = (idx % (D / 8)) * 8;
    bf16x8 v8 = {};
    if (row0 + row < S)
      v8 = *reinterpret_cast<const bf16x8*>(
          src + (int64_t)(row0 + row) * rs + c);
    *reinterpret_cast<bf16x8*>(nat + loff<D>(row, c)) = v8;
  }
  constexpr int NSLAB = (T / 8) * (D / 2);
#pragma unroll
  for (int u = 0; u < (NSLAB + NT - 1) / NT; ++u) {
    const int idx = threadIdx.x + u * NT;
    if (idx < NSLAB) {
      const int f = 2 * (idx % (D / 2));
      const int kb = idx / (D / 2);
      uint32_t r[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int row = row0 + kb * 8 + j;
        bf16x2 v2 = {};
        if (row < S)
          v2 = *reinterpret_cast<const bf16x2*>(src + (int64_t)row * rs + f);
        r[j] = __builtin_bit_cast(uint32_t, v2);
      }
      uint32_t o0[4], o1[4];
#pragma unroll
      for (int d2 = 0; d2 < 4; ++d2) {
        o0[d2] = __builtin_amdgcn_perm(r[2 * d2 + 1], r[2 * d2], 0x05040100u);
        o1[d2] = __builtin_amdgcn_perm(r[2 * d2 + 1], r[2 * d2], 0x07060302u);
      }
      *reinterpret_cast<uint4*>(tr + loff<T>(f, kb * 8)) =
          make_uint4(o0[0], o0[1], o0[2], o0[3]);
      *reinterpret_cast<uint4*>(tr + loff<T>(f + 1, kb * 8)) =
          make_uint4(o1[0], o1[1], o1[2], o1[3]);
    }
  }
}

// one block = 64 kv rows of one (b,h); 4 waves x 16 kv rows.
template <int D>
__launch_bounds__(NT) __global__
void flash_bwd_kernel(const bf16_t* __restrict__ Q,
                      const bf16_t* __restrict__ K,
                      const bf16_t* __restrict__ V,
                      const bf16_t* __restrict__ dO,
                      const float* __restrict__ LSE,
                      const float* __restrict__ DELTA,
                      float* __restrict__ dQws, bf16_t* __restrict__ dK,
                      bf16_t* __restrict__ dV, int S, int H, float scale,
                      bool causal, int64_t q_bs, int64_t q_hs, int64_t q_rs,
                      int64_t o_bs, int64_t o_hs, int64_t o_rs) {
  constexpr int DK = D / 32;
  constexpr int DF = D / 16;
  constexpr int QT = 64;
  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int kv0 = blockIdx.x * KB;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wk0 = kv0 + wave * 16;   // this wave's 16 kv rows

  const int64_t qoff = (int64_t)b * q_bs + (int64_t)h * q_hs;
  const int64_t ooff = (int64_t)b * o_bs + (int64_t)h * o_hs;
  const bf16_t* qp = Q + qoff;
  const bf16_t* kp = K + qoff;
  const bf16_t* vp = V + qoff;
  const bf16_t* dop = dO + ooff;

  __shared__ bf16_t smem[KB * D * 3 + D * KB + QT * D * 2 + D * QT +
                         QT * KB + 4 * 16 * QT];
  bf16_t* sKb = smem;                       // [KB][D] natural
  bf16_t* sKT = sKb + KB * D;               // [D][KB]
  bf16_t* sVb = sKT + D * KB;               // [KB][D] natural
  bf16_t* sQ = sVb + KB * D;                // [QT][D] natural
  bf16_t* sQT = sQ + QT * D;                // [D][QT]
  bf16_t* sdO = sQT + D * QT;               // [QT][D] natural
  bf16_t* sdOT = sdO + QT * D;              // [D][QT]
  bf16_t* sdS = sdOT + D * QT;              // [QT][KB] shared
  bf16_t* sPT = sdS + QT * KB + wave * 16 * QT;  // [16][QT] wave-private
  __shared__ float sLSE[QT], sDELTA[QT];

  // stage K, V tiles (fixed for the block)
  stage_nat_t<KB, D>(kp, kv0, S, q_rs, sKb, sKT);
  {
    constexpr int UN = KB * D / 8 / NT;
#pragma unroll
    for (int u = 0; u < UN; ++u) {
      const int idx = threadIdx.x + u * NT;
      const int row = idx / (D / 8);
      const int c = (idx % (D / 8)) * 8;
      bf16x8 v8 = {};
      if (kv0 + row < S)
        v8 = *reinterpret_cast<const bf16x8*>(
            vp + (int64_t)(kv0 + row) * q_rs + c);
      *reinterpret_cast<bf16x8*>(sVb + loff<D>(row, c)) = v8;
    }
  }
  __syncthreads();

  // A fragments of this wave's K and V rows (constant across q tiles)
  bf16x8 kf[DK], vf[DK];
#pragma unroll
  for (int kk = 0; kk < DK; ++kk) {
    kf[kk] = *reinterpret_cast<const bf16x8*>(
        sKb + loff<D>(wave * 16 + (lane & 15), 8 * (lane >> 4) + 32 * kk));
    vf[kk] = *reinterpret_cast<const bf16x8*>(
        sVb + loff<D>(wave * 16 + (lane & 15), 8 * (lane >> 4) + 32 * kk));
  }

  f32x4 acc_dk[DF] = {};
  f32x4 acc_dv[DF] = {};

  // issue-early / write-late staging registers for the Q and dO tiles
  constexpr int QUN = QT * D / 8 / NT;
  bf16x8 qn[QUN], don[QUN];
  uint32_t qt[8], dot_[8];
  float lse2[QT / NT + 1], dl2[QT / NT + 1];

  auto tile_load = [&](int q0) {
#pragma unroll
    for (int u = 0; u < QUN; ++u) {
      const int idx = threadIdx.x + u * NT;
      const int row = idx / (D / 8);
      const int c = (idx % (D / 8)) * 8;
      bf16x8 a = {}, b2 = {};
      if (q0 + row < S) {
        a = *reinterpret_cast<const bf16x8*>(
            qp + (int64_t)(q0 + row) * q_rs + c);
        b2 = *reinterpret_cast<const bf16x8*>(
            dop + (int64_t)(q0 + row) * o_rs + c);
      }
      qn[u] = a;
      don[u] = b2;
    }
    const int f = 2 * (threadIdx.x % (D / 2));
    const int kb = threadIdx.x / (D / 2);
    if (kb < QT / 8) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int row = q0 + kb * 8 + j;
        bf16x2 a = {}, b2 = {};
        if (row < S) {
          a = *reinterpret_cast<const bf16x2*>(
              qp + (int64_t)row * q_rs + f);
          b2 = *reinterpret_cast<const bf16x2*>(
              dop + (int64_t)row * o_rs + f);
        }
        qt[j] = __builtin_bit_cast(uint32_t, a);
        dot_[j] = __builtin_bit_cast(uint32_t, b2);
      }
    }
    for (int i = threadIdx.x, s2 = 0; i < QT; i += NT, ++s2) {
      const int qg = q0 + i;
      lse2[s2] = (qg < S) ? LSE[(int64_t)bh * S + qg] : -3.0e38f;
      dl2[s2] = (qg < S) ? DELTA[(int64_t)bh * S + qg] : 0.f;
    }
  };
  auto tile_write = [&]() {
#pragma unroll
    for (int u = 0; u < QUN; ++u) {
      const int idx = threadIdx.x + u * NT;
      const int row = idx / (D / 8);
      const int c = (idx % (D / 8)) * 8;
      *reinterpret_cast<bf16x8*>(sQ + loff<D>(row, c)) = qn[u];
      *reinterpret_cast<bf16x8*>(sdO + loff<D>(row, c)) = don[u];
    }
    const int f = 2 * (threadIdx.x % (D / 2));
    const int kb = threadIdx.x / (D / 2);
    if (kb < QT / 8) {
      uint32_t o0[4], o1[4];
#pragma unroll
      for (int d2 = 0; d2 < 4; ++d2) {
        o0[d2] = __builtin_amdgcn_perm(qt[2 * d2 + 1], qt[2 * d2],
                                       0x05040100u);
        o1[d2] = __builtin_amdgcn_perm(qt[2 * d2 + 1], qt[2 * d2],
                                       0x07060302u);
      }
      *reinterpret_cast<uint4*>(sQT + loff<QT>(f, kb * 8)) =
          make_uint4(o0[0], o0[1], o0[2], o0[3]);
      *reinterpret_cast<uint4*>(sQT + loff<QT>(f + 1, kb * 8)) =
          make_uint4(o1[0], o1[1], o1[2], o1[3]);
#pragma unroll
      for (int d2 = 0; d2 < 4; ++d2) {
        o0[d2] = __builtin_amdgcn_perm(dot_[2 * d2 + 1], dot_[2 * d2],
                                       0x05040100u);
        o1[d2] = __builtin_amdgcn_perm(dot_[2 * d2 + 1], dot_[2 * d2],
                                       0x07060302u);
      }
      *reinterpret_cast<uint4*>(sdOT + loff<QT>(f, kb * 8)) =
          make_uint4(o0[0], o0[1], o0[2], o0[3]);
      *reinterpret_cast<uint4*>(sdOT + loff<QT>(f + 1, kb * 8)) =
          make_uint4(o1[0], o1[1], o1[2], o1[3]);
    }
    for (int i = threadIdx.x, s2 = 0; i < QT; i += NT, ++s2) {
      sLSE[i] = lse2[s2];
      sDELTA[i] = dl2[s2];
    }
  };

  const int q_start = causal ? (kv0 / QT) * QT : 0;
  tile_load(q_start);
  tile_write();
  __syncthreads();
  for (int q0 = q_start; q0 < S; q0 += QT) {
    if (q0 + QT < S) tile_load(q0 + QT);  // overlap with compute

    // S^T = K Q^T ; dP^T = V dO^T   (both natural-layout B reads)
    f32x4 st[4] = {};
    f32x4 dpt[4] = {};
#pragma unroll
    for (int kk = 0; kk < DK; ++kk)
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        const bf16x8 qb = *reinterpret_cast<const bf16x8*>(
            sQ + loff<D>((lane & 15) + 16 * nf, 8 * (lane >> 4) + 32 * kk));
        st[nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kf[kk], qb, st[nf],
                                                         0, 0, 0);
        const bf16x8 db = *reinterpret_cast<const bf16x8*>(
            sdO + loff<D>((lane & 15) + 16 * nf, 8 * (lane >> 4) + 32 * kk));
        dpt[nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(vf[kk], db,
                                                          dpt[nf], 0, 0, 0);
      }

    // P^T and dS^T (elementwise, C layout: kv row = 4*(lane>>4)+e within
    // this wave's 16; q col = q0 + nf*16 + (lane&15))
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      const int qcol = nf * 16 + (lane & 15);
      const int qg = q0 + qcol;
      const float lse = sLSE[qcol];
      const float dl = sDELTA[qcol];
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        const int kvg = wk0 + 4 * (lane >> 4) + e;
        float pt = 0.f;
        if (qg < S && kvg < S && (!causal || kvg <= qg) && lse > -1.0e38f)
          pt = __expf(st[nf][e] * scale - lse);
        st[nf][e] = pt;                                  // now P^T
        dpt[nf][e] = pt * (dpt[nf][e] - dl) * scale;     // now dS^T
      }
    }

    // write P^T to the wave tile, dV += P^T dO (via sdOT)
#pragma unroll
    for (int nf = 0; nf < 4; ++nf)
#pragma unroll
      for (int e = 0; e < 4; ++e)
        sPT[loff<QT>(4 * (lane >> 4) + e, nf * 16 + (lane & 15))] =
            f2bf(st[nf][e]);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
    for (int kk2 = 0; kk2 < QT / 32; ++kk2) {
      const bf16x8 pa = *reinterpret_cast<const bf16x8*>(
          sPT + loff<QT>(lane & 15, 8 * (lane >> 4) + 32 * kk2));
#pragma unroll
      for (int df = 0; df < DF; ++df) {
        const bf16x8 dob = *reinterpret_cast<const bf16x8*>(
            sdOT + loff<QT>((lane & 15) + 16 * df,
                            8 * (lane >> 4) + 32 * kk2));
        acc_dv[df] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, dob,
                                                             acc_dv[df],
                                                             0, 0, 0);
      }
    }

    // write dS^T to the wave tile (reuse) and to the shared [q][kv] image
#pragma unroll
    for (int nf = 0; nf < 4; ++nf)
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        const bf16_t dsv = f2bf(dpt[nf][e]);
        sPT[loff<QT>(4 * (lane >> 4) + e, nf * 16 + (lane & 15))] = dsv;
        sdS[loff<KB>(nf * 16 + (lane & 15),
                     wave * 16 + 4 * (lane >> 4) + e)] = dsv;
      }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    // dK += dS^T Q (via sQT)
#pragma unroll
    for (int kk2 = 0; kk2 < QT / 32; ++kk2) {
      const bf16x8 da = *reinterpret_cast<const bf16x8*>(
          sPT + loff<QT>(lane & 15, 8 * (lane >> 4) + 32 * kk2));
#pragma unroll
      for (int df = 0; df < DF; ++df) {
        const bf16x8 qb = *reinterpret_cast<const bf16x8*>(
            sQT + loff<QT>((lane & 15) + 16 * df,
                           8 * (lane >> 4) + 32 * kk2));
        acc_dk[df] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(da, qb,
                                                             acc_dk[df],
                                                             0, 0, 0);
      }
    }
    __syncthreads();  // sdS complete; sQ/sQT/sdO/sdOT reads done
    if (q0 + QT < S) tile_write();  // overlaps the dQ phase (sdS/sKT only)

    // dQ partial: wave w owns q rows [q0+16w, +16): dQ = dS @ K (via sKT)
    {
      f32x4 acc_dq[DF] = {};
#pragma unroll
      for (int kk2 = 0; kk2 < KB / 32; ++kk2) {
        const bf16x8 dsa = *reinterpret_cast<const bf16x8*>(
            sdS + loff<KB>(wave * 16 + (lane & 15),
                           8 * (lane >> 4) + 32 * kk2));
#pragma unroll
        for (int df = 0; df < DF; ++df) {
          const bf16x8 kb = *reinterpret_cast<const bf16x8*>(
              sKT + loff<KB>((lane & 15) + 16 * df,
                             8 * (lane >> 4) + 32 * kk2));
          acc_dq[df] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsa, kb,
                                                               acc_dq[df],
                                                               0, 0, 0);
        }
      }
#pragma unroll
      for (int df = 0; df < DF; ++df)
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          const int qg = q0 + wave * 16 + 4 * (lane >> 4) + e;
          if (qg < S)
            atomicAdd(&dQws[(int64_t)bh * S * D + (int64_t)qg * D +
                            df * 16 + (lane & 15)],
                      acc_dq[df][e]);
        }
    }
    __syncthreads();  // next tile's images complete before its reads
  }

  // write dK, dV (C layout scatter)
#pragma unroll
  for (int df = 0; df < DF; ++df)
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      const int kvg = wk0 + 4 * (lane >> 4) + e;
      if (kvg >= S) continue;
      const int64_t off = qoff + (int64_t)kvg * q_rs + df * 16 + (lane & 15);
      dK[off] = f2bf(acc_dk[df][e]);
      dV[off] = f2bf(acc_dv[df][e]);
    }
}



// delta kernel needs contiguous dO/O rows: the host passes per-(b,h)
// strided views by iterating bh in the kernel via strides
template <int D>
__global__ void flash_bwd_delta_strided_kernel(
    const bf16_t* __restrict__ dO, const bf16_t* __restrict__ O,
    float* __restrict__ delta, int H, int S, int64_t o_bs, int64_t o_hs,
    int64_t o_rs) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int64_t rows = (int64_t)gridDim.y * S;  // gridDim.y == B*H
  const int64_t nw = (int64_t)gridDim.x * (NT / WAVE);
  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int64_t off = (int64_t)b * o_bs + (int64_t)h * o_hs;
  for (int64_t r = (int64_t)blockIdx.x * (NT / WAVE) + wid; r < S; r += nw) {
    float s = 0.f;
    for (int c = lane * 8; c < D; c += WAVE * 8) {
      const bf16x8 a =
          *reinterpret_cast<const bf16x8*>(dO + off + r * o_rs + c);
      const bf16x8 bb =
          *reinterpret_cast<const bf16x8*>(O + off + r * o_rs + c);
#pragma unroll
      for (int e = 0; e < 8; ++e) s += bf2f(a[e]) * bf2f(bb[e]);
    }
    s = wave_allreduce_sum(s);
    if (lane == 0) delta[(int64_t)bh * S + r] = s;
  }
}

}  // namespace


namespace {
__global__ void cast_scatter_kernel(const float* __restrict__ src,
                                    bf16_t* __restrict__ dst, int H, int S,
                                    int D, int64_t bs, int64_t hs,
                                    int64_t rs) {
  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int64_t base = (int64_t)b * bs + (int64_t)h * hs;
  const int64_t n = (int64_t)S * D;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * 4;
  for (int64_t i = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 4;
       i < n; i += stride) {
    const int64_t row = i / D;
    const int c = (int)(i - row * D);
    const f32x4 v = *reinterpret_cast<const f32x4*>(
        src + (int64_t)bh * n + i);
    bf16x4 o;
#pragma unroll
    for (int e = 0; e < 4; ++e) o[e] = f2bf(v[e]);
    *reinterpret_cast<bf16x4*>(dst + base + row * rs + c) = o;
  }
}
}  // namespace

void cast_scatter_bf16(const float* src, void* dst, int B, int H, int S,
                       int D, int64_t bs, int64_t hs, int64_t rs,
                       hipStream_t stream) {
  const int blocks = (int)std::min<int64_t>(
      ((int64_t)S * D / 4 + NT - 1) / NT, 512);
  hipLaunchKernelGGL(cast_scatter_kernel, dim3(std::max(blocks, 1), B * H),
                     dim3(NT), 0, stream, src, static_cast<bf16_t*>(dst), H,
                     S, D, bs, hs, rs);
}

void attention_bwd_bf16(const void* q, const void* k, const void* v,
                        const void* o, const void* dout, const float* lse,
                        float* delta, float* dq_ws, void* dk, void* dv,
                        int B, int H, int S, int D, float scale, bool causal,
                        int64_t q_bs, int64_t q_hs, int64_t q_rs,
                        int64_t o_bs, int64_t o_hs, int64_t o_rs,
                        hipStream_t stream) {
  dim3 block(NT);
#define BWD_D(DD)                                                            \
  do {                                                                       \
    dim3 dgrid((S / 4 + (NT / WAVE) - 1) / (NT / WAVE) > 256                 \
                   ? 256 : (S + 3) / 4, B * H);                              \
    hipLaunchKernelGGL(flash_bwd_delta_strided_kernel<DD>, dgrid, block, 0,  \
                       stream, static_cast<const bf16_t*>(dout),             \
                       static_cast<const bf16_t*>(o), delta, H, S, o_bs,     \
                       o_hs, o_rs);                                          \
    dim3 grid((S + KB - 1) / KB, B * H);                                     \
    hipLaunchKernelGGL(flash_bwd_kernel<DD>, grid, block, 0, stream,         \
                       static_cast<const bf16_t*>(q),                        \
                       static_cast<const bf16_t*>(k),                        \
                       static_cast<const bf16_t*>(v),                        \
                       static_cast<const bf16_t*>(dout), lse, delta, dq_ws,  \
                       static_cast<bf16_t*>(dk), static_cast<bf16_t*>(dv),   \
                       S, H, scale, causal, q_bs, q_hs, q_rs, o_bs, o_hs,    \
                       o_rs);                                                \
  } while (0)
  if (D == 64) BWD_D(64);
  else if (D == 128) BWD_D(128);
  else throw std::runtime_error("flash bwd: head dim must be 64 or 128");
#undef BWD_D
}
}  // namespace tepdist
