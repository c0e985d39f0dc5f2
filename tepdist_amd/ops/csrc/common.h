// Common helpers for the TePDist-AMD CDNA4 (gfx950 / MI355X) kernel library.
// Hand-written HIP; no CUDA shims, no hipify output.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>
#include <stdexcept>
#include <string>

#define DEV_INLINE __device__ __forceinline__

// CDNA wavefront is 64 lanes (not 32).
constexpr int WAVE = 64;

typedef __bf16 bf16_t;
typedef __bf16 bf16x2 __attribute__((ext_vector_type(2)));
typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef short s16x8 __attribute__((ext_vector_type(8)));
typedef unsigned int u32x4 __attribute__((ext_vector_type(4)));

DEV_INLINE float bf2f(bf16_t v) { return static_cast<float>(v); }
DEV_INLINE bf16_t f2bf(float v) { return static_cast<bf16_t>(v); }

// ---------------------------------------------------------------------------
// Wave / block reductions
// ---------------------------------------------------------------------------

DEV_INLINE float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;  // valid in lane 0
}

DEV_INLINE float wave_allreduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

DEV_INLINE float wave_allreduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// Block-level all-reduce over `nwaves` waves through LDS (caller provides
// a float scratch of >= nwaves entries).
template <typename Op>
DEV_INLINE float block_allreduce(float v, float* scratch, int nwaves, Op op,
                                 float init) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = op(v, __shfl_xor(v, off, 64));
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float r = init;
  for (int i = 0; i < nwaves; ++i) r = op(r, scratch[i]);
  return r;
}

// ---------------------------------------------------------------------------
// Philox 4x32-10 counter-based RNG (for dropout: mask reproducible from
// (seed, offset, linear index) without storing it)
// ---------------------------------------------------------------------------

DEV_INLINE uint2 philox_mulhilo(uint32_t a, uint32_t b) {
  uint64_t p = static_cast<uint64_t>(a) * b;
  return make_uint2(static_cast<uint32_t>(p >> 32), static_cast<uint32_t>(p));
}

struct Philox4 {
  uint2 key;
  uint4 ctr;
  DEV_INLINE Philox4(uint64_t seed, uint64_t subseq, uint64_t offset) {
    key = make_uint2(static_cast<uint32_t>(seed),
                     static_cast<uint32_t>(seed >> 32));
    ctr = make_uint4(static_cast<uint32_t>(offset),
                     static_cast<uint32_t>(offset >> 32),
                     static_cast<uint32_t>(subseq),
                     static_cast<uint32_t>(subseq >> 32));
  }
  DEV_INLINE uint4 operator()() {
    uint4 c = ctr;
    uint2 k = key;
#pragma unroll
    for (int i = 0; i < 10; ++i) {
      uint2 r0 = philox_mulhilo(0xD2511F53u, c.x);
      uint2 r1 = philox_mulhilo(0xCD9E8D57u, c.z);
      c = make_uint4(r1.x ^ c.y ^ k.x, r1.y, r0.x ^ c.w ^ k.y, r0.y);
      k.x += 0x9E3779B9u;
      k.y += 0xBB67AE85u;
    }
    return c;
  }
};

DEV_INLINE float u32_to_uniform(uint32_t x) {
  // (0, 1]
  return (x >> 8) * (1.0f / 16777216.0f) + (1.0f / 33554432.0f);
}

// ---------------------------------------------------------------------------
// GELU (tanh approximation, matches ops/reference.py)
// ---------------------------------------------------------------------------

constexpr float kSqrt2OverPi = 0.7978845608028654f;
constexpr float kGeluC = 0.044715f;

// tanh via the hardware exp2 (v_exp_f32): tanh(z) = 1 - 2/(e^{2z}+1).
// ocml tanhf is a branching libcall (~40 instr) that made the gelu
// passes VALU-bound at ~55% of HBM bandwidth. Saturation behaves: e ->
// inf gives rcp 0 -> 1; e -> 0 gives -1. Accuracy ~1e-6 relative (rcp is
// ~22-bit) — far below bf16 output quantization.
DEV_INLINE float fast_tanh_f(float z) {
  const float e = __builtin_amdgcn_exp2f(z * 2.8853900817779268f);  // 2z*log2e
  return 1.0f - 2.0f * __builtin_amdgcn_rcpf(e + 1.0f);
}

DEV_INLINE float gelu_f(float x) {
  float t = fast_tanh_f(kSqrt2OverPi * (x + kGeluC * x * x * x));
  return 0.5f * x * (1.0f + t);
}

DEV_INLINE float gelu_grad_f(float x) {
  float t = fast_tanh_f(kSqrt2OverPi * (x + kGeluC * x * x * x));
  float dt = (1.0f - t * t) * kSqrt2OverPi * (1.0f + 3.0f * kGeluC * x * x);
  return 0.5f * (1.0f + t) + 0.5f * x * dt;
}

// ---------------------------------------------------------------------------

#define HIP_CHECK(expr)                                              \
  do {                                                               \
    hipError_t _e = (expr);                                          \
    if (_e != hipSuccess) {                                          \
      throw std::runtime_error(std::string("HIP error: ") +          \
                               hipGetErrorString(_e) + " at " +      \
                               __FILE__ + ":" + std::to_string(__LINE__)); \
    }                                                                \
  } while (0)

DEV_INLINE int cdiv_d(int a, int b) { return (a + b - 1) / b; }
inline long cdiv(long a, long b) { return (a + b - 1) / b; }
