// Diagnostic probes (not part of the training path).
// tr16_probe: determines the exact lane->element mapping of
// ds_read_b64_tr_b16 on this chip. LDS is filled with bf16(element_index);
// each lane reads at addr = base + (lane&15)*2B + (lane>>4)*128B and we dump
// the 4 returned values per lane, plus a uniform-address variant.

#include "common.h"
#include "kernels.h"

namespace tepdist {

namespace {

typedef __attribute__((address_space(3))) bf16x4* lds_tr_ptr;

__global__ void tr16_probe_kernel(float* out_pattern, float* out_uniform) {
  __shared__ bf16_t lds[512];
  for (int i = threadIdx.x; i < 512; i += blockDim.x)
    lds[i] = f2bf((float)i);
  __syncthreads();
  if (threadIdx.x < 64) {
    const int lane = threadIdx.x;
    const bf16_t* p =
        &lds[0] + (lane & 15) + (lane >> 4) * 64;  // elements (2B each)
    bf16x4 v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16((lds_tr_ptr)p);
#pragma unroll
    for (int j = 0; j < 4; ++j) out_pattern[lane * 4 + j] = bf2f(v[j]);
    bf16x4 u = __builtin_amdgcn_ds_read_tr16_b64_v4bf16((lds_tr_ptr)&lds[0]);
#pragma unroll
    for (int j = 0; j < 4; ++j) out_uniform[lane * 4 + j] = bf2f(u[j]);
  }
}

}  // namespace

void tr16_probe(float* out_pattern, float* out_uniform, hipStream_t stream) {
  hipLaunchKernelGGL(tr16_probe_kernel, dim3(1), dim3(64), 0, stream,
                     out_pattern, out_uniform);
  HIP_CHECK(hipGetLastError());
}

}  // namespace tepdist
