// Direct hipBLASLt driver for fused-epilogue GEMMs (host-only TU).
//
// torch's F.linear path cannot express hipBLASLt's fused epilogues; this
// binding drives the library directly for the two MLP fusions that remove
// whole HBM round-trips on the bf16 training step:
//   mode 0  GELU_AUX_BIAS : y = gelu(x @ w^T + b), aux := pre-gelu
//           (the separate gelu_fwd kernel's read of the 537 MB pre-act
//            tensor disappears)
//   mode 1  DGELU_BGRAD   : dh = dgelu(dout @ w2, aux), db := rowsum
//           (kills the separate gelu_bwd kernel AND the fc bias-grad
//            reduction)
// Row-major torch tensors are described to the column-major API as their
// transposes: D(N,M)=op(A)op(B) with A=w(K,N,ld=K,OP_T), B=x(K,M,ld=K).
//
// Plans (desc + layouts + heuristic algo + workspace) are cached per
// (mode, M, N, K); bias/aux pointers are re-set per call. First use of a
// shape happens during eager warmup, so plan creation (and its hipMalloc
// workspace growth) never lands inside a hipGraph capture.

#include <hip/hip_runtime.h>
#include <hipblaslt/hipblaslt.h>

#include <pybind11/pybind11.h>

#include <cstdint>
#include <map>
#include <mutex>
#include <stdexcept>
#include <string>
#include <tuple>

namespace {

#define BLT_CHECK(expr)                                                   \
  do {                                                                    \
    hipblasStatus_t s_ = (expr);                                          \
    if (s_ != HIPBLAS_STATUS_SUCCESS)                                     \
      throw std::runtime_error(std::string("hipblaslt error ") +          \
                               std::to_string(int(s_)) + " at " #expr);   \
  } while (0)

#define HIP_CHECK(expr)                                                   \
  do {                                                                    \
    hipError_t e_ = (expr);                                               \
    if (e_ != hipSuccess)                                                 \
      throw std::runtime_error(std::string("hip error: ") +               \
                               hipGetErrorString(e_));                    \
  } while (0)

constexpr size_t kMaxWorkspace = 64u << 20;

struct Plan {
  hipblasLtMatmulDesc_t desc;
  hipblasLtMatrixLayout_t la, lb, ld;
  hipblasLtMatmulAlgo_t algo;
  size_t ws_size;
};

hipblasLtHandle_t handle() {
  static hipblasLtHandle_t h = [] {
    hipblasLtHandle_t hh;
    BLT_CHECK(hipblasLtCreate(&hh));
    return hh;
  }();
  return h;
}

void* workspace() {
  static void* ws = [] {
    void* p;
    HIP_CHECK(hipMalloc(&p, kMaxWorkspace));
    return p;
  }();
  return ws;
}

std::map<std::tuple<int, long, long, long>, Plan> g_plans;
std::mutex g_mu;

Plan& get_plan(int mode, long M, long N, long K) {
  std::lock_guard<std::mutex> lk(g_mu);
  auto key = std::make_tuple(mode, M, N, K);
  auto it = g_plans.find(key);
  if (it != g_plans.end()) return it->second;

  Plan p{};
  BLT_CHECK(hipblasLtMatmulDescCreate(&p.desc, HIPBLAS_COMPUTE_32F,
                                      HIP_R_32F));
  int32_t opT = HIPBLAS_OP_T, opN = HIPBLAS_OP_N;
  hipblasLtEpilogue_t epi;
  int64_t aux_ld;
  if (mode == 0) {
    // y[M,N] = gelu(x[M,K] @ w[N,K]^T + b[N]); col-major D(N,M)
    BLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.desc, HIPBLASLT_MATMUL_DESC_TRANSA, &opT, sizeof(opT)));
    BLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.desc, HIPBLASLT_MATMUL_DESC_TRANSB, &opN, sizeof(opN)));
    BLT_CHECK(hipblasLtMatrixLayoutCreate(&p.la, HIP_R_16BF, K, N, K));
    BLT_CHECK(hipblasLtMatrixLayoutCreate(&p.lb, HIP_R_16BF, K, M, K));
    BLT_CHECK(hipblasLtMatrixLayoutCreate(&p.ld, HIP_R_16BF, N, M, N));
    epi = HIPBLASLT_EPILOGUE_GELU_AUX_BIAS;
    aux_ld = N;
  } else {
    // dh[M,K] = dgelu(dout[M,N] @ w2[N,K], aux[M,K]) + bgrad db[K];
    // col-major D(K,M) = w2(K,N,OP_N) x dout(N,M,OP_N)
    BLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.desc, HIPBLASLT_MATMUL_DESC_TRANSA, &opN, sizeof(opN)));
    BLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.desc, HIPBLASLT_MATMUL_DESC_TRANSB, &opN, sizeof(opN)));
    BLT_CHECK(hipblasLtMatrixLayoutCreate(&p.la, HIP_R_16BF, K, N, K));
    BLT_CHECK(hipblasLtMatrixLayoutCreate(&p.lb, HIP_R_16BF, N, M, N));
    BLT_CHECK(hipblasLtMatrixLayoutCreate(&p.ld, HIP_R_16BF, K, M, K));
    epi = HIPBLASLT_EPILOGUE_DGELU_BGRAD;
    aux_ld = K;
  }
  BLT_CHECK(hipblasLtMatmulDescSetAttribute(
      p.desc, HIPBLASLT_MATMUL_DESC_EPILOGUE, &epi, sizeof(epi)));
  BLT_CHECK(hipblasLtMatmulDescSetAttribute(
      p.desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_LD, &aux_ld,
      sizeof(aux_ld)));
  int32_t bias_t = HIP_R_16BF;
  BLT_CHECK(hipblasLtMatmulDescSetAttribute(
      p.desc, HIPBLASLT_MATMUL_DESC_BIAS_DATA_TYPE, &bias_t,
      sizeof(bias_t)));
  // pointers must be non-null for the heuristic to accept the epilogue;
  // real ones are set per call
  void* dummy = workspace();
  BLT_CHECK(hipblasLtMatmulDescSetAttribute(
      p.desc, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &dummy, sizeof(dummy)));
  BLT_CHECK(hipblasLtMatmulDescSetAttribute(
      p.desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_POINTER, &dummy,
      sizeof(dummy)));

  hipblasLtMatmulPreference_t pref;
  BLT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
  size_t ws = kMaxWorkspace;
  BLT_CHECK(hipblasLtMatmulPreferenceSetAttribute(
      pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws, sizeof(ws)));
  hipblasLtMatmulHeuristicResult_t res[4];
  int nres = 0;
  BLT_CHECK(hipblasLtMatmulAlgoGetHeuristic(handle(), p.desc, p.la, p.lb,
                                            p.ld, p.ld, pref, 4, res,
                                            &nres));
  hipblasLtMatmulPreferenceDestroy(pref);
  if (nres == 0)
    throw std::runtime_error("hipblaslt: no algo for fused epilogue " +
                             std::to_string(mode));
  p.algo = res[0].algo;
  p.ws_size = res[0].workspaceSize;
  return g_plans.emplace(key, p).first->second;
}

// mode 0: A=w1[N,K], B=x[M,K], D=y[M,N], bias=b1[N], aux out [M,N]
// mode 1: A=w2[N,K], B=dout[M,N], D=dh[M,K], bias=db out [K], aux in [M,K]
void blt_fused(int mode, long M, long N, long K, uintptr_t A, uintptr_t B,
               uintptr_t D, uintptr_t bias, uintptr_t aux,
               uintptr_t stream) {
  Plan& p = get_plan(mode, M, N, K);
  void* bp = reinterpret_cast<void*>(bias);
  void* ap = reinterpret_cast<void*>(aux);
  {
    std::lock_guard<std::mutex> lk(g_mu);
    BLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.desc, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &bp, sizeof(bp)));
    BLT_CHECK(hipblasLtMatmulDescSetAttribute(
        p.desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_POINTER, &ap,
        sizeof(ap)));
  }
  float alpha = 1.0f, beta = 0.0f;
  BLT_CHECK(hipblasLtMatmul(
      handle(), p.desc, &alpha, reinterpret_cast<const void*>(A), p.la,
      reinterpret_cast<const void*>(B), p.lb, &beta,
      reinterpret_cast<const void*>(D), p.ld,
      reinterpret_cast<void*>(D), p.ld, &p.algo, workspace(),
      p.ws_size, reinterpret_cast<hipStream_t>(stream)));
}

// Support probe: which epilogue / bias-dtype / aux-dtype combinations
// does the library's heuristic actually offer algorithms for on this
// device + shape? Returns {label: n_algos}.
pybind11::dict probe_epilogues(long M, long N, long K) {
  pybind11::dict out;
  struct Case {
    const char* label;
    hipblasLtEpilogue_t epi;
    int bias_t;  // hipDataType or -1 to skip attr
    int aux_t;   // hipDataType or -1 to skip attr
    bool need_aux;
  };
  const Case cases[] = {
      {"default", HIPBLASLT_EPILOGUE_DEFAULT, -1, -1, false},
      {"bias_bf16", HIPBLASLT_EPILOGUE_BIAS, HIP_R_16BF, -1, false},
      {"bias_f32", HIPBLASLT_EPILOGUE_BIAS, HIP_R_32F, -1, false},
      {"gelu", HIPBLASLT_EPILOGUE_GELU, -1, -1, false},
      {"gelu_bias_bf16", HIPBLASLT_EPILOGUE_GELU_BIAS, HIP_R_16BF, -1,
       false},
      {"gelu_aux", HIPBLASLT_EPILOGUE_GELU_AUX, -1, -1, true},
      {"gelu_aux_bias_bf16", HIPBLASLT_EPILOGUE_GELU_AUX_BIAS, HIP_R_16BF,
       -1, true},
      {"gelu_aux_bias_f32", HIPBLASLT_EPILOGUE_GELU_AUX_BIAS, HIP_R_32F,
       -1, true},
      {"gelu_aux_bias_bf16_auxf32", HIPBLASLT_EPILOGUE_GELU_AUX_BIAS,
       HIP_R_16BF, HIP_R_32F, true},
      {"dgelu", HIPBLASLT_EPILOGUE_DGELU, -1, -1, true},
      {"dgelu_bgrad_bf16", HIPBLASLT_EPILOGUE_DGELU_BGRAD, HIP_R_16BF, -1,
       true},
      {"dgelu_bgrad_f32", HIPBLASLT_EPILOGUE_DGELU_BGRAD, HIP_R_32F, -1,
       true},
      {"bgradb_bf16", HIPBLASLT_EPILOGUE_BGRADB, HIP_R_16BF, -1, false},
  };
  for (const Case& c : cases) {
    hipblasLtMatmulDesc_t desc;
    BLT_CHECK(hipblasLtMatmulDescCreate(&desc, HIPBLAS_COMPUTE_32F,
                                        HIP_R_32F));
    int32_t opT = HIPBLAS_OP_T, opN = HIPBLAS_OP_N;
    hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_TRANSA,
                                    &opT, sizeof(opT));
    hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_TRANSB,
                                    &opN, sizeof(opN));
    hipblasLtEpilogue_t e = c.epi;
    hipblasLtMatmulDescSetAttribute(desc, HIPBLASLT_MATMUL_DESC_EPILOGUE,
                                    &e, sizeof(e));
    if (c.bias_t >= 0) {
      int32_t bt = c.bias_t;
      hipblasLtMatmulDescSetAttribute(
          desc, HIPBLASLT_MATMUL_DESC_BIAS_DATA_TYPE, &bt, sizeof(bt));
      void* dummy = workspace();
      hipblasLtMatmulDescSetAttribute(
          desc, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &dummy, sizeof(dummy));
    }
    if (c.need_aux) {
      int64_t ld = N;
      void* dummy = workspace();
      hipblasLtMatmulDescSetAttribute(
          desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_POINTER, &dummy,
          sizeof(dummy));
      hipblasLtMatmulDescSetAttribute(
          desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_LD, &ld, sizeof(ld));
      if (c.aux_t >= 0) {
        int32_t at = c.aux_t;
        hipblasLtMatmulDescSetAttribute(
            desc, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_DATA_TYPE, &at,
            sizeof(at));
      }
    }
    hipblasLtMatrixLayout_t la, lb, ld_;
    BLT_CHECK(hipblasLtMatrixLayoutCreate(&la, HIP_R_16BF, K, N, K));
    BLT_CHECK(hipblasLtMatrixLayoutCreate(&lb, HIP_R_16BF, K, M, K));
    BLT_CHECK(hipblasLtMatrixLayoutCreate(&ld_, HIP_R_16BF, N, M, N));
    hipblasLtMatmulPreference_t pref;
    BLT_CHECK(hipblasLtMatmulPreferenceCreate(&pref));
    size_t ws = kMaxWorkspace;
    hipblasLtMatmulPreferenceSetAttribute(
        pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws, sizeof(ws));
    hipblasLtMatmulHeuristicResult_t res[8];
    int nres = 0;
    hipblasStatus_t st = hipblasLtMatmulAlgoGetHeuristic(
        handle(), desc, la, lb, ld_, ld_, pref, 8, res, &nres);
    out[c.label] =
        (st == HIPBLAS_STATUS_SUCCESS) ? nres : -int(st);
    hipblasLtMatmulPreferenceDestroy(pref);
    hipblasLtMatrixLayoutDestroy(la);
    hipblasLtMatrixLayoutDestroy(lb);
    hipblasLtMatrixLayoutDestroy(ld_);
    hipblasLtMatmulDescDestroy(desc);
  }
  return out;
}

}  // namespace

PYBIND11_MODULE(_tepdist_blt, m) {
  m.doc() = "hipBLASLt fused-epilogue GEMMs (GELU_AUX_BIAS / DGELU_BGRAD)";
  m.def("blt_fused", &blt_fused, "fused-epilogue matmul");
  m.def("probe_epilogues", &probe_epilogues, "heuristic support probe");
}
