// pybind11 bindings for the CDNA4 kernel library. Deliberately torch-free:
// tensors cross as raw device pointers + shapes (tepdist_amd/ops/hip.py owns
// allocation via torch and passes torch.cuda.current_stream().cuda_stream).
// This keeps the extension a pure hipcc build (no hipify, no libtorch ABI).

#include <pybind11/pybind11.h>

#include <hip/hip_runtime.h>

#include "common.h"
#include "kernels.h"

namespace py = pybind11;
using namespace tepdist;

namespace {

hipStream_t S(uintptr_t s) { return reinterpret_cast<hipStream_t>(s); }

void check_launch() { HIP_CHECK(hipGetLastError()); }

}  // namespace

PYBIND11_MODULE(_tepdist_hip, m) {
  m.doc() = "TePDist-AMD gfx950 kernel library";

  m.def("gemm", [](uintptr_t A, uintptr_t B, uintptr_t C, uintptr_t Cpre,
                   uintptr_t bias, int M, int N, int K, int lda, int ldb,
                   int ldc, int64_t sa, int64_t sb, int64_t sc, int batch,
                   bool a_kc, bool b_kc, int epi, int split_k,
                   uintptr_t stream) {
    gemm_bf16(reinterpret_cast<void*>(A), reinterpret_cast<void*>(B),
              reinterpret_cast<void*>(C), reinterpret_cast<void*>(Cpre),
              reinterpret_cast<void*>(bias), M, N, K, lda, ldb, ldc, sa, sb,
              sc, batch, a_kc, b_kc, epi, split_k, S(stream));
    check_launch();
  });

  m.def("splitk_reduce", [](uintptr_t parts, uintptr_t out, int nparts,
                            int64_t mn, uintptr_t stream) {
    splitk_reduce(reinterpret_cast<const float*>(parts),
                  reinterpret_cast<void*>(out), nparts, mn, S(stream));
    check_launch();
  });

  m.def("transpose", [](uintptr_t in, uintptr_t out, int R, int C,
                        int64_t si, int64_t so, int batch,
                        uintptr_t stream) {
    transpose_bf16(reinterpret_cast<void*>(in), reinterpret_cast<void*>(out),
                   R, C, si, so, batch, S(stream));
    check_launch();
  });

  m.def("layernorm_fwd", [](uintptr_t x, uintptr_t g, uintptr_t b, uintptr_t y,
                            uintptr_t mean, uintptr_t rstd, int rows, int cols,
                            float eps, uintptr_t stream, uintptr_t res,
                            uintptr_t sum_out) {
    layernorm_fwd_bf16(reinterpret_cast<void*>(x), reinterpret_cast<void*>(g),
                       reinterpret_cast<void*>(b), reinterpret_cast<void*>(y),
                       reinterpret_cast<float*>(mean),
                       reinterpret_cast<float*>(rstd), rows, cols, eps,
                       S(stream), reinterpret_cast<void*>(res),
                       reinterpret_cast<void*>(sum_out));
    check_launch();
  }, pybind11::arg("x"), pybind11::arg("g"), pybind11::arg("b"),
     pybind11::arg("y"), pybind11::arg("mean"), pybind11::arg("rstd"),
     pybind11::arg("rows"), pybind11::arg("cols"), pybind11::arg("eps"),
     pybind11::arg("stream"), pybind11::arg("res") = 0,
     pybind11::arg("sum_out") = 0);

  m.def("layernorm_bwd", [](uintptr_t dy, uintptr_t x, uintptr_t g,
                            uintptr_t mean, uintptr_t rstd, uintptr_t dx,
                            uintptr_t dg_part, uintptr_t db_part,
                            uintptr_t dgamma, uintptr_t dbeta, int rows,
                            int cols, int part_rows, uintptr_t stream,
                            uintptr_t dsum) {
    layernorm_bwd_bf16(reinterpret_cast<void*>(dy), reinterpret_cast<void*>(x),
                       reinterpret_cast<void*>(g),
                       reinterpret_cast<float*>(mean),
                       reinterpret_cast<float*>(rstd),
                       reinterpret_cast<void*>(dx),
                       reinterpret_cast<float*>(dg_part),
                       reinterpret_cast<float*>(db_part), rows, cols,
                       part_rows, S(stream), reinterpret_cast<void*>(dsum));
    check_launch();
    layernorm_bwd_reduce(reinterpret_cast<float*>(dg_part),
                         reinterpret_cast<float*>(db_part),
                         reinterpret_cast<void*>(dgamma),
                         reinterpret_cast<void*>(dbeta), part_rows, cols,
                         S(stream));
    check_launch();
  }, pybind11::arg("dy"), pybind11::arg("x"), pybind11::arg("g"),
     pybind11::arg("mean"), pybind11::arg("rstd"), pybind11::arg("dx"),
     pybind11::arg("dg_part"), pybind11::arg("db_part"),
     pybind11::arg("dgamma"), pybind11::arg("dbeta"), pybind11::arg("rows"),
     pybind11::arg("cols"), pybind11::arg("part_rows"),
     pybind11::arg("stream"), pybind11::arg("dsum") = 0);

  m.def("softmax_fwd", [](uintptr_t x, uintptr_t p, int64_t rows, int cols,
                          int sq, float scale, bool causal, uintptr_t stream) {
    softmax_fwd_bf16(reinterpret_cast<void*>(x), reinterpret_cast<void*>(p),
                     rows, cols, sq, scale, causal, S(stream));
    check_launch();
  });

  m.def("softmax_bwd", [](uintptr_t dp, uintptr_t p, uintptr_t ds,
                          int64_t rows, int cols, float scale,
                          uintptr_t stream) {
    softmax_bwd_bf16(reinterpret_cast<void*>(dp), reinterpret_cast<void*>(p),
                     reinterpret_cast<void*>(ds), rows, cols, scale,
                     S(stream));
    check_launch();
  });

  m.def("attention_fwd", [](uintptr_t q, uintptr_t k, uintptr_t v,
                            uintptr_t o, uintptr_t lse, int B, int H,
                            int seq, int D, float scale, bool causal,
                            int64_t q_bs, int64_t q_hs, int64_t q_rs,
                            int64_t o_bs, int64_t o_hs, int64_t o_rs,
                            uintptr_t stream) {
    attention_fwd_bf16(reinterpret_cast<void*>(q), reinterpret_cast<void*>(k),
                       reinterpret_cast<void*>(v), reinterpret_cast<void*>(o),
                       reinterpret_cast<float*>(lse), B, H, seq, D, scale,
                       causal, q_bs, q_hs, q_rs, o_bs, o_hs, o_rs, S(stream));
    check_launch();
  });

  m.def("attention_bwd", [](uintptr_t q, uintptr_t k, uintptr_t v,
                            uintptr_t o, uintptr_t dout, uintptr_t lse,
                            uintptr_t delta, uintptr_t dq, uintptr_t dk,
                            uintptr_t dv, int B, int H, int seq, int D,
                            float scale, bool causal, int64_t q_bs,
                            int64_t q_hs, int64_t q_rs, int64_t o_bs,
                            int64_t o_hs, int64_t o_rs, uintptr_t stream) {
    attention_bwd_bf16(
        reinterpret_cast<void*>(q), reinterpret_cast<void*>(k),
        reinterpret_cast<void*>(v), reinterpret_cast<void*>(o),
        reinterpret_cast<void*>(dout), reinterpret_cast<float*>(lse),
        reinterpret_cast<float*>(delta), reinterpret_cast<void*>(dq),
        reinterpret_cast<void*>(dk), reinterpret_cast<void*>(dv), B, H, seq,
        D, scale, causal, q_bs, q_hs, q_rs, o_bs, o_hs, o_rs, S(stream));
    check_launch();
  });


  m.def("embedding_fwd", [](uintptr_t ids, uintptr_t table, uintptr_t out,
                            int64_t n_ids, int dim, uintptr_t stream) {
    embedding_fwd_bf16(reinterpret_cast<const int64_t*>(ids),
                       reinterpret_cast<void*>(table),
                       reinterpret_cast<void*>(out), n_ids, dim, S(stream));
    check_launch();
  });

  m.def("embedding_bwd", [](uintptr_t dy, uintptr_t ids, uintptr_t grad_f32,
                            uintptr_t grad_bf16, int64_t n_ids, int vocab,
                            int dim, uintptr_t stream) {
    embedding_bwd_bf16(reinterpret_cast<void*>(dy),
                       reinterpret_cast<const int64_t*>(ids),
                       reinterpret_cast<float*>(grad_f32),
                       reinterpret_cast<void*>(grad_bf16), n_ids, vocab, dim,
                       S(stream));
    check_launch();
  });

  m.def("cross_entropy_fwd", [](uintptr_t logits, uintptr_t targets,
                                uintptr_t nll, uintptr_t lse, int64_t rows,
                                int cols, int ignore, uintptr_t stream) {
    cross_entropy_fwd_bf16(reinterpret_cast<void*>(logits),
                           reinterpret_cast<const int64_t*>(targets),
                           reinterpret_cast<float*>(nll),
                           reinterpret_cast<float*>(lse), rows, cols, ignore,
                           S(stream));
    check_launch();
  });

  m.def("cross_entropy_bwd", [](uintptr_t logits, uintptr_t targets,
                                uintptr_t lse, uintptr_t dscale,
                                uintptr_t dlogits, int64_t rows, int cols,
                                int ignore, uintptr_t stream) {
    cross_entropy_bwd_bf16(reinterpret_cast<void*>(logits),
                           reinterpret_cast<const int64_t*>(targets),
                           reinterpret_cast<const float*>(lse),
                           reinterpret_cast<const float*>(dscale),
                           reinterpret_cast<void*>(dlogits), rows, cols,
                           ignore, S(stream));
    check_launch();
  });

  m.def("dropout_fwd", [](uintptr_t x, uintptr_t y, uintptr_t mask, int64_t n,
                          float p, uint64_t seed, uint64_t offset,
                          uintptr_t stream) {
    dropout_fwd_bf16(reinterpret_cast<void*>(x), reinterpret_cast<void*>(y),
                     reinterpret_cast<uint8_t*>(mask), n, p, seed, offset,
                     S(stream));
    check_launch();
  });

  m.def("dropout_bwd", [](uintptr_t dy, uintptr_t mask, uintptr_t dx,
                          int64_t n, float p, uintptr_t stream) {
    dropout_bwd_bf16(reinterpret_cast<void*>(dy),
                     reinterpret_cast<const uint8_t*>(mask),
                     reinterpret_cast<void*>(dx), n, p, S(stream));
    check_launch();
  });

  m.def("gelu_fwd", [](uintptr_t x, uintptr_t y, int64_t n, uintptr_t stream) {
    gelu_fwd_bf16(reinterpret_cast<void*>(x), reinterpret_cast<void*>(y), n,
                  S(stream));
    check_launch();
  });

  m.def("cast_ws", [](uintptr_t ws, uintptr_t db, int cols,
                      uintptr_t stream) {
    cast_ws_f32_bf16(reinterpret_cast<const float*>(ws),
                     reinterpret_cast<void*>(db), cols, S(stream));
    check_launch();
  });
  m.def("transpose_dy", [](uintptr_t dy, uintptr_t pre, uintptr_t dgt,
                           uintptr_t dgn, uintptr_t bias_ws, int R, int C,
                           uintptr_t stream) {
    transpose_dy_bf16(reinterpret_cast<void*>(dy),
                      reinterpret_cast<void*>(pre),
                      reinterpret_cast<void*>(dgt),
                      reinterpret_cast<void*>(dgn),
                      reinterpret_cast<float*>(bias_ws), R, C, S(stream));
    check_launch();
  });
  m.def("rmsnorm_fwd", [](uintptr_t x, uintptr_t g, uintptr_t y,
                          uintptr_t rstd, int64_t rows, int cols, float eps,
                          uintptr_t stream) {
    rmsnorm_fwd_bf16(reinterpret_cast<void*>(x), reinterpret_cast<void*>(g),
                     reinterpret_cast<void*>(y),
                     reinterpret_cast<float*>(rstd), rows, cols, eps,
                     S(stream));
    check_launch();
  });
  m.def("rmsnorm_bwd", [](uintptr_t dy, uintptr_t x, uintptr_t g,
                          uintptr_t rstd, uintptr_t dx, uintptr_t dg,
                          uintptr_t dg_part, int part_rows, int64_t rows,
                          int cols, uintptr_t stream) {
    rmsnorm_bwd_bf16(reinterpret_cast<void*>(dy), reinterpret_cast<void*>(x),
                     reinterpret_cast<void*>(g),
                     reinterpret_cast<const float*>(rstd),
                     reinterpret_cast<void*>(dx), reinterpret_cast<void*>(dg),
                     reinterpret_cast<float*>(dg_part), part_rows, rows,
                     cols, S(stream));
    check_launch();
  });
  m.def("rope", [](uintptr_t x, uintptr_t y, int64_t tokens, int heads,
                   int D, int seq_len, float theta, bool backward,
                   uintptr_t stream) {
    rope_bf16(reinterpret_cast<void*>(x), reinterpret_cast<void*>(y), tokens,
              heads, D, seq_len, theta, backward, S(stream));
    check_launch();
  });
  m.def("swiglu_fwd", [](uintptr_t a, uintptr_t b, uintptr_t y, int64_t n,
                         uintptr_t stream) {
    swiglu_fwd_bf16(reinterpret_cast<void*>(a), reinterpret_cast<void*>(b),
                    reinterpret_cast<void*>(y), n, S(stream));
    check_launch();
  });
  m.def("swiglu_bwd", [](uintptr_t dy, uintptr_t a, uintptr_t b,
                         uintptr_t da, uintptr_t db, int64_t n,
                         uintptr_t stream) {
    swiglu_bwd_bf16(reinterpret_cast<void*>(dy), reinterpret_cast<void*>(a),
                    reinterpret_cast<void*>(b), reinterpret_cast<void*>(da),
                    reinterpret_cast<void*>(db), n, S(stream));
    check_launch();
  });
  m.def("gelu_bwd", [](uintptr_t dy, uintptr_t x, uintptr_t dx, int64_t n,
                       uintptr_t stream) {
    gelu_bwd_bf16(reinterpret_cast<void*>(dy), reinterpret_cast<void*>(x),
                  reinterpret_cast<void*>(dx), n, S(stream));
    check_launch();
  });

  m.def("bias_sum", [](uintptr_t dy, uintptr_t db, uintptr_t ws, int64_t rows,
                       int cols, uintptr_t stream) {
    bias_sum_bf16(reinterpret_cast<void*>(dy), reinterpret_cast<void*>(db),
                  reinterpret_cast<float*>(ws), rows, cols, S(stream));
    check_launch();
  });

  m.def("adamw", [](uintptr_t param, uintptr_t master, uintptr_t grad_bf16,
                    uintptr_t grad_f32, uintptr_t m_, uintptr_t v_, int64_t n,
                    float lr, float b1, float b2, float eps, float wd,
                    float bc1, float bc2, uintptr_t stream) {
    adamw_bf16(reinterpret_cast<void*>(param),
               reinterpret_cast<float*>(master),
               reinterpret_cast<void*>(grad_bf16),
               reinterpret_cast<const float*>(grad_f32),
               reinterpret_cast<float*>(m_), reinterpret_cast<float*>(v_), n,
               lr, b1, b2, eps, wd, bc1, bc2, S(stream));
    check_launch();
  });

  m.def("tr16_probe", [](uintptr_t a, uintptr_t b, uintptr_t stream) {
    tr16_probe(reinterpret_cast<float*>(a), reinterpret_cast<float*>(b),
               S(stream));
  });

  m.def("adamw_mt", [](uintptr_t tabs, uintptr_t numel, uintptr_t wds,
                       uintptr_t ptypes, uintptr_t chunks, int nchunks,
                       int nt, float lr, float b1, float b2, float eps,
                       float bc1, float bc2, uintptr_t hyper,
                       uintptr_t stream) {
    adamw_mt_bf16(reinterpret_cast<const int64_t*>(tabs),
                  reinterpret_cast<const int64_t*>(numel),
                  reinterpret_cast<const float*>(wds),
                  reinterpret_cast<const unsigned char*>(ptypes),
                  reinterpret_cast<const int*>(chunks), nchunks, nt, lr, b1,
                  b2, eps, bc1, bc2,
                  reinterpret_cast<const float*>(hyper), S(stream));
    check_launch();
  });

  m.def("device_sync", []() { HIP_CHECK(hipDeviceSynchronize()); });
}
