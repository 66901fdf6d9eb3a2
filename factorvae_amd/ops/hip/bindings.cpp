// Torch extension bindings for the FactorVAE gfx950 kernels.
// Thin layer: validate tensors, extract raw pointers, forward to the
// extern "C" launchers defined in the .hip translation units.

#include <torch/extension.h>
// Direct ATen/hip include (torch-ROCm's native header; no CUDA-named
// compatibility path — the hipify build pass is a no-op on this file).
#include <ATen/hip/HIPContext.h>

#include <hip/hip_runtime_api.h>

extern "C" {
hipError_t fv_gemm_nt(const float*, const float*, const float*, float*, int,
                      int, int, float, int, int, hipStream_t);
hipError_t fv_gemm_nn(const float*, const float*, const float*, float*, int,
                      int, int, float, int, int, hipStream_t);
hipError_t fv_gemm_tn(const float*, const float*, float*, float*, float*,
                      float*, int, int, int, int, int, hipStream_t);
hipError_t fv_colsum(const float*, float*, int, int, int, hipStream_t);
hipError_t fv_gemm_nt_bf16(const void*, const void*, const float*, float*,
                           void*, int, int, int, float, int, int,
                           hipStream_t);
hipError_t fv_gemm_nn_bf16(const void*, const void*, const float*, float*,
                           void*, const void*, int, int, int, float, int,
                           int, hipStream_t);
hipError_t fv_gemm_nt_bf16_rs(const void*, const void*, const float*,
                              float*, void*, const void*, int, int, int,
                              int, float, int, hipStream_t);
hipError_t fv_cast_shadows(const float*, void*, void*, int, int, int, int,
                           const float*, void*, void*, int, int, int, int,
                           const float*, void*, long, hipStream_t);
hipError_t fv_gemm_tn_bf16(const void*, const void*, float*, float*, float*,
                           float*, int, int, int, int, int, hipStream_t);
hipError_t fv_cast_f32_bf16(const float*, void*, long, hipStream_t);
hipError_t fv_cast3_f32_bf16(const float*, void*, long, const float*, void*,
                             long, const float*, void*, long, hipStream_t);
hipError_t fv_lrelu_bwd_bf16(const void*, const void*, void*, long,
                             hipStream_t);
hipError_t fv_lrelu_bwd(const float*, const float*, float*, long, hipStream_t);
hipError_t fv_ln_fwd(const float*, const float*, const float*, float*, void*,
                     void*, int, float*, float*, long, int, float,
                     hipStream_t);
hipError_t fv_gemm_nt_fp8(const void*, const void*, const float*,
                          const float*, float*, void*, void*, int, int, int,
                          int, int, int, float, int, int, hipStream_t);
hipError_t fv_absmax_scale(const float*, long, float*, float*, hipStream_t);
hipError_t fv_gemm_nt_fp8_rs(const void*, const void*, const float*,
                             const float*, const float*, const void*,
                             const float*, float*, float*, void*, void*,
                             int, int, int, int, int, float, int, int,
                             hipStream_t);
hipError_t fv_scale_from_amax2(float*, float*, float*, float*, float*,
                               float*, hipStream_t);
hipError_t fv_cast_f32_fp8_damax(const float*, void*, const float*, float*,
                                 long, int, int, hipStream_t);
hipError_t fv_cast_f32_fp8_scaled_t(const float*, void*, const float*, int,
                                    int, int, hipStream_t);
hipError_t fv_cast_f32_fp8_scaled(const float*, void*, const float*, long,
                                  int, int, hipStream_t);
hipError_t fv_ln_bwd_params(const float*, const float*, const float*,
                            const float*, float*, float*, float*, long, int,
                            int, hipStream_t);
hipError_t fv_gru_fwd(const float*, const float*, const float*, float*, float*,
                      float*, float*, int, int, int, hipStream_t);
hipError_t fv_gru_bwd(const float*, const float*, const float*, const float*,
                      float*, float*, void*, void*, int, int, int,
                      hipStream_t);
hipError_t fv_gru_fwd_mfma(const float*, const void*, const float*, float*,
                           float*, float*, float*, int, int, int,
                           hipStream_t);
hipError_t fv_gru_bwd_mfma(const float*, const float*, const float*,
                           const void*, float*, float*, void*, void*,
                           void*, int, const float*, float*, float*,
                           float*, int, int, int, hipStream_t);
hipError_t fv_wgrad_reduce(const float*, float*, long, const float*,
                           float*, long, int, int, hipStream_t);
hipError_t fv_enc_softmax_fwd(const float*, const float*, float*, float*, int,
                              int, hipStream_t);
hipError_t fv_enc_fused_fwd(const float*, const float*, const float*,
                            const float*, float*, float*, float*,
                            const float*, const float*, const float*,
                            const float*, float*, float*, float*, float*,
                            int*, int, int, int, int, hipStream_t);
hipError_t fv_enc_bwd_fused(const float*, const float*, const float*,
                            const float*, const float*, const float*,
                            const float*, const float*, const float*,
                            float*, float*, float*, float*, float*, int,
                            int, int, hipStream_t);
hipError_t fv_enc_softmax_bwd(const float*, const float*, const float*, float*,
                              int, int, hipStream_t);
hipError_t fv_enc_heads_fwd(const float*, const float*, const float*,
                            const float*, const float*, float*, float*, float*,
                            float*, int, int, hipStream_t);
hipError_t fv_enc_heads_bwd(const float*, const float*, const float*,
                            const float*, const float*, const float*,
                            const float*, float*, float*, float*, float*,
                            float*, int, int, hipStream_t);
hipError_t fv_attn_qk_fwd(const float*, const float*, const float*, float*,
                          float*, int, int, hipStream_t);
hipError_t fv_attn_fused_bwd(const float*, const float*, const float*,
                             const float*, const float*, const float*,
                             const float*, const float*, const float*,
                             const float*, const float*, const float*,
                             const int*, const float*, const float*,
                             const float*, const float*, const float*,
                             float*, float*, float*, float*, float*, float*,
                             float*, float*, float*, float*, float*, float*,
                             float*, float*, int, int, int, float, float,
                             hipStream_t);
hipError_t fv_attn_fused_fwd(const float*, const float*, const float*,
                             const float*, const float*, const float*,
                             const float*, const float*, const float*,
                             const float*, const float*, const float*,
                             float*, float*, int*, float*, float*, float*,
                             float*, float*, float*, float*, int, int, int,
                             float, float, hipStream_t);
hipError_t fv_attn_softmax_fwd(const float*, const float*, float*, float*,
                               int*, int, int, float, hipStream_t);
hipError_t fv_attn_ctx_fwd(const float*, const float*, const float*,
                           const int*, float*, int, int, hipStream_t);
hipError_t fv_attn_head_bwd(const float*, const float*, const float*,
                            const int*, float*, float*, float*, int, int,
                            hipStream_t);
hipError_t fv_attn_softmax_bwd(const float*, const float*, const float*,
                               const float*, const int*, float*, float*, int,
                               int, float, float, hipStream_t);
hipError_t fv_attn_qk_bwd(const float*, const float*, const float*,
                          const float*, const float*, float*, float*, float*,
                          int, int, hipStream_t);
hipError_t fv_pred_mlp_fwd(const float*, const float*, const float*,
                           const float*, const float*, const float*,
                           const float*, float*, float*, float*, float*,
                           float*, int, int, hipStream_t);
hipError_t fv_pred_mlp_bwd(const float*, const float*, const float*,
                           const float*, const float*, const float*,
                           const float*, float*, float*, float*, float*,
                           float*, float*, int, int, hipStream_t);
hipError_t fv_dec_fwd(const float*, const float*, const float*, const float*,
                      const float*, const float*, const float*, const float*,
                      const float*, const float*, const float*, const float*,
                      float*, float*, float*, float*, float*, int, int, int,
                      hipStream_t);
hipError_t fv_dec_bwd(const float*, const float*, const float*, const float*,
                      const float*, const float*, const float*, const float*,
                      const float*, const float*, const float*, const float*,
                      const float*, float*, float*, float*, float*, float*,
                      float*, float*, float*, float*, float*, int, int, int,
                      hipStream_t);
hipError_t fv_loss_fwd(const float*, const float*, const float*, const float*,
                       const float*, const float*, float*, float*, float*, int,
                       int, hipStream_t);
hipError_t fv_loss_bwd(const float*, const float*, const float*, const float*,
                       const float*, const float*, float*, float*, float*,
                       float*, float*, int, int, float, hipStream_t);
hipError_t fv_loss_fused(const float*, const float*, const float*,
                         const float*, const float*, const float*, float*,
                         float*, float*, float*, float*, float*, float*,
                         float*, int, int, float, hipStream_t);
hipError_t fv_dh_combine(float*, const float*, const float*, const float*,
                         const float*, const float*, const float*, int, int,
                         int, int, hipStream_t);
hipError_t fv_step_inc(int*, hipStream_t);
hipError_t fv_adam(float*, const float*, float*, float*, const int*, long,
                   float, float, float, float, float, float, hipStream_t);
}

namespace {

inline hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

inline const float* fp(const torch::Tensor& t) { return t.data_ptr<float>(); }
inline float* fpm(torch::Tensor& t) { return t.data_ptr<float>(); }

void check_f32(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == torch::kFloat32, name, " must be fp32");
}

#define CK(t) check_f32(t, #t)

inline void check_bf16(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
}
#define CKB(t) check_bf16(t, #t)
inline void* bfp(torch::Tensor& t) { return t.data_ptr(); }
inline const void* bfpc(const torch::Tensor& t) { return t.data_ptr(); }
#define RUN(call)                                                      \
  do {                                                                 \
    hipError_t e_ = (call);                                            \
    TORCH_CHECK(e_ == hipSuccess, "HIP kernel failed: ", #call, " (", \
                (int)e_, ")");                                         \
  } while (0)

void gemm_nt(torch::Tensor A, torch::Tensor W,
             c10::optional<torch::Tensor> bias, torch::Tensor out,
             double alpha, bool accumulate, bool act_lrelu) {
  CK(A); CK(W); CK(out);
  const int R = A.size(0), Ci = A.size(1), Co = W.size(0);
  TORCH_CHECK(W.size(1) == Ci && out.size(0) == R && out.size(1) == Co);
  const float* b = nullptr;
  if (bias.has_value()) { CK(*bias); b = fp(*bias); }
  RUN(fv_gemm_nt(fp(A), fp(W), b, fpm(out), R, Ci, Co, (float)alpha,
                 accumulate, act_lrelu, cur_stream()));
}

void gemm_nn(torch::Tensor A, torch::Tensor B,
             c10::optional<torch::Tensor> bias, torch::Tensor out,
             double alpha, bool accumulate, bool act_lrelu) {
  CK(A); CK(B); CK(out);
  const int R = A.size(0), Ci = A.size(1), Co = B.size(1);
  TORCH_CHECK(B.size(0) == Ci && out.size(0) == R && out.size(1) == Co);
  const float* b = nullptr;
  if (bias.has_value()) { CK(*bias); b = fp(*bias); }
  RUN(fv_gemm_nn(fp(A), fp(B), b, fpm(out), R, Ci, Co, (float)alpha,
                 accumulate, act_lrelu, cur_stream()));
}

// ---------------------------------------------------------------- bf16
void gemm_nt_bf16(torch::Tensor A, torch::Tensor W,
                  c10::optional<torch::Tensor> bias,
                  c10::optional<torch::Tensor> out_f32,
                  c10::optional<torch::Tensor> out_bf16,
                  double alpha, bool accumulate, bool act_lrelu) {
  CKB(A); CKB(W);
  const int R = A.size(0), Ci = A.size(1), Co = W.size(0);
  TORCH_CHECK(W.size(1) == Ci);
  const float* b = nullptr;
  if (bias.has_value()) { CK(*bias); b = fp(*bias); }
  float* of = nullptr; void* ob = nullptr;
  if (out_f32.has_value()) {
    CK(*out_f32);
    TORCH_CHECK(out_f32->size(0) == R && out_f32->size(1) == Co);
    of = fpm(*out_f32);
  }
  if (out_bf16.has_value()) {
    CKB(*out_bf16);
    TORCH_CHECK(out_bf16->size(0) == R && out_bf16->size(1) == Co);
    ob = bfp(*out_bf16);
  }
  TORCH_CHECK(of || ob, "need at least one output");
  RUN(fv_gemm_nt_bf16(bfpc(A), bfpc(W), b, of, ob, R, Ci, Co, (float)alpha,
                      accumulate, act_lrelu, cur_stream()));
}

void gemm_nn_bf16(torch::Tensor A, torch::Tensor B,
                  c10::optional<torch::Tensor> bias,
                  c10::optional<torch::Tensor> out_f32,
                  c10::optional<torch::Tensor> out_bf16,
                  double alpha, bool accumulate, bool act_lrelu,
                  c10::optional<torch::Tensor> lrelu_bwd_of = c10::nullopt) {
  CKB(A); CKB(B);
  const int R = A.size(0), Ci = A.size(1), Co = B.size(1);
  TORCH_CHECK(B.size(0) == Ci);
  const float* b = nullptr;
  if (bias.has_value()) { CK(*bias); b = fp(*bias); }
  float* of = nullptr; void* ob = nullptr;
  if (out_f32.has_value()) {
    CK(*out_f32);
    TORCH_CHECK(out_f32->size(0) == R && out_f32->size(1) == Co);
    of = fpm(*out_f32);
  }
  if (out_bf16.has_value()) {
    CKB(*out_bf16);
    TORCH_CHECK(out_bf16->size(0) == R && out_bf16->size(1) == Co);
    ob = bfp(*out_bf16);
  }
  TORCH_CHECK(of || ob, "need at least one output");
  const void* yp = nullptr;
  if (lrelu_bwd_of.has_value()) {
    CKB(*lrelu_bwd_of);
    TORCH_CHECK(lrelu_bwd_of->size(0) == R && lrelu_bwd_of->size(1) == Co);
    yp = bfpc(*lrelu_bwd_of);
  }
  RUN(fv_gemm_nn_bf16(bfpc(A), bfpc(B), b, of, ob, yp, R, Ci, Co,
                      (float)alpha, accumulate, act_lrelu, cur_stream()));
}

void gemm_tn_bf16(torch::Tensor A, torch::Tensor B, torch::Tensor out,
                  c10::optional<torch::Tensor> part, long r_chunks,
                  bool accumulate, c10::optional<torch::Tensor> db,
                  c10::optional<torch::Tensor> db_part) {
  CKB(A); CKB(B); CK(out);
  const int R = A.size(0), M = A.size(1), N = B.size(1);
  TORCH_CHECK(B.size(0) == R && out.size(0) == M && out.size(1) == N);
  long zmax = (R + 511) / 512;
  if (zmax > 128) zmax = 128;
  if (zmax < 1 || r_chunks <= 1) zmax = 1;
  float* pp = nullptr;
  if (part.has_value()) {
    CK(*part);
    TORCH_CHECK(part->numel() >= zmax * M * N, "partials too small");
    pp = fpm(*part);
  }
  float* dbp = nullptr; float* dbpp = nullptr;
  if (db.has_value()) {
    CK(*db);
    TORCH_CHECK(db->numel() == M);
    dbp = fpm(*db);
    if (db_part.has_value()) { CK(*db_part); dbpp = fpm(*db_part); }
  }
  RUN(fv_gemm_tn_bf16(bfpc(A), bfpc(B), fpm(out), pp, dbp, dbpp, R, M, N,
                      (int)r_chunks, accumulate, cur_stream()));
}

void gemm_nt_bf16_rs(torch::Tensor A, torch::Tensor Wp,
                     c10::optional<torch::Tensor> bias,
                     c10::optional<torch::Tensor> out_f32,
                     c10::optional<torch::Tensor> out_bf16,
                     c10::optional<torch::Tensor> lrelu_bwd_of,
                     double alpha, bool act_lrelu) {
  CKB(A); CKB(Wp);
  const int R = A.size(0), Ci = A.size(1), KP = Wp.size(1);
  TORCH_CHECK(KP == ((Ci + 31) / 32) * 32, "Wp must be k-padded to 32");
  int Co = -1;
  const float* b = nullptr;
  if (bias.has_value()) { CK(*bias); b = fp(*bias); Co = bias->numel(); }
  float* of = nullptr; void* ob = nullptr;
  if (out_f32.has_value()) {
    CK(*out_f32);
    Co = out_f32->size(1);
    TORCH_CHECK(out_f32->size(0) == R);
    of = fpm(*out_f32);
  }
  if (out_bf16.has_value()) {
    CKB(*out_bf16);
    Co = out_bf16->size(1);
    TORCH_CHECK(out_bf16->size(0) == R);
    ob = bfp(*out_bf16);
  }
  TORCH_CHECK(of || ob, "need at least one output");
  TORCH_CHECK(Wp.size(0) >= Co, "Wp rows < Co");
  const void* yp = nullptr;
  if (lrelu_bwd_of.has_value()) {
    CKB(*lrelu_bwd_of);
    TORCH_CHECK(lrelu_bwd_of->size(0) == R && lrelu_bwd_of->size(1) == Co);
    yp = bfpc(*lrelu_bwd_of);
  }
  RUN(fv_gemm_nt_bf16_rs(bfpc(A), bfpc(Wp), b, of, ob, yp, R, Ci, Co, KP,
                         (float)alpha, act_lrelu, cur_stream()));
}

void cast_shadows(torch::Tensor s1, torch::Tensor d1, torch::Tensor d1t,
                  torch::Tensor s2, torch::Tensor d2, torch::Tensor d2t,
                  torch::Tensor s3, torch::Tensor d3) {
  CK(s1); CKB(d1); CKB(d1t); CK(s2); CKB(d2); CKB(d2t); CK(s3); CKB(d3);
  const int M1 = s1.size(0), N1 = s1.size(1);
  const int M2 = s2.size(0), N2 = s2.size(1);
  TORCH_CHECK(d1.size(0) == M1 && d1t.size(0) == N1);
  TORCH_CHECK(d2.size(0) == M2 && d2t.size(0) == N2);
  RUN(fv_cast_shadows(fp(s1), bfp(d1), bfp(d1t), M1, N1, (int)d1.size(1),
                      (int)d1t.size(1), fp(s2), bfp(d2), bfp(d2t), M2, N2,
                      (int)d2.size(1), (int)d2t.size(1), fp(s3), bfp(d3),
                      s3.numel(), cur_stream()));
}

void cast3_f32_bf16(torch::Tensor s0, torch::Tensor d0, torch::Tensor s1,
                    torch::Tensor d1, torch::Tensor s2, torch::Tensor d2) {
  CK(s0); CKB(d0); CK(s1); CKB(d1); CK(s2); CKB(d2);
  RUN(fv_cast3_f32_bf16(fp(s0), bfp(d0), s0.numel(), fp(s1), bfp(d1),
                        s1.numel(), fp(s2), bfp(d2), s2.numel(),
                        cur_stream()));
}

void cast_f32_bf16(torch::Tensor src, torch::Tensor dst) {
  CK(src); CKB(dst);
  TORCH_CHECK(src.numel() == dst.numel());
  RUN(fv_cast_f32_bf16(fp(src), bfp(dst), src.numel(), cur_stream()));
}

void lrelu_bwd_bf16(torch::Tensor dY, torch::Tensor Y, torch::Tensor dZ) {
  CKB(dY); CKB(Y); CKB(dZ);
  RUN(fv_lrelu_bwd_bf16(bfpc(dY), bfpc(Y), bfp(dZ), dY.numel(),
                        cur_stream()));
}

void gemm_tn(torch::Tensor A, torch::Tensor B, torch::Tensor out,
             c10::optional<torch::Tensor> part, long r_chunks,
             bool accumulate, c10::optional<torch::Tensor> db,
             c10::optional<torch::Tensor> db_part) {
  CK(A); CK(B); CK(out);
  const int R = A.size(0), M = A.size(1), N = B.size(1);
  TORCH_CHECK(B.size(0) == R && out.size(0) == M && out.size(1) == N);
  float* pp = nullptr;
  if (part.has_value()) {
    CK(*part);
    TORCH_CHECK(part->numel() >= (long)32 * M * N,
                "tn partial workspace too small");
    pp = fpm(*part);
  }
  float* dbp = nullptr;
  float* dbpp = nullptr;
  if (db.has_value()) {
    CK(*db);
    TORCH_CHECK(db->numel() == M, "db must be (M,)");
    dbp = fpm(*db);
    if (db_part.has_value()) {
      CK(*db_part);
      TORCH_CHECK(db_part->numel() >= (long)32 * M,
                  "db partial workspace too small");
      dbpp = fpm(*db_part);
    }
  }
  RUN(fv_gemm_tn(fp(A), fp(B), fpm(out), pp, dbp, dbpp, R, M, N,
                 (int)r_chunks, accumulate, cur_stream()));
}

void colsum(torch::Tensor A, torch::Tensor out, long r_chunks) {
  CK(A); CK(out);
  const int R = A.size(0), C = A.size(1);
  TORCH_CHECK(out.numel() == C);
  RUN(fv_colsum(fp(A), fpm(out), R, C, (int)r_chunks, cur_stream()));
}

void lrelu_bwd(torch::Tensor dY, torch::Tensor Y, torch::Tensor dZ) {
  CK(dY); CK(Y); CK(dZ);
  RUN(fv_lrelu_bwd(fp(dY), fp(Y), fpm(dZ), dY.numel(), cur_stream()));
}

void ln_fwd(torch::Tensor x, torch::Tensor gamma, torch::Tensor beta,
            c10::optional<torch::Tensor> xln,
            torch::Tensor mean, torch::Tensor rstd,
            double eps, c10::optional<torch::Tensor> xln_bf = c10::nullopt,
            c10::optional<torch::Tensor> xln_f8 = c10::nullopt) {
  CK(x); CK(gamma); CK(beta); CK(mean); CK(rstd);
  const long R = x.numel() / x.size(-1);
  const int C = x.size(-1);
  float* xo = nullptr;
  if (xln.has_value()) { CK(*xln); xo = fpm(*xln); }
  void* xb = nullptr;
  if (xln_bf.has_value()) { CKB(*xln_bf); xb = bfp(*xln_bf); }
  void* x8 = nullptr;
  int f8_ld = 0;
  if (xln_f8.has_value()) {
    TORCH_CHECK(xln_f8->is_cuda() && xln_f8->is_contiguous());
    TORCH_CHECK(xln_f8->scalar_type() == torch::kFloat8_e4m3fn);
    x8 = xln_f8->data_ptr();
    f8_ld = xln_f8->size(-1);
  }
  TORCH_CHECK(xo || xb || x8, "ln_fwd needs an output");
  RUN(fv_ln_fwd(fp(x), fp(gamma), fp(beta), xo, xb, x8, f8_ld, fpm(mean),
                fpm(rstd), R, C, (float)eps, cur_stream()));
}

inline void check_fp8(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == torch::kFloat8_e4m3fn, name,
              " must be float8_e4m3fn");
}
#define CK8(t) check_fp8(t, #t)

// A/W fp8 (R, lda)/(Co, ldw) padded; logical K = Ci. Outputs optional.
void gemm_nt_fp8(torch::Tensor A, torch::Tensor W,
                 c10::optional<torch::Tensor> bias,
                 c10::optional<torch::Tensor> inv_sw,
                 c10::optional<torch::Tensor> out_f32,
                 c10::optional<torch::Tensor> out_bf16,
                 c10::optional<torch::Tensor> out_fp8,
                 long R, long Ci, long Co, double alpha, bool act_lrelu) {
  CK8(A); CK8(W);
  const int lda = A.size(1), ldw = W.size(1);
  TORCH_CHECK(A.size(0) >= R && lda >= Ci && W.size(0) >= Co && ldw >= Ci);
  const float* b = nullptr;
  if (bias.has_value()) { CK(*bias); b = fp(*bias); }
  const float* isw = nullptr;
  if (inv_sw.has_value()) { CK(*inv_sw); isw = fp(*inv_sw); }
  float* of = nullptr; void* ob = nullptr; void* o8 = nullptr; int ldo = 0;
  if (out_f32.has_value()) { CK(*out_f32); of = fpm(*out_f32); }
  if (out_bf16.has_value()) { CKB(*out_bf16); ob = bfp(*out_bf16); }
  if (out_fp8.has_value()) {
    CK8(*out_fp8);
    o8 = out_fp8->data_ptr();
    ldo = out_fp8->size(-1);
  }
  TORCH_CHECK(of || ob || o8, "need at least one output");
  RUN(fv_gemm_nt_fp8(A.data_ptr(), W.data_ptr(), b, isw, of, ob, o8, ldo,
                     (int)R, (int)Ci, (int)Co, lda, ldw, (float)alpha,
                     act_lrelu, b != nullptr, cur_stream()));
}

void gemm_nt_fp8_rs(torch::Tensor A, torch::Tensor Wp,
                    c10::optional<torch::Tensor> bias,
                    c10::optional<torch::Tensor> inv_sw,
                    c10::optional<torch::Tensor> out_f32,
                    c10::optional<torch::Tensor> out_bf16,
                    c10::optional<torch::Tensor> out_fp8,
                    long R, long Ci, long Co, double alpha, bool act_lrelu,
                    c10::optional<torch::Tensor> inv_sa = c10::nullopt,
                    c10::optional<torch::Tensor> lrelu_bwd_of = c10::nullopt,
                    c10::optional<torch::Tensor> s_out = c10::nullopt,
                    c10::optional<torch::Tensor> amax_out = c10::nullopt) {
  CK8(A); CK8(Wp);
  const int KP = A.size(1);
  TORCH_CHECK(Wp.size(1) == KP && (KP & 127) == 0 && KP >= Ci,
              "A/Wp must share a 128-padded k stride");
  TORCH_CHECK(A.size(0) >= R && Wp.size(0) >= Co);
  const float* b = nullptr;
  if (bias.has_value()) { CK(*bias); b = fp(*bias); }
  const float* isw = nullptr;
  if (inv_sw.has_value()) { CK(*inv_sw); isw = fp(*inv_sw); }
  const float* isa = nullptr;
  if (inv_sa.has_value()) { CK(*inv_sa); isa = fp(*inv_sa); }
  const void* yp = nullptr;
  if (lrelu_bwd_of.has_value()) {
    CKB(*lrelu_bwd_of);
    yp = bfpc(*lrelu_bwd_of);
  }
  const float* sop = nullptr;
  if (s_out.has_value()) { CK(*s_out); sop = fp(*s_out); }
  float* amp = nullptr;
  if (amax_out.has_value()) { CK(*amax_out); amp = fpm(*amax_out); }
  float* of = nullptr; void* ob = nullptr; void* o8 = nullptr; int ldo = 0;
  if (out_f32.has_value()) { CK(*out_f32); of = fpm(*out_f32); }
  if (out_bf16.has_value()) { CKB(*out_bf16); ob = bfp(*out_bf16); }
  if (out_fp8.has_value()) {
    CK8(*out_fp8);
    o8 = out_fp8->data_ptr();
    ldo = out_fp8->size(-1);
  }
  TORCH_CHECK(of || ob || o8, "need at least one output");
  RUN(fv_gemm_nt_fp8_rs(A.data_ptr(), Wp.data_ptr(), b, isw, isa, yp, sop,
                        amp, of, ob, o8, ldo, (int)R, (int)Ci, (int)Co, KP,
                        (float)alpha, act_lrelu, b != nullptr,
                        cur_stream()));
}

void scale_from_amax2(torch::Tensor amax1, torch::Tensor s1,
                      torch::Tensor is1, torch::Tensor amax2,
                      torch::Tensor s2, torch::Tensor is2) {
  CK(amax1); CK(s1); CK(is1); CK(amax2); CK(s2); CK(is2);
  RUN(fv_scale_from_amax2(fpm(amax1), fpm(s1), fpm(is1), fpm(amax2),
                          fpm(s2), fpm(is2), cur_stream()));
}

void cast_f32_fp8_damax(torch::Tensor src, torch::Tensor dst,
                        torch::Tensor s, torch::Tensor amax_out) {
  CK(src); CK8(dst); CK(s); CK(amax_out);
  const long rows = src.numel() / src.size(-1);
  const int cols = src.size(-1);
  TORCH_CHECK(dst.size(-1) >= cols && dst.numel() / dst.size(-1) >= rows);
  RUN(fv_cast_f32_fp8_damax(fp(src), dst.data_ptr(), fp(s), fpm(amax_out),
                            rows, cols, (int)dst.size(-1), cur_stream()));
}

void cast_f32_fp8_scaled_t(torch::Tensor src, torch::Tensor dstT,
                           torch::Tensor scale) {
  CK(src); CK8(dstT); CK(scale);
  const int M = src.size(0), N = src.size(1);
  TORCH_CHECK(dstT.size(0) >= N && dstT.size(1) >= M);
  RUN(fv_cast_f32_fp8_scaled_t(fp(src), dstT.data_ptr(), fp(scale), M, N,
                               (int)dstT.size(1), cur_stream()));
}

void absmax_scale(torch::Tensor src, torch::Tensor scale,
                  torch::Tensor inv_scale) {
  CK(src); CK(scale); CK(inv_scale);
  RUN(fv_absmax_scale(fp(src), src.numel(), fpm(scale), fpm(inv_scale),
                      cur_stream()));
}

void cast_f32_fp8_scaled(torch::Tensor src, torch::Tensor dst,
                         c10::optional<torch::Tensor> scale) {
  CK(src); CK8(dst);
  const long rows = src.numel() / src.size(-1);
  const int cols = src.size(-1);
  TORCH_CHECK(dst.size(-1) >= cols && dst.numel() / dst.size(-1) >= rows);
  const float* sp = nullptr;
  if (scale.has_value()) { CK(*scale); sp = fp(*scale); }
  RUN(fv_cast_f32_fp8_scaled(fp(src), dst.data_ptr(), sp, rows, cols,
                             (int)dst.size(-1), cur_stream()));
}

void ln_bwd_params(torch::Tensor x, torch::Tensor dxln, torch::Tensor mean,
                   torch::Tensor rstd, torch::Tensor part,
                   torch::Tensor dgamma, torch::Tensor dbeta, long r_chunks) {
  CK(x); CK(dxln); CK(mean); CK(rstd); CK(part); CK(dgamma); CK(dbeta);
  const long R = x.numel() / x.size(-1);
  const int C = x.size(-1);
  RUN(fv_ln_bwd_params(fp(x), fp(dxln), fp(mean), fp(rstd), fpm(part),
                       fpm(dgamma), fpm(dbeta), R, C, (int)r_chunks,
                       cur_stream()));
}

void gru_fwd(torch::Tensor gi, torch::Tensor Whh, torch::Tensor bhh,
             torch::Tensor h_final, c10::optional<torch::Tensor> h_seq,
             torch::Tensor h_prev, torch::Tensor gates4, long N, long T,
             long H) {
  CK(gi); CK(Whh); CK(bhh); CK(h_final); CK(h_prev); CK(gates4);
  float* hs = nullptr;
  if (h_seq.has_value()) { CK(*h_seq); hs = fpm(*h_seq); }
  RUN(fv_gru_fwd(fp(gi), fp(Whh), fp(bhh), fpm(h_final), hs,
                 fpm(h_prev), fpm(gates4), (int)N, (int)T, (int)H,
                 cur_stream()));
}

void gru_bwd(torch::Tensor dh_final, torch::Tensor h_prev, torch::Tensor gates4,
             torch::Tensor Whh, torch::Tensor dgi, torch::Tensor dgh, long N,
             long T, long H,
             c10::optional<torch::Tensor> dgi_bf = c10::nullopt,
             c10::optional<torch::Tensor> dgh_bf = c10::nullopt) {
  CK(dh_final); CK(h_prev); CK(gates4); CK(Whh); CK(dgi); CK(dgh);
  void* gib = nullptr; void* ghb = nullptr;
  if (dgi_bf.has_value()) { CKB(*dgi_bf); gib = bfp(*dgi_bf); }
  if (dgh_bf.has_value()) { CKB(*dgh_bf); ghb = bfp(*dgh_bf); }
  RUN(fv_gru_bwd(fp(dh_final), fp(h_prev), fp(gates4), fp(Whh), fpm(dgi),
                 fpm(dgh), gib, ghb, (int)N, (int)T, (int)H, cur_stream()));
}

void gru_fwd_mfma(torch::Tensor gi, torch::Tensor whh_bf, torch::Tensor bhh,
                  torch::Tensor h_final, c10::optional<torch::Tensor> h_seq,
                  torch::Tensor h_prev, torch::Tensor gates4, long N, long T,
                  long H) {
  CK(gi); CKB(whh_bf); CK(bhh); CK(h_final); CK(h_prev);
  CK(gates4);
  float* hs = nullptr;
  if (h_seq.has_value()) { CK(*h_seq); hs = fpm(*h_seq); }
  RUN(fv_gru_fwd_mfma(fp(gi), bfpc(whh_bf), fp(bhh), fpm(h_final),
                      hs, fpm(h_prev), fpm(gates4), (int)N, (int)T,
                      (int)H, cur_stream()));
}

void gru_bwd_mfma(torch::Tensor dh_final, torch::Tensor h_prev,
                  torch::Tensor gates4, torch::Tensor whh_bf,
                  c10::optional<torch::Tensor> dgi,
                  c10::optional<torch::Tensor> dgh, long N, long T,
                  long H,
                  c10::optional<torch::Tensor> dgi_bf = c10::nullopt,
                  c10::optional<torch::Tensor> dgh_bf = c10::nullopt,
                  c10::optional<torch::Tensor> dgi_f8 = c10::nullopt,
                  c10::optional<torch::Tensor> s_dgi = c10::nullopt,
                  c10::optional<torch::Tensor> amax_dgi = c10::nullopt,
                  c10::optional<torch::Tensor> whh_part = c10::nullopt,
                  c10::optional<torch::Tensor> bhh_part = c10::nullopt) {
  CK(dh_final); CK(h_prev); CK(gates4); CKB(whh_bf);
  float* gi = nullptr; float* gh = nullptr;
  if (dgi.has_value()) { CK(*dgi); gi = fpm(*dgi); }
  if (dgh.has_value()) { CK(*dgh); gh = fpm(*dgh); }
  void* gib = nullptr; void* ghb = nullptr;
  if (dgi_bf.has_value()) { CKB(*dgi_bf); gib = bfp(*dgi_bf); }
  if (dgh_bf.has_value()) { CKB(*dgh_bf); ghb = bfp(*dgh_bf); }
  void* g8 = nullptr; int ld8 = 0;
  const float* sd = nullptr; float* am = nullptr;
  if (dgi_f8.has_value()) {
    CK8(*dgi_f8);
    g8 = dgi_f8->data_ptr();
    ld8 = dgi_f8->size(-1);
    TORCH_CHECK(s_dgi.has_value() && amax_dgi.has_value(),
                "dgi_f8 needs s_dgi + amax_dgi");
    CK(*s_dgi); CK(*amax_dgi);
    sd = fp(*s_dgi);
    am = fpm(*amax_dgi);
  }
  TORCH_CHECK(gi || gib || g8, "gru_bwd_mfma needs a dgi output");
  float* wp = nullptr; float* bp = nullptr;
  if (whh_part.has_value()) {
    CK(*whh_part);
    TORCH_CHECK(bhh_part.has_value(), "whh_part needs bhh_part");
    CK(*bhh_part);
    const long nblk = (N + 15) / 16;
    TORCH_CHECK(whh_part->numel() >= nblk * 192 * 64 &&
                bhh_part->numel() >= nblk * 192, "wgrad partials too small");
    wp = fpm(*whh_part);
    bp = fpm(*bhh_part);
  }
  RUN(fv_gru_bwd_mfma(fp(dh_final), fp(h_prev), fp(gates4), bfpc(whh_bf),
                      gi, gh, gib, ghb, g8, ld8, sd, am, wp, bp, (int)N,
                      (int)T, (int)H, cur_stream()));
}

void wgrad_reduce(torch::Tensor part, torch::Tensor out,
                  torch::Tensor db_part, torch::Tensor db, long z,
                  bool accumulate) {
  CK(part); CK(out); CK(db_part); CK(db);
  RUN(fv_wgrad_reduce(fp(part), fpm(out), out.numel(), fp(db_part),
                      fpm(db), db.numel(), (int)z, accumulate,
                      cur_stream()));
}

void attn_fused_fwd(torch::Tensor h, torch::Tensor qk, torch::Tensor cb,
                    c10::optional<torch::Tensor> mask, torch::Tensor Wv,
                    torch::Tensor bv, torch::Tensor Wl, torch::Tensor bl,
                    torch::Tensor wmu, torch::Tensor bmu, torch::Tensor wsig,
                    torch::Tensor bsig, torch::Tensor a, torch::Tensor sd,
                    torch::Tensor guard, torch::Tensor u, torch::Tensor ctx,
                    torch::Tensor hm2, torch::Tensor pmu,
                    torch::Tensor psig_pre, torch::Tensor psig,
                    torch::Tensor psig_c, double alpha, double keep_inv) {
  CK(h); CK(qk); CK(cb); CK(Wv); CK(bv); CK(Wl); CK(bl); CK(wmu); CK(bmu);
  CK(wsig); CK(bsig); CK(a); CK(sd); CK(u); CK(ctx); CK(hm2); CK(pmu);
  CK(psig_pre); CK(psig); CK(psig_c);
  TORCH_CHECK(guard.scalar_type() == torch::kInt32);
  const int N = h.size(0), H = h.size(1), K = qk.size(0);
  const float* mp = nullptr;
  if (mask.has_value()) { CK(*mask); mp = fp(*mask); }
  RUN(fv_attn_fused_fwd(fp(h), fp(qk), fp(cb), mp, fp(Wv), fp(bv), fp(Wl),
                        fp(bl), fp(wmu), fp(bmu), fp(wsig), fp(bsig),
                        fpm(a), fpm(sd), guard.data_ptr<int>(), fpm(u),
                        fpm(ctx), fpm(hm2), fpm(pmu), fpm(psig_pre),
                        fpm(psig), fpm(psig_c), N, K, H, (float)alpha,
                        (float)keep_inv, cur_stream()));
}

void attn_fused_bwd(torch::Tensor dpmu, torch::Tensor dpsig_c,
                    torch::Tensor psig, torch::Tensor psig_pre,
                    torch::Tensor hm2, torch::Tensor wmu, torch::Tensor wsig,
                    torch::Tensor Wl, torch::Tensor h, torch::Tensor a,
                    torch::Tensor sd, c10::optional<torch::Tensor> mask,
                    torch::Tensor guard, torch::Tensor u, torch::Tensor Wv,
                    torch::Tensor q, torch::Tensor Wk, torch::Tensor bk,
                    torch::Tensor dz2, torch::Tensor du, torch::Tensor ds,
                    torch::Tensor dc, torch::Tensor dWv, torch::Tensor dbv,
                    torch::Tensor dq, torch::Tensor dWk, torch::Tensor dbk,
                    torch::Tensor hpart,
                    torch::Tensor dwmu, torch::Tensor dbmu,
                    torch::Tensor dwsig, torch::Tensor dbsig,
                    double alpha, double keep_inv) {
  CK(h); CK(a); CK(sd); CK(u); CK(Wv); CK(q); CK(Wk); CK(bk); CK(dz2);
  CK(du); CK(ds); CK(dc); CK(dWv); CK(dbv); CK(dq); CK(dWk); CK(dbk);
  CK(hpart);
  const int N = h.size(0), H = h.size(1), K = q.size(0);
  TORCH_CHECK(hpart.numel() >= (long)K * (2 * H + 2));
  const float* mp = nullptr;
  if (mask.has_value()) { CK(*mask); mp = fp(*mask); }
  RUN(fv_attn_fused_bwd(fp(dpmu), fp(dpsig_c), fp(psig), fp(psig_pre),
                        fp(hm2), fp(wmu), fp(wsig), fp(Wl), fp(h), fp(a),
                        fp(sd), mp, guard.data_ptr<int>(), fp(u), fp(Wv),
                        fp(q), fp(Wk), fp(bk), fpm(dz2), fpm(du), fpm(ds),
                        fpm(dc), fpm(dWv), fpm(dbv), fpm(dq), fpm(dWk),
                        fpm(dbk), fpm(hpart), fpm(dwmu), fpm(dbmu),
                        fpm(dwsig), fpm(dbsig), N, K, H, (float)alpha,
                        (float)keep_inv, cur_stream()));
}

void enc_fused_fwd(torch::Tensor h, torch::Tensor Wenc, torch::Tensor benc,
                   torch::Tensor y, torch::Tensor scores, torch::Tensor a,
                   torch::Tensor yp,
                   c10::optional<torch::Tensor> Wmu = c10::nullopt,
                   c10::optional<torch::Tensor> bmu = c10::nullopt,
                   c10::optional<torch::Tensor> Wsig = c10::nullopt,
                   c10::optional<torch::Tensor> bsig = c10::nullopt,
                   c10::optional<torch::Tensor> fmu = c10::nullopt,
                   c10::optional<torch::Tensor> fsig_pre = c10::nullopt,
                   c10::optional<torch::Tensor> fsig = c10::nullopt,
                   c10::optional<torch::Tensor> fsig_c = c10::nullopt,
                   c10::optional<torch::Tensor> done = c10::nullopt) {
  CK(h); CK(Wenc); CK(benc); CK(y); CK(scores); CK(a); CK(yp);
  const int N = h.size(0), H = h.size(1), M = Wenc.size(0);
  const float* wm = nullptr; const float* bm = nullptr;
  const float* ws = nullptr; const float* bs = nullptr;
  float* fm = nullptr; float* fsp = nullptr; float* fs = nullptr;
  float* fsc = nullptr; int* dn = nullptr; int K = 0;
  if (fmu.has_value()) {
    CK(*Wmu); CK(*bmu); CK(*Wsig); CK(*bsig); CK(*fmu); CK(*fsig_pre);
    CK(*fsig); CK(*fsig_c);
    TORCH_CHECK(done.has_value() &&
                done->scalar_type() == torch::kInt32 && done->is_cuda(),
                "fused heads need an int32 done counter");
    wm = fp(*Wmu); bm = fp(*bmu); ws = fp(*Wsig); bs = fp(*bsig);
    fm = fpm(*fmu); fsp = fpm(*fsig_pre); fs = fpm(*fsig);
    fsc = fpm(*fsig_c);
    dn = done->data_ptr<int>();
    K = fmu->numel();
    TORCH_CHECK(Wmu->size(0) == K && Wmu->size(1) == M);
  }
  RUN(fv_enc_fused_fwd(fp(h), fp(Wenc), fp(benc), fp(y), fpm(scores),
                       fpm(a), fpm(yp), wm, bm, ws, bs, fm, fsp, fs, fsc,
                       dn, K, N, M, H, cur_stream()));
}

void enc_bwd_fused(torch::Tensor dfmu, torch::Tensor dfsig_c,
                   torch::Tensor fsig, torch::Tensor fsig_pre,
                   torch::Tensor yp, torch::Tensor Wmu, torch::Tensor Wsig,
                   torch::Tensor a, torch::Tensor y, torch::Tensor dscores,
                   torch::Tensor dWmu, torch::Tensor dbmu,
                   torch::Tensor dWsig, torch::Tensor dbsig) {
  CK(dfmu); CK(fsig); CK(yp); CK(Wmu); CK(Wsig); CK(a); CK(y); CK(dscores);
  CK(dWmu); CK(dbmu); CK(dWsig); CK(dbsig);
  const int N = a.size(0), M = a.size(1), K = Wmu.size(0);
  RUN(fv_enc_bwd_fused(fp(dfmu), fp(dfsig_c), fp(fsig), fp(fsig_pre),
                       fp(yp), fp(Wmu), fp(Wsig), fp(a), fp(y),
                       fpm(dscores), fpm(dWmu), fpm(dbmu), fpm(dWsig),
                       fpm(dbsig), N, M, K, cur_stream()));
}

void enc_softmax_fwd(torch::Tensor scores, torch::Tensor y, torch::Tensor a,
                     torch::Tensor yp) {
  CK(scores); CK(y); CK(a); CK(yp);
  RUN(fv_enc_softmax_fwd(fp(scores), fp(y), fpm(a), fpm(yp),
                         scores.size(0), scores.size(1), cur_stream()));
}

void enc_softmax_bwd(torch::Tensor dyp, torch::Tensor a, torch::Tensor y,
                     torch::Tensor dscores) {
  CK(dyp); CK(a); CK(y); CK(dscores);
  RUN(fv_enc_softmax_bwd(fp(dyp), fp(a), fp(y), fpm(dscores), a.size(0),
                         a.size(1), cur_stream()));
}

void enc_heads_fwd(torch::Tensor yp, torch::Tensor Wmu, torch::Tensor bmu,
                   torch::Tensor Wsig, torch::Tensor bsig, torch::Tensor fmu,
                   torch::Tensor fsig_pre, torch::Tensor fsig,
                   torch::Tensor fsig_c) {
  CK(yp); CK(Wmu); CK(bmu); CK(Wsig); CK(bsig); CK(fmu); CK(fsig_pre);
  CK(fsig); CK(fsig_c);
  RUN(fv_enc_heads_fwd(fp(yp), fp(Wmu), fp(bmu), fp(Wsig), fp(bsig), fpm(fmu),
                       fpm(fsig_pre), fpm(fsig), fpm(fsig_c), yp.numel(),
                       Wmu.size(0), cur_stream()));
}

void enc_heads_bwd(torch::Tensor dfmu, torch::Tensor dfsig_c,
                   torch::Tensor fsig, torch::Tensor fsig_pre,
                   torch::Tensor yp, torch::Tensor Wmu, torch::Tensor Wsig,
                   torch::Tensor dyp, torch::Tensor dWmu, torch::Tensor dbmu,
                   torch::Tensor dWsig, torch::Tensor dbsig) {
  CK(dfmu); CK(dfsig_c); CK(fsig); CK(fsig_pre); CK(yp); CK(Wmu); CK(Wsig);
  CK(dyp); CK(dWmu); CK(dbmu); CK(dWsig); CK(dbsig);
  RUN(fv_enc_heads_bwd(fp(dfmu), fp(dfsig_c), fp(fsig), fp(fsig_pre), fp(yp),
                       fp(Wmu), fp(Wsig), fpm(dyp), fpm(dWmu), fpm(dbmu),
                       fpm(dWsig), fpm(dbsig), yp.numel(), Wmu.size(0),
                       cur_stream()));
}

void attn_qk_fwd(torch::Tensor q, torch::Tensor Wk, torch::Tensor bk,
                 torch::Tensor qk, torch::Tensor c) {
  CK(q); CK(Wk); CK(bk); CK(qk); CK(c);
  RUN(fv_attn_qk_fwd(fp(q), fp(Wk), fp(bk), fpm(qk), fpm(c), q.size(0),
                     q.size(1), cur_stream()));
}

void attn_softmax_fwd(torch::Tensor s, c10::optional<torch::Tensor> mask,
                      torch::Tensor a, torch::Tensor sd, torch::Tensor guard,
                      double keep_inv) {
  CK(s); CK(a); CK(sd);
  TORCH_CHECK(guard.scalar_type() == torch::kInt32 && guard.is_cuda());
  const float* mp = nullptr;
  if (mask.has_value()) { CK(*mask); mp = fp(*mask); }
  RUN(fv_attn_softmax_fwd(fp(s), mp, fpm(a), fpm(sd), guard.data_ptr<int>(),
                          s.size(0), s.size(1), (float)keep_inv,
                          cur_stream()));
}

void attn_ctx_fwd(torch::Tensor u, torch::Tensor Wv, torch::Tensor bv,
                  torch::Tensor guard, torch::Tensor ctx) {
  CK(u); CK(Wv); CK(bv); CK(ctx);
  RUN(fv_attn_ctx_fwd(fp(u), fp(Wv), fp(bv), guard.data_ptr<int>(), fpm(ctx),
                      u.size(0), u.size(1), cur_stream()));
}

void attn_head_bwd(torch::Tensor dctx, torch::Tensor u, torch::Tensor Wv,
                   torch::Tensor guard, torch::Tensor du, torch::Tensor dWv,
                   torch::Tensor dbv) {
  CK(dctx); CK(u); CK(Wv); CK(du); CK(dWv); CK(dbv);
  RUN(fv_attn_head_bwd(fp(dctx), fp(u), fp(Wv), guard.data_ptr<int>(),
                       fpm(du), fpm(dWv), fpm(dbv), u.size(0), u.size(1),
                       cur_stream()));
}

void attn_softmax_bwd(torch::Tensor da, torch::Tensor a, torch::Tensor sd,
                      c10::optional<torch::Tensor> mask, torch::Tensor guard,
                      torch::Tensor ds, torch::Tensor dc, double keep_inv,
                      double alpha) {
  CK(da); CK(a); CK(sd); CK(ds); CK(dc);
  const float* mp = nullptr;
  if (mask.has_value()) { CK(*mask); mp = fp(*mask); }
  RUN(fv_attn_softmax_bwd(fp(da), fp(a), fp(sd), mp, guard.data_ptr<int>(),
                          fpm(ds), fpm(dc), a.size(0), a.size(1),
                          (float)keep_inv, (float)alpha, cur_stream()));
}

void attn_qk_bwd(torch::Tensor dqk, torch::Tensor dc, torch::Tensor q,
                 torch::Tensor Wk, torch::Tensor bk, torch::Tensor dq,
                 torch::Tensor dWk, torch::Tensor dbk) {
  CK(dqk); CK(dc); CK(q); CK(Wk); CK(bk); CK(dq); CK(dWk); CK(dbk);
  RUN(fv_attn_qk_bwd(fp(dqk), fp(dc), fp(q), fp(Wk), fp(bk), fpm(dq),
                     fpm(dWk), fpm(dbk), q.size(0), q.size(1), cur_stream()));
}

void pred_mlp_fwd(torch::Tensor ctx, torch::Tensor Wl, torch::Tensor bl,
                  torch::Tensor wmu, torch::Tensor bmu, torch::Tensor wsig,
                  torch::Tensor bsig, torch::Tensor hm2, torch::Tensor pmu,
                  torch::Tensor psig_pre, torch::Tensor psig,
                  torch::Tensor psig_c) {
  CK(ctx); CK(Wl); CK(bl); CK(wmu); CK(bmu); CK(wsig); CK(bsig); CK(hm2);
  CK(pmu); CK(psig_pre); CK(psig); CK(psig_c);
  RUN(fv_pred_mlp_fwd(fp(ctx), fp(Wl), fp(bl), fp(wmu), fp(bmu), fp(wsig),
                      fp(bsig), fpm(hm2), fpm(pmu), fpm(psig_pre), fpm(psig),
                      fpm(psig_c), ctx.size(0), ctx.size(1), cur_stream()));
}

void pred_mlp_bwd(torch::Tensor dpmu, torch::Tensor dpsig_c, torch::Tensor psig,
                  torch::Tensor psig_pre, torch::Tensor hm2, torch::Tensor wmu,
                  torch::Tensor wsig, torch::Tensor dz2, torch::Tensor hpart,
                  torch::Tensor dwmu, torch::Tensor dbmu, torch::Tensor dwsig,
                  torch::Tensor dbsig) {
  CK(dpmu); CK(dpsig_c); CK(psig); CK(psig_pre); CK(hm2); CK(wmu); CK(wsig);
  CK(dz2); CK(hpart); CK(dwmu); CK(dbmu); CK(dwsig); CK(dbsig);
  TORCH_CHECK(hpart.numel() >= (long)hm2.size(0) * (2 * hm2.size(1) + 2));
  RUN(fv_pred_mlp_bwd(fp(dpmu), fp(dpsig_c), fp(psig), fp(psig_pre), fp(hm2),
                      fp(wmu), fp(wsig), fpm(dz2), fpm(hpart), fpm(dwmu),
                      fpm(dbmu), fpm(dwsig), fpm(dbsig), hm2.size(0),
                      hm2.size(1), cur_stream()));
}

void dec_fwd(torch::Tensor h, torch::Tensor W1, torch::Tensor b1,
             torch::Tensor wmu, torch::Tensor bmu, torch::Tensor wsig,
             torch::Tensor bsig, torch::Tensor Wb, torch::Tensor bb,
             torch::Tensor fmu, torch::Tensor fsig_c, torch::Tensor eps,
             torch::Tensor recon, torch::Tensor a1, torch::Tensor beta,
             torch::Tensor asig_pre, torch::Tensor sigma) {
  CK(h); CK(W1); CK(b1); CK(Wb); CK(bb); CK(fmu); CK(fsig_c); CK(eps);
  CK(recon); CK(a1); CK(beta); CK(asig_pre); CK(sigma);
  RUN(fv_dec_fwd(fp(h), fp(W1), fp(b1), fp(wmu), fp(bmu), fp(wsig), fp(bsig),
                 fp(Wb), fp(bb), fp(fmu), fp(fsig_c), fp(eps), fpm(recon),
                 fpm(a1), fpm(beta), fpm(asig_pre), fpm(sigma), h.size(0),
                 Wb.size(0), h.size(1), cur_stream()));
}

void dec_bwd(torch::Tensor drecon, torch::Tensor h, torch::Tensor a1,
             torch::Tensor beta, torch::Tensor asig_pre, torch::Tensor sigma,
             torch::Tensor eps, torch::Tensor fmu, torch::Tensor fsig_c,
             torch::Tensor W1, torch::Tensor wmu, torch::Tensor wsig,
             torch::Tensor Wb, torch::Tensor dh, torch::Tensor dz1,
             torch::Tensor dbeta, torch::Tensor part, torch::Tensor dfmu,
             torch::Tensor dfsig_c, torch::Tensor dwmu, torch::Tensor dbmu,
             torch::Tensor dwsig, torch::Tensor dbsig) {
  CK(drecon); CK(h); CK(a1); CK(beta); CK(asig_pre); CK(sigma); CK(eps);
  CK(fmu); CK(fsig_c); CK(W1); CK(wmu); CK(wsig); CK(Wb); CK(dh); CK(dz1);
  CK(dbeta); CK(part); CK(dfmu); CK(dfsig_c); CK(dwmu); CK(dbmu); CK(dwsig);
  CK(dbsig);
  const int N_ = h.size(0), K_ = Wb.size(0), H_ = h.size(1);
  long iters = (N_ + 4 * 128 - 1) / (4 * 128);
  if (iters > 8) iters = 8;
  if (iters < 1) iters = 1;
  const long nblk = (N_ + 4 * iters - 1) / (4 * iters);
  TORCH_CHECK(part.numel() >= nblk * (2 * K_ + 2 * H_ + 2));
  RUN(fv_dec_bwd(fp(drecon), fp(h), fp(a1), fp(beta), fp(asig_pre), fp(sigma),
                 fp(eps), fp(fmu), fp(fsig_c), fp(W1), fp(wmu), fp(wsig),
                 fp(Wb), fpm(dh), fpm(dz1), fpm(dbeta), fpm(part), fpm(dfmu),
                 fpm(dfsig_c), fpm(dwmu), fpm(dbmu), fpm(dwsig), fpm(dbsig),
                 N_, K_, H_, cur_stream()));
}

void loss_fwd(torch::Tensor recon, torch::Tensor y, torch::Tensor fmu,
              torch::Tensor fsig_c, torch::Tensor pmu, torch::Tensor psig_c,
              torch::Tensor loss, torch::Tensor mse, torch::Tensor kl) {
  CK(recon); CK(y); CK(fmu); CK(fsig_c); CK(pmu); CK(psig_c); CK(loss);
  CK(mse); CK(kl);
  RUN(fv_loss_fwd(fp(recon), fp(y), fp(fmu), fp(fsig_c), fp(pmu), fp(psig_c),
                  fpm(loss), fpm(mse), fpm(kl), recon.numel(), fmu.numel(),
                  cur_stream()));
}

void loss_bwd(torch::Tensor recon, torch::Tensor y, torch::Tensor fmu,
              torch::Tensor fsig_c, torch::Tensor pmu, torch::Tensor psig_c,
              torch::Tensor drecon, torch::Tensor dfmu, torch::Tensor dfsig_c,
              torch::Tensor dpmu, torch::Tensor dpsig_c, double gscale) {
  CK(recon); CK(y); CK(drecon); CK(dfmu); CK(dfsig_c); CK(dpmu); CK(dpsig_c);
  RUN(fv_loss_bwd(fp(recon), fp(y), fp(fmu), fp(fsig_c), fp(pmu), fp(psig_c),
                  fpm(drecon), fpm(dfmu), fpm(dfsig_c), fpm(dpmu),
                  fpm(dpsig_c), recon.numel(), fmu.numel(), (float)gscale,
                  cur_stream()));
}

void loss_fused(torch::Tensor recon, torch::Tensor y, torch::Tensor fmu,
                torch::Tensor fsig_c, torch::Tensor pmu, torch::Tensor psig_c,
                torch::Tensor loss, torch::Tensor mse, torch::Tensor kl,
                torch::Tensor drecon, torch::Tensor dfmu,
                torch::Tensor dfsig_c, torch::Tensor dpmu,
                torch::Tensor dpsig_c, double gscale) {
  CK(recon); CK(y); CK(fmu); CK(fsig_c); CK(pmu); CK(psig_c); CK(loss);
  CK(mse); CK(kl); CK(drecon); CK(dfmu); CK(dfsig_c); CK(dpmu); CK(dpsig_c);
  RUN(fv_loss_fused(fp(recon), fp(y), fp(fmu), fp(fsig_c), fp(pmu),
                    fp(psig_c), fpm(loss), fpm(mse), fpm(kl), fpm(drecon),
                    fpm(dfmu), fpm(dfsig_c), fpm(dpmu), fpm(dpsig_c),
                    recon.numel(), fmu.numel(), (float)gscale, cur_stream()));
}

void dh_combine(torch::Tensor dh, torch::Tensor ds, torch::Tensor qk,
                torch::Tensor a, torch::Tensor du, torch::Tensor dscores,
                torch::Tensor Wenc) {
  CK(dh); CK(ds); CK(qk); CK(a); CK(du); CK(dscores); CK(Wenc);
  const int N_ = dh.size(0), H_ = dh.size(1);
  const int K_ = qk.size(0), M_ = Wenc.size(0);
  TORCH_CHECK(ds.size(0) == N_ && ds.size(1) == K_, "ds shape");
  TORCH_CHECK(a.sizes() == ds.sizes(), "a shape");
  TORCH_CHECK(du.sizes() == qk.sizes(), "du shape");
  TORCH_CHECK(dscores.size(0) == N_ && dscores.size(1) == M_, "dscores");
  TORCH_CHECK(Wenc.size(1) == H_ && qk.size(1) == H_, "B cols");
  RUN(fv_dh_combine(fpm(dh), fp(ds), fp(qk), fp(a), fp(du), fp(dscores),
                    fp(Wenc), N_, K_, M_, H_, cur_stream()));
}

void step_inc(torch::Tensor step_t) {
  TORCH_CHECK(step_t.scalar_type() == torch::kInt32 && step_t.is_cuda());
  RUN(fv_step_inc(step_t.data_ptr<int>(), cur_stream()));
}

void adam(torch::Tensor p, torch::Tensor g, torch::Tensor m, torch::Tensor v,
          torch::Tensor step_t, double lr0, double eta_min, double t_max,
          double beta1, double beta2, double eps) {
  CK(p); CK(g); CK(m); CK(v);
  TORCH_CHECK(step_t.scalar_type() == torch::kInt32 && step_t.is_cuda());
  RUN(fv_adam(fpm(p), fp(g), fpm(m), fpm(v), step_t.data_ptr<int>(), p.numel(),
              (float)lr0, (float)eta_min, (float)t_max, (float)beta1,
              (float)beta2, (float)eps, cur_stream()));
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("gemm_nt", &gemm_nt);
  mod.def("gemm_nn", &gemm_nn);
  mod.def("gemm_tn", &gemm_tn, py::arg("A"), py::arg("B"), py::arg("out"),
          py::arg("part"), py::arg("r_chunks"), py::arg("accumulate"),
          py::arg("db") = py::none(), py::arg("db_part") = py::none());
  mod.def("colsum", &colsum);
  mod.def("gemm_nt_bf16", &gemm_nt_bf16, py::arg("A"), py::arg("W"),
          py::arg("bias"), py::arg("out_f32"), py::arg("out_bf16"),
          py::arg("alpha"), py::arg("accumulate"), py::arg("act_lrelu"));
  mod.def("gemm_nn_bf16", &gemm_nn_bf16, py::arg("A"), py::arg("B"),
          py::arg("bias"), py::arg("out_f32"), py::arg("out_bf16"),
          py::arg("alpha"), py::arg("accumulate"), py::arg("act_lrelu"),
          py::arg("lrelu_bwd_of") = py::none());
  mod.def("gemm_tn_bf16", &gemm_tn_bf16, py::arg("A"), py::arg("B"),
          py::arg("out"), py::arg("part"), py::arg("r_chunks"),
          py::arg("accumulate"), py::arg("db") = py::none(),
          py::arg("db_part") = py::none());
  mod.def("cast_f32_bf16", &cast_f32_bf16);
  mod.def("cast3_f32_bf16", &cast3_f32_bf16);
  mod.def("lrelu_bwd_bf16", &lrelu_bwd_bf16);
  mod.def("lrelu_bwd", &lrelu_bwd);
  mod.def("ln_fwd", &ln_fwd, py::arg("x"), py::arg("gamma"), py::arg("beta"),
          py::arg("xln"), py::arg("mean"), py::arg("rstd"), py::arg("eps"),
          py::arg("xln_bf") = py::none(), py::arg("xln_f8") = py::none());
  mod.def("gemm_nt_fp8", &gemm_nt_fp8, py::arg("A"), py::arg("W"),
          py::arg("bias"), py::arg("inv_sw"), py::arg("out_f32"),
          py::arg("out_bf16"), py::arg("out_fp8"), py::arg("R"),
          py::arg("Ci"), py::arg("Co"), py::arg("alpha"),
          py::arg("act_lrelu"));
  mod.def("absmax_scale", &absmax_scale);
  mod.def("cast_f32_fp8_scaled", &cast_f32_fp8_scaled);
  mod.def("ln_bwd_params", &ln_bwd_params);
  mod.def("gru_fwd", &gru_fwd);
  mod.def("gru_fwd_mfma", &gru_fwd_mfma);
  mod.def("attn_fused_fwd", &attn_fused_fwd);
  mod.def("enc_fused_fwd", &enc_fused_fwd, py::arg("h"),
          py::arg("Wenc"), py::arg("benc"), py::arg("y"),
          py::arg("scores"), py::arg("a"), py::arg("yp"),
          py::arg("Wmu") = py::none(), py::arg("bmu") = py::none(),
          py::arg("Wsig") = py::none(), py::arg("bsig") = py::none(),
          py::arg("fmu") = py::none(),
          py::arg("fsig_pre") = py::none(),
          py::arg("fsig") = py::none(),
          py::arg("fsig_c") = py::none(),
          py::arg("done") = py::none());
  mod.def("enc_bwd_fused", &enc_bwd_fused);
  mod.def("attn_fused_bwd", &attn_fused_bwd);
  mod.def("gru_bwd_mfma", &gru_bwd_mfma, py::arg("dh_final"),
          py::arg("h_prev"), py::arg("gates4"), py::arg("whh_bf"),
          py::arg("dgi"), py::arg("dgh"), py::arg("N"), py::arg("T"),
          py::arg("H"), py::arg("dgi_bf") = py::none(),
          py::arg("dgh_bf") = py::none(), py::arg("dgi_f8") = py::none(),
          py::arg("s_dgi") = py::none(), py::arg("amax_dgi") = py::none(),
          py::arg("whh_part") = py::none(),
          py::arg("bhh_part") = py::none());
  mod.def("wgrad_reduce", &wgrad_reduce, py::arg("part"), py::arg("out"),
          py::arg("db_part"), py::arg("db"), py::arg("z"),
          py::arg("accumulate") = false);
  mod.def("gru_bwd", &gru_bwd, py::arg("dh_final"), py::arg("h_prev"),
          py::arg("gates4"), py::arg("Whh"), py::arg("dgi"), py::arg("dgh"),
          py::arg("N"), py::arg("T"), py::arg("H"),
          py::arg("dgi_bf") = py::none(), py::arg("dgh_bf") = py::none());
  mod.def("enc_softmax_fwd", &enc_softmax_fwd);
  mod.def("enc_softmax_bwd", &enc_softmax_bwd);
  mod.def("enc_heads_fwd", &enc_heads_fwd);
  mod.def("enc_heads_bwd", &enc_heads_bwd);
  mod.def("attn_qk_fwd", &attn_qk_fwd);
  mod.def("attn_softmax_fwd", &attn_softmax_fwd);
  mod.def("attn_ctx_fwd", &attn_ctx_fwd);
  mod.def("attn_head_bwd", &attn_head_bwd);
  mod.def("attn_softmax_bwd", &attn_softmax_bwd);
  mod.def("attn_qk_bwd", &attn_qk_bwd);
  mod.def("loss_fused", &loss_fused);
  mod.def("dh_combine", &dh_combine);
  mod.def("gemm_nt_bf16_rs", &gemm_nt_bf16_rs, py::arg("A"), py::arg("Wp"),
          py::arg("bias") = py::none(), py::arg("out_f32") = py::none(),
          py::arg("out_bf16") = py::none(),
          py::arg("lrelu_bwd_of") = py::none(), py::arg("alpha") = 1.0,
          py::arg("act_lrelu") = false);
  mod.def("cast_shadows", &cast_shadows);
  mod.def("gemm_nt_fp8_rs", &gemm_nt_fp8_rs, py::arg("A"), py::arg("Wp"),
          py::arg("bias") = py::none(), py::arg("inv_sw") = py::none(),
          py::arg("out_f32") = py::none(), py::arg("out_bf16") = py::none(),
          py::arg("out_fp8") = py::none(), py::arg("R"), py::arg("Ci"),
          py::arg("Co"), py::arg("alpha") = 1.0,
          py::arg("act_lrelu") = false, py::arg("inv_sa") = py::none(),
          py::arg("lrelu_bwd_of") = py::none(),
          py::arg("s_out") = py::none(), py::arg("amax_out") = py::none());
  mod.def("scale_from_amax2", &scale_from_amax2);
  mod.def("cast_f32_fp8_damax", &cast_f32_fp8_damax);
  mod.def("cast_f32_fp8_scaled_t", &cast_f32_fp8_scaled_t);
  mod.def("pred_mlp_fwd", &pred_mlp_fwd);
  mod.def("pred_mlp_bwd", &pred_mlp_bwd);
  mod.def("dec_fwd", &dec_fwd);
  mod.def("dec_bwd", &dec_bwd);
  mod.def("loss_fwd", &loss_fwd);
  mod.def("loss_bwd", &loss_bwd);
  mod.def("step_inc", &step_inc);
  mod.def("adam", &adam);
}
