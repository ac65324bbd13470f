// Internal shared declarations between ob_kernels.hip and ob_layer.hip.
#pragma once
#include <hip/hip_runtime.h>

#include <cstdarg>
#include <cstdint>
#include <cstdio>

#include "../../include/oobleck_stage.h"

// thread-local last-error buffer (single driving thread per GPU by contract,
// but thread-local keeps the reconfig listener thread safe too).
int ob_fail(const char* fmt, ...);

#define OB_HIP(x)                                                     \
  do {                                                                \
    hipError_t e_ = (x);                                              \
    if (e_ != hipSuccess)                                             \
      return ob_fail("%s:%d: %s", __FILE__, __LINE__,                 \
                     hipGetErrorString(e_));                          \
  } while (0)

#define OB_LAUNCH_CHECK()                                             \
  do {                                                                \
    hipError_t e_ = hipGetLastError();                                \
    if (e_ != hipSuccess)                                             \
      return ob_fail("%s:%d: launch: %s", __FILE__, __LINE__,         \
                     hipGetErrorString(e_));                          \
  } while (0)

// ---------------------------------------------------------------------------
// In-step profiler (state in ob_layer.hip).  When enabled (ob_profile_enable,
// public ABI), every wrapped launch region is bracketed by HIP events ON THE
// STREAM IT LAUNCHES ON and accumulated per family, so bench.py can report
// the production dispatch's in-step per-launch times (roofline) and a
// per-family step-time split.  Off by default: zero overhead in the timed
// region (one branch per call).
enum {
  OB_PF_FC_FWD = 0,     // the roofline kernel: MLP fc forward GEMM
  OB_PF_GEMM_FWD = 1,   // other forward GEMMs (qkv/attnproj/mlpproj/lm_head)
  OB_PF_GEMM_DX = 2,    // backward activation-grad GEMMs
  OB_PF_GEMM_DW = 3,    // weight-grad GEMMs (side stream)
  OB_PF_FLASH_FWD = 4,  // flash fwd + its V^T staging
  OB_PF_FLASH_BWD = 5,  // flash bwd (transposes + dsum + dkdv/dq)
  OB_PF_ATTN_MAT = 6,   // non-flash attention matmuls + softmax
  OB_PF_LN = 7,
  OB_PF_CE = 8,
  OB_PF_ELEM = 9,       // gelu / colsum / embed / misc elementwise
  OB_PF_ADAMW = 10,
  OB_PF_NFAM = 11
};
bool ob_prof_on();
int ob_prof_beg(int fam, hipStream_t s);
void ob_prof_end(int fam, int slot, hipStream_t s);

#define OB_PROF(fid, strm, call)                                        \
  ({                                                                    \
    int _r;                                                             \
    if (ob_prof_on()) {                                                 \
      hipStream_t _ps = reinterpret_cast<hipStream_t>(strm);            \
      const int _fi = (fid);                                            \
      const int _slot = ob_prof_beg(_fi, _ps);                          \
      _r = (call);                                                      \
      ob_prof_end(_fi, _slot, _ps);                                     \
    } else {                                                            \
      _r = (call);                                                      \
    }                                                                   \
    _r;                                                                 \
  })

// internal launchers used by the layer orchestration (same semantics as the
// public ob_* wrappers but C++ linkage, no error wrapping duplication).
int ob_embed_fwd_f32(const int64_t* ids, const float* wte, const float* wpe,
                     float* out, int64_t B, int64_t S, int64_t H, void* stream);
int ob_embed_bwd_f32(const int64_t* ids, const float* dout, float* dwte,
                     float* dwpe, int64_t B, int64_t S, int64_t H, void* stream);
int ob_ce_fwd_f32(const float* logits, const int64_t* labels, float* lse,
                  float* loss, int64_t B, int64_t S, int64_t V, void* stream);
int ob_ce_bwd_f32(float* logits_to_dlogits, const int64_t* labels,
                  const float* lse, const float* dloss_or_null, int64_t B,
                  int64_t S, int64_t V, void* stream);

// bf16-path launchers (defined in ob_kernels_bf16.hip)
extern "C" {
int ob_f32_to_bf16_t_ld(const void* x, void* y, int64_t rows, int64_t cols,
                        int64_t out_ld, void* stream);
int ob_layernorm_fwd_bf16(const void* x, const void* w, const void* b,
                          void* y, void* mean, void* rstd, int64_t rows,
                          int64_t H, float eps, void* stream);
int ob_layernorm_bwd_bf16(const void* x, const void* w, const void* mean,
                          const void* rstd, const void* dy, void* dx,
                          void* dw, void* db, int64_t rows, int64_t H,
                          int dx_accum, void* stream);
int ob_softmax_causal_fwd_bf16(void* scores, int64_t batch, int64_t Sq,
                               float scale, void* stream);
int ob_softmax_causal_bwd_bf16(const void* P, void* dP, int64_t batch,
                               int64_t Sq, void* stream);
int ob_gelu_fwd_bf16(const void* u, void* g, int64_t n, void* stream);
int ob_gelu_bwd_bf16(const void* u, const void* dg, void* du, int64_t n,
                     void* stream);
int ob_colsum_bf16(const void* X, void* db, int64_t M, int64_t N,
                   void* stream);
int ob_embed_fwd_bf16(const void* ids, const void* wte, const void* wpe,
                      void* out, int64_t B, int64_t Sq, int64_t H,
                      void* stream);
int ob_embed_bwd_bf16(const void* ids, const void* dout, void* dwte,
                      void* dwpe, int64_t B, int64_t Sq, int64_t H,
                      void* stream);
int ob_ce_fwd_bf16(const void* logits, const void* labels, void* lse,
                   void* loss, int64_t B, int64_t Sq, int64_t V, int64_t ld,
                   void* stream);
int ob_ce_bwd_bf16(void* logits, const void* labels, const void* lse,
                   const void* dloss, int64_t B, int64_t Sq, int64_t V,
                   int64_t ld, void* stream);
int ob_transpose_bf16(const void* in, void* out, int64_t R, int64_t C,
                      void* stream);
int ob_flash_fwd_bf16(const void* qkv, const void* VT, void* O, void* lse,
                      int64_t B, int64_t Sq, int64_t H, int64_t nh,
                      float scale, void* stream);
int ob_transpose_bf16_b(const void* in, void* out, int64_t R, int64_t C,
                        int64_t sIn1, int64_t sIn2, int64_t ldin, int64_t n1,
                        int64_t n2, void* stream);
int ob_flash_dsum_bf16(const void* O, const void* dO, void* D, int64_t B,
                       int64_t Sq, int64_t H, int64_t nh, void* stream);
int ob_flash_bwd_bf16(const void* qkv, const void* QT, const void* KT,
                      const void* dOT, const void* dO, const void* lse,
                      const void* D, void* dqkv, int64_t B, int64_t Sq,
                      int64_t H, int64_t nh, float scale, void* stream);
}
// hipBLASLt path for plain GEMMs (ob_blaslt.hip); returns -1 when the
// heuristic offers no algo (caller falls back to the hand-written path)
extern "C" int ob_gemm_lt(int tA, int tB, int64_t M, int64_t N, int64_t K,
                          float alpha, const void* A, int64_t lda,
                          const void* B, int64_t ldb, float beta, void* C,
                          int64_t ldc, int c_f32, void* stream);
extern "C" int ob_gemm_lt_bias(int tA, int tB, int64_t M, int64_t N,
                               int64_t K, float alpha, const void* A,
                               int64_t lda, const void* B, int64_t ldb,
                               float beta, void* C, int64_t ldc, int c_f32,
                               const void* bias, void* stream);
extern "C" int ob_gemm_lt_bias_res(int tA, int tB, int64_t M, int64_t N,
                                   int64_t K, float alpha, const void* A,
                                   int64_t lda, const void* B, int64_t ldb,
                                   void* C, int64_t ldc, int c_f32,
                                   const void* bias, const void* residual,
                                   void* stream);
extern "C" int ob_gemm_lt_f32(int tA, int tB, int64_t M, int64_t N,
                              int64_t K, float alpha, const void* A,
                              int64_t lda, const void* B, int64_t ldb,
                              float beta, void* C, int64_t ldc,
                              void* stream);
extern "C" int ob_gemm_bf16_nt_8ph(const void* A, const void* B, void* C,
                                   const void* bias, const void* residual,
                                   int64_t M, int64_t N, int64_t K,
                                   int64_t lda, int64_t ldb, int64_t ldc,
                                   int64_t sA1, int64_t sA2, int64_t sB1,
                                   int64_t sB2, int64_t sC1, int64_t sC2,
                                   int64_t n1, int64_t n2, float alpha,
                                   float beta, int out_kind, int splitk,
                                   void* stream, int64_t Mr);
// fast NT glds dispatch (interior 128-tiled M/N; Mr = store-row guard)
int ob_gemm_bf16_nt_dispatch(const void* A, const void* B, void* C,
                             const void* bias, const void* residual,
                             int64_t M, int64_t N, int64_t K, int64_t lda,
                             int64_t ldb, int64_t ldc, int64_t sA1,
                             int64_t sA2, int64_t sB1, int64_t sB2,
                             int64_t sC1, int64_t sC2, int64_t n1, int64_t n2,
                             float alpha, float beta, int out_kind, int splitk,
                             void* stream, int64_t Mr);
