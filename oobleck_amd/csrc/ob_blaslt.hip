// ob_blaslt.hip — hipBLASLt path for the PLAIN GEMMs of the hot path.
//
// The hand-written MFMA kernels (ob_kernels_bf16.hip) carry every FUSED
// op (bias/residual epilogues, flash attention, LN, CE); the plain
// library GEMMs — dX (activation grads), the bias-free lm_head family,
// and the TN weight-grad GEMMs (beta=1 accumulation straight into the
// fp32 flat grads, which also deletes the transpose materialization) —
// go to AMD's own hipBLASLt, per the MI355X playbook ("hipBLASLt /
// rocBLAS only for plain library GEMMs").  Measured on the hot shapes
// (tools/blas_probe.py): fc 815 TF, dX_fc 963, lm_head 1166, dW-TN 694
// vs 436-692 for our kernels.
//
// Row-major calls are expressed through the standard column-major swap:
//   C_rm[M,N] = op(A_rm) op(B_rm)  ==>  C̄(N,M) = op'(B̄) op'(Ā)
// with X̄ = the stored bytes viewed column-major (X^T).
#include "ob_internal.h"

#include <hipblaslt/hipblaslt.h>

#include <cstdlib>
#include <map>
#include <tuple>

namespace {

struct LtPlan {
  hipblasLtMatmulDesc_t desc = nullptr;
  hipblasLtMatrixLayout_t la = nullptr, lb = nullptr, lc = nullptr;
  hipblasLtMatmulAlgo_t algo{};
  bool ok = false;
};

constexpr size_t kLtWs = 64u << 20;
// key: tA,tB (row-major semantics), M, N, K, ldc, out f32?, beta!=0?,
// bias epilogue?, A/B f32?
using Key = std::tuple<int, int, int64_t, int64_t, int64_t, int64_t, int,
                       int, int, int>;

// ONE FULL CONTEXT PER STREAM — handle + workspace + plan cache.
// Concurrent enqueue of Lt matmuls on different streams through a single
// shared handle stalls pathologically at queue depth (measured: the pp1
// fwd/bwd dual-stream bench never finishes a 16-microbatch step with a
// shared handle, and runs fine with hipBLASLt disabled) — the handle
// carries internal per-launch state that cross-stream reuse serializes
// against.  Streams are few (main + fwd/bwd + side) and long-lived.
struct LtCtx {
  hipblasLtHandle_t h = nullptr;
  void* ws = nullptr;
  std::map<Key, LtPlan> plans;
};
std::map<void*, LtCtx> g_lt_by_stream;

LtCtx* lt_ctx(void* stream) {
  auto it = g_lt_by_stream.find(stream);
  if (it != g_lt_by_stream.end()) return &it->second;
  LtCtx ctx;
  if (hipblasLtCreate(&ctx.h) != HIPBLAS_STATUS_SUCCESS) return nullptr;
  if (hipMalloc(&ctx.ws, kLtWs) != hipSuccess) return nullptr;
  return &g_lt_by_stream.emplace(stream, ctx).first->second;
}

}  // namespace

// Plain row-major GEMM: C[M,N] = alpha * op(A) op(B) + beta * C.
// A is [M,K] (or [K,M] if tA), B is [K,N] (or [N,K] if tB), bf16,
// fp32 accumulate; C bf16 (c_f32 == 0) or fp32 (c_f32 == 1).
// Returns 0 on success, -1 if no algo (caller falls back), 1 on error.
// bias (optional, fp32, length N) is applied via the BIAS epilogue —
// in the column-major swap D = C-bar (N x M), whose rows are our output
// columns, exactly the broadcast hipBLASLt defines.
// Cin: the C-INPUT operand (D = alpha*op(A)op(B) + beta*Cin, written to
// C) — hipblasLt supports C != D, which folds a residual WITHOUT the
// 12.6 MB copy the round-1 path paid per forward projection GEMM.
static int lt_matmul(int tA, int tB, int64_t M, int64_t N, int64_t K,
                     float alpha, const void* A, int64_t lda, const void* B,
                     int64_t ldb, float beta, void* C, int64_t ldc,
                     int c_f32, const void* bias, int ab_f32, void* stream,
                     const void* Cin = nullptr) {
  if (!Cin) Cin = C;
  LtCtx* ctx = lt_ctx(stream);
  if (!ctx) return ob_fail("hipblasLt per-stream context init failed");
  const Key key{tA,     tB,           M,
                N,      K,            ldc,
                c_f32,  beta != 0.f,  bias != nullptr,
                ab_f32};
  auto it = ctx->plans.find(key);
  if (it == ctx->plans.end()) {
    LtPlan p;
    // column-major swap: A-slot <- B bytes, B-slot <- A bytes
    const hipblasOperation_t opA = tB ? HIPBLAS_OP_T : HIPBLAS_OP_N;
    const hipblasOperation_t opB = tA ? HIPBLAS_OP_T : HIPBLAS_OP_N;
    // cm dims of the A-slot (B̄): stored rm [K,N] or [N,K] -> cm (N,K)^T…
    // rows/cols BEFORE op, with ld = the rm row stride:
    //   B rm [K,N] (tB=0): B̄ is (N x K) cm, ld = ldb; op N gives (N x K)?
    // We need op(Aslot) = (N x K).  tB=0: B̄ = (N x K) already -> op N.
    // tB=1: B rm [N,K] -> B̄ = (K x N) -> op T gives (N x K).  Symmetric
    // for the B-slot: op(Bslot) = (K x M) from Ā.
    const int64_t a_rows = tB ? K : N, a_cols = tB ? N : K;
    const int64_t b_rows = tA ? M : K, b_cols = tA ? K : M;
    hipblasLtMatmulDescCreate(&p.desc, HIPBLAS_COMPUTE_32F, HIP_R_32F);
    hipblasLtMatmulDescSetAttribute(p.desc, HIPBLASLT_MATMUL_DESC_TRANSA,
                                    &opA, sizeof(opA));
    hipblasLtMatmulDescSetAttribute(p.desc, HIPBLASLT_MATMUL_DESC_TRANSB,
                                    &opB, sizeof(opB));
    if (bias) {
      const hipblasLtEpilogue_t ep = HIPBLASLT_EPILOGUE_BIAS;
      hipblasLtMatmulDescSetAttribute(p.desc, HIPBLASLT_MATMUL_DESC_EPILOGUE,
                                      &ep, sizeof(ep));
      const int32_t bt = HIP_R_32F;
      hipblasLtMatmulDescSetAttribute(
          p.desc, HIPBLASLT_MATMUL_DESC_BIAS_DATA_TYPE, &bt, sizeof(bt));
      hipblasLtMatmulDescSetAttribute(
          p.desc, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &bias, sizeof(bias));
    }
    const hipDataType abt = ab_f32 ? HIP_R_32F : HIP_R_16BF;
    hipblasLtMatrixLayoutCreate(&p.la, abt, a_rows, a_cols, ldb);
    hipblasLtMatrixLayoutCreate(&p.lb, abt, b_rows, b_cols, lda);
    hipblasLtMatrixLayoutCreate(&p.lc, c_f32 ? HIP_R_32F : HIP_R_16BF, N, M,
                                ldc);
    hipblasLtMatmulPreference_t pref;
    hipblasLtMatmulPreferenceCreate(&pref);
    const uint64_t ws = kLtWs;
    hipblasLtMatmulPreferenceSetAttribute(
        pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws, sizeof(ws));
    hipblasLtMatmulHeuristicResult_t res[1];
    int nres = 0;
    hipblasLtMatmulAlgoGetHeuristic(ctx->h, p.desc, p.la, p.lb, p.lc, p.lc,
                                    pref, 1, res, &nres);
    hipblasLtMatmulPreferenceDestroy(pref);
    if (nres > 0 && res[0].state == HIPBLAS_STATUS_SUCCESS) {
      p.algo = res[0].algo;
      p.ok = true;
    }
    it = ctx->plans.emplace(key, p).first;
  }
  const LtPlan& p = it->second;
  if (!p.ok) return -1;
  if (bias)
    hipblasLtMatmulDescSetAttribute(
        p.desc, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &bias, sizeof(bias));
  const hipblasStatus_t st = hipblasLtMatmul(
      ctx->h, p.desc, &alpha, B, p.la, A, p.lb, &beta, Cin, p.lc, C, p.lc,
      &p.algo, ctx->ws, kLtWs, reinterpret_cast<hipStream_t>(stream));
  if (st != HIPBLAS_STATUS_SUCCESS)
    return ob_fail("hipblasLtMatmul failed (%d)", (int)st);
  return 0;
}

extern "C" int ob_gemm_lt(int tA, int tB, int64_t M, int64_t N, int64_t K,
                          float alpha, const void* A, int64_t lda,
                          const void* B, int64_t ldb, float beta, void* C,
                          int64_t ldc, int c_f32, void* stream) {
  return lt_matmul(tA, tB, M, N, K, alpha, A, lda, B, ldb, beta, C, ldc,
                   c_f32, nullptr, 0, stream);
}

extern "C" int ob_gemm_lt_bias(int tA, int tB, int64_t M, int64_t N,
                               int64_t K, float alpha, const void* A,
                               int64_t lda, const void* B, int64_t ldb,
                               float beta, void* C, int64_t ldc, int c_f32,
                               const void* bias, void* stream) {
  return lt_matmul(tA, tB, M, N, K, alpha, A, lda, B, ldb, beta, C, ldc,
                   c_f32, bias, 0, stream);
}

/* bias epilogue + residual folded as the C-input (no copy) */
extern "C" int ob_gemm_lt_bias_res(int tA, int tB, int64_t M, int64_t N,
                                   int64_t K, float alpha, const void* A,
                                   int64_t lda, const void* B, int64_t ldb,
                                   void* C, int64_t ldc, int c_f32,
                                   const void* bias, const void* residual,
                                   void* stream) {
  return lt_matmul(tA, tB, M, N, K, alpha, A, lda, B, ldb, 1.f, C, ldc,
                   c_f32, bias, 0, stream, residual);
}

// fp32 operands, fp32 accumulate/out (the reference-dtype leg)
extern "C" int ob_gemm_lt_f32(int tA, int tB, int64_t M, int64_t N,
                              int64_t K, float alpha, const void* A,
                              int64_t lda, const void* B, int64_t ldb,
                              float beta, void* C, int64_t ldc,
                              void* stream) {
  return lt_matmul(tA, tB, M, N, K, alpha, A, lda, B, ldb, beta, C, ldc, 1,
                   nullptr, 1, stream);
}
