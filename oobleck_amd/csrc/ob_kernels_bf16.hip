// ob_kernels_bf16.hip — bf16 MFMA GEMM for the mixed-precision path
// (§8 f4: bf16 storage + fp32 master weights; the reference never
// implemented mixed precision, README.md:98).
//
// v_mfma_f32_32x32x16_bf16: fp32 accumulate, dense peak ~2.5 PF/s
// (16x the f32 matrix rate).  Fragment layout (derived from the CDNA
// pattern the guides document — f32 ops use k = lane-group index, the
// bf16 K-deep ops give each lane a CONTIGUOUS k-run of K/2 elements;
// verified empirically by tests/test_gpu_bf16.py against torch matmul):
//   A: lane l holds A[i = l&31][k = 8*(l>>5) + j], j = 0..7  (one 16-byte
//      ds_read_b128 when the LDS image is [row][k] with k contiguous)
//   B: lane l holds B[k = 8*(l>>5) + j][n = l&31]
//   C/D: col = l&31, row = (r&3) + 8*(r>>2) + 4*(l>>5)  (dtype-independent)
//
// LDS images are [row][k] (row = M-row for A, N-col for B) with the row
// stride padded to 40 halfs (80 B) so the 16-lane ds_read_b128 groups hit
// 16 distinct bank quads (bank = (20*row + 4*khalf) % 64 covers 0..60
// step 4 across a group).  Operands whose k is contiguous in memory
// (activations [M,K]; weight shadows kept in BOTH [in,out] and [out,in]
// layouts by the layer) stage directly; k-strided operands (the TN weight
// -grad case) transpose-stage through scattered 2-byte LDS writes.
#include "ob_internal.h"

#include <cmath>

using f32x16 = __attribute__((ext_vector_type(16))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;

static inline hipStream_t S(void* s) { return reinterpret_cast<hipStream_t>(s); }

#define BF_BM 128
#define BF_BN 128
#define BF_BK 32
#define BF_LDS_K 40  // padded row stride in halfs (80 B)

// out-kind for the epilogue
enum { BF_OUT_BF16 = 0, BF_OUT_F32 = 1, BF_OUT_F32_ATOMIC = 2 };

__device__ __forceinline__ float bf2f(__bf16 h) { return (float)h; }

__device__ __forceinline__ unsigned bf_bits(__bf16 h) {
  return (unsigned)__builtin_bit_cast(unsigned short, h);
}
__device__ __forceinline__ __bf16 bits_bf(unsigned u) {
  return __builtin_bit_cast(__bf16, (unsigned short)(u & 0xffffu));
}

// pack 8 guarded scalar loads into a uint4 without an address-taken local
template <bool EDGE>
__device__ __forceinline__ uint4 bf_load8(const __bf16* p, int valid) {
  uint4 v = {0, 0, 0, 0};
  unsigned e[8];
#pragma unroll
  for (int j = 0; j < 8; ++j)
    e[j] = (!EDGE || j < valid) ? bf_bits(p[j]) : 0u;
  v.x = e[0] | (e[1] << 16);
  v.y = e[2] | (e[3] << 16);
  v.z = e[4] | (e[5] << 16);
  v.w = e[6] | (e[7] << 16);
  return v;
}

__device__ __forceinline__ __bf16 bf_extract(const uint4& v, int j) {
  // j is a compile-time constant at every call site (unrolled loops)
  const unsigned word = (j < 2) ? v.x : (j < 4) ? v.y : (j < 6) ? v.z : v.w;
  return bits_bf((j & 1) ? (word >> 16) : word);
}

__device__ __forceinline__ uint4 bf_pack8(const float* v) {
  uint4 o;
  o.x = bf_bits((__bf16)v[0]) | (bf_bits((__bf16)v[1]) << 16);
  o.y = bf_bits((__bf16)v[2]) | (bf_bits((__bf16)v[3]) << 16);
  o.z = bf_bits((__bf16)v[4]) | (bf_bits((__bf16)v[5]) << 16);
  o.w = bf_bits((__bf16)v[6]) | (bf_bits((__bf16)v[7]) << 16);
  return o;
}

// direct staging: img[row][k] = src[row][k], k contiguous in memory.
// 256 threads; NR = tile rows (64 or 128).
template <bool EDGE, int NR>
__device__ __forceinline__ void bf_stage_direct_n(__bf16* img,
                                                  const __bf16* src,
                                                  int64_t ld, int rmax,
                                                  int kmax) {
  constexpr int TPR = 1024 / (NR * 4);  // k-chunks per thread pass
  const int r = threadIdx.x / (256 / NR);
  const int kb0 = (threadIdx.x % (256 / NR)) * 8;
#pragma unroll
  for (int it = 0; it < (NR * 4) / 256; ++it) {
    const int kb = kb0 + it * (256 / NR) * 8;
    uint4 v = {0, 0, 0, 0};
    if (!EDGE || (r < rmax && kb + 7 < kmax)) {
      v = *reinterpret_cast<const uint4*>(src + (int64_t)r * ld + kb);
    } else if (r < rmax && kb < kmax) {
      v = bf_load8<EDGE>(src + (int64_t)r * ld + kb, kmax - kb);
    }
    *reinterpret_cast<uint4*>(img + r * BF_LDS_K + kb) = v;
  }
  (void)sizeof(char[TPR >= 0 ? 1 : -1]);
}

template <bool EDGE>
__device__ __forceinline__ void bf_stage_direct(__bf16* img,
                                                const __bf16* src, int64_t ld,
                                                int rmax, int kmax) {
  bf_stage_direct_n<EDGE, 128>(img, src, ld, rmax, kmax);
}

// transpose staging: img[row][k] = src[k][row] (src k-major, ld = src row
// stride).  Reads 16 B along the row dim (coalesced), writes 8 scattered
// 2-byte LDS stores.  NR = tile rows.
template <bool EDGE, int NR>
__device__ __forceinline__ void bf_stage_transpose_n(__bf16* img,
                                                     const __bf16* src,
                                                     int64_t ld, int rmax,
                                                     int kmax) {
  constexpr int RGROUPS = NR / 8;  // 8-row chunks
  const int rb = (threadIdx.x % RGROUPS) * 8;
  const int k0 = threadIdx.x / RGROUPS;
#pragma unroll
  for (int it = 0; it < (RGROUPS * 32) / 256; ++it) {
    const int k = k0 + it * (256 / RGROUPS);
    uint4 v = {0, 0, 0, 0};
    bool any = true;
    if (!EDGE || (k < kmax && rb + 7 < rmax)) {
      v = *reinterpret_cast<const uint4*>(src + (int64_t)k * ld + rb);
    } else if (k < kmax && rb < rmax) {
      v = bf_load8<EDGE>(src + (int64_t)k * ld + rb, rmax - rb);
    } else {
      any = (k < BF_BK);  // still must zero-fill the image
    }
    if (any) {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        img[(rb + j) * BF_LDS_K + k] = bf_extract(v, j);
    }
  }
}

template <bool EDGE>
__device__ __forceinline__ void bf_stage_transpose(__bf16* img,
                                                   const __bf16* src,
                                                   int64_t ld, int rmax,
                                                   int kmax) {
  bf_stage_transpose_n<EDGE, 128>(img, src, ld, rmax, kmax);
}

template <bool TA, bool TB, int OUT, bool EDGE, int BN>
__global__ __launch_bounds__(256, 2) void k_gemm_bf16(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ Cv, const float* __restrict__ bias,
    const __bf16* __restrict__ R, int M, int N, int K, int64_t lda,
    int64_t ldb, int64_t ldc, int64_t sA1, int64_t sA2, int64_t sB1,
    int64_t sB2, int64_t sC1, int64_t sC2, int n2, float alpha, float beta,
    int nbn) {
  constexpr int FN = BN / 64;  // B fragments per wave (1 or 2)
  __shared__ __bf16 As[2][BF_BM * BF_LDS_K];
  __shared__ __bf16 Bs[2][BN * BF_LDS_K];

  const int tile = blockIdx.x;
  const int bm = tile / nbn, bn = tile % nbn;
  const int m0 = bm * BF_BM, n0 = bn * BN;

  const int z = blockIdx.z;
  const int i1 = z / n2, i2 = z % n2;
  A += (int64_t)i1 * sA1 + (int64_t)i2 * sA2;
  B += (int64_t)i1 * sB1 + (int64_t)i2 * sB2;
  float* Cf = reinterpret_cast<float*>(Cv);
  __bf16* Cb = reinterpret_cast<__bf16*>(Cv);
  const int64_t coff = (int64_t)i1 * sC1 + (int64_t)i2 * sC2;
  Cf += coff;
  Cb += coff;
  if (R) R += coff;

  const int splitk = gridDim.y;
  const int kchunk = ((K + splitk * BF_BK - 1) / (splitk * BF_BK)) * BF_BK;
  const int kbeg = blockIdx.y * kchunk;
  const int kend = min(K, kbeg + kchunk);
  if (kbeg >= kend) return;

  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const int wr = w >> 1, wc = w & 1;
  const int il = lane & 31, kh = lane >> 5;

#define OB_BF_STAGE(BUF, KT)                                                  \
  {                                                                           \
    const int kmax_ = min(kend - (KT), BF_BK);                                \
    if (TA)                                                                   \
      bf_stage_transpose<EDGE>(As[BUF], A + (int64_t)(KT)*lda + m0, lda,      \
                               min(M - m0, BF_BM), kmax_);                    \
    else                                                                      \
      bf_stage_direct<EDGE>(As[BUF], A + (int64_t)m0 * lda + (KT), lda,       \
                            min(M - m0, BF_BM), kmax_);                       \
    if (TB)                                                                   \
      bf_stage_direct_n<EDGE, BN>(Bs[BUF], B + (int64_t)n0 * ldb + (KT), ldb, \
                                  min(N - n0, BN), kmax_);                    \
    else                                                                      \
      bf_stage_transpose_n<EDGE, BN>(Bs[BUF], B + (int64_t)(KT)*ldb + n0,     \
                                     ldb, min(N - n0, BN), kmax_);            \
  }

  f32x16 acc00 = {}, acc01 = {}, acc10 = {}, acc11 = {};

#define OB_BF_MFMA(BUF)                                                       \
  _Pragma("unroll") for (int ks = 0; ks < BF_BK / 16; ++ks) {                 \
    const int kb = ks * 16 + kh * 8;                                          \
    const bf16x8 a0 = *reinterpret_cast<const bf16x8*>(                       \
        As[BUF] + (wr * 64 + il) * BF_LDS_K + kb);                            \
    const bf16x8 a1 = *reinterpret_cast<const bf16x8*>(                       \
        As[BUF] + (wr * 64 + 32 + il) * BF_LDS_K + kb);                       \
    const bf16x8 b0 = *reinterpret_cast<const bf16x8*>(                       \
        Bs[BUF] + (wc * (BN / 2) + il) * BF_LDS_K + kb);                      \
    acc00 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b0, acc00, 0, 0, 0);  \
    acc10 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b0, acc10, 0, 0, 0);  \
    if (FN == 2) {                                                            \
      const bf16x8 b1 = *reinterpret_cast<const bf16x8*>(                     \
          Bs[BUF] + (wc * (BN / 2) + 32 + il) * BF_LDS_K + kb);               \
      acc01 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b1, acc01, 0, 0, 0);\
      acc11 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b1, acc11, 0, 0, 0);\
    }                                                                         \
  }                                                                      \
  __builtin_amdgcn_s_setprio(0);

  OB_BF_STAGE(0, kbeg)
  __syncthreads();
  int cur = 0;
  for (int kt = kbeg; kt + BF_BK < kend; kt += BF_BK) {
    OB_BF_STAGE(cur ^ 1, kt + BF_BK)
    OB_BF_MFMA(cur)
    __syncthreads();
    cur ^= 1;
  }
  OB_BF_MFMA(cur)
#undef OB_BF_STAGE
#undef OB_BF_MFMA

  const int mw = m0 + wr * 64, nw = n0 + wc * (BN / 2);
#define OB_BF_EPI(ACC, TI, TJ)                                                \
  {                                                                           \
    const int nn = nw + (TJ)*32 + il;                                         \
    if (!EDGE || nn < N) {                                                    \
      _Pragma("unroll") for (int r = 0; r < 16; ++r) {                        \
        const int mm = mw + (TI)*32 + (r & 3) + 8 * (r >> 2) + 4 * kh;        \
        if (!EDGE || mm < M) {                                                \
          float v = alpha * ACC[r];                                           \
          if (OUT == BF_OUT_F32_ATOMIC) {                                     \
            atomicAdd(&Cf[(int64_t)mm * ldc + nn], v);                        \
          } else {                                                            \
            if (bias) v += bias[nn];                                          \
            if (R) v += bf2f(R[(int64_t)mm * ldc + nn]);                      \
            if (OUT == BF_OUT_F32) {                                          \
              if (beta != 0.f) v += beta * Cf[(int64_t)mm * ldc + nn];        \
              Cf[(int64_t)mm * ldc + nn] = v;                                 \
            } else {                                                          \
              if (beta != 0.f) v += beta * bf2f(Cb[(int64_t)mm * ldc + nn]);  \
              Cb[(int64_t)mm * ldc + nn] = (__bf16)v;                         \
            }                                                                 \
          }                                                                   \
        }                                                                     \
      }                                                                       \
    }                                                                         \
  }
  OB_BF_EPI(acc00, 0, 0)
  OB_BF_EPI(acc10, 1, 0)
  if (FN == 2) {
    OB_BF_EPI(acc01, 0, 1)
    OB_BF_EPI(acc11, 1, 1)
  }
#undef OB_BF_EPI
}

extern "C" int ob_gemm_bf16(int transA, int transB, int64_t M, int64_t N,
                            int64_t K, float alpha, const void* A, int64_t lda,
                            int64_t strideA1, int64_t strideA2, const void* B,
                            int64_t ldb, int64_t strideB1, int64_t strideB2,
                            float beta, void* C, int64_t ldc, int64_t strideC1,
                            int64_t strideC2, int64_t n1, int64_t n2,
                            const void* bias, const void* residual,
                            int out_kind, int splitk, void* stream) {
  if (M <= 0 || N <= 0 || K <= 0) return ob_fail("gemm_bf16: bad dims");
  if (splitk < 1) splitk = 1;
  if (splitk > 1 && out_kind != BF_OUT_F32_ATOMIC)
    return ob_fail("gemm_bf16: splitk needs atomic f32 out");
  // 16-byte staging requires 8-half-aligned leading dims and bases
  if ((lda | ldb) & 7) return ob_fail("gemm_bf16: lda/ldb must be 8-aligned");
  // probe hook (tools/fc_probe.py): OB_BF16_FORCE picks one NT kernel
  // unconditionally ("glds" | "n128" | "n256" | "8ph"); read per call so
  // one process can sweep variants.
  const char* force = getenv("OB_BF16_FORCE");
  if (force && transA == 0 && transB == 1) {
    const int64_t nbm = (M + 255) / 256;
    if (force[0] == 'g')
      return ob_gemm_bf16_nt_dispatch(A, B, C, bias, residual, M, N, K, lda,
                                      ldb, ldc, strideA1, strideA2, strideB1,
                                      strideB2, strideC1, strideC2, n1, n2,
                                      alpha, beta, out_kind, splitk, stream,
                                      M);
    if (force[0] == 'n') {
      extern int ob_gemm_bf16_nt256_dispatch(
          const void* A, const void* B, void* C, const void* bias,
          const void* residual, int64_t M, int64_t N, int64_t K, int64_t lda,
          int64_t ldb, int64_t ldc, int64_t sA1, int64_t sA2, int64_t sB1,
          int64_t sB2, int64_t sC1, int64_t sC2, int64_t n1, int64_t n2,
          float alpha, float beta, int out_kind, int splitk, void* stream,
          int64_t Mr, int BN);
      return ob_gemm_bf16_nt256_dispatch(
          A, B, C, bias, residual, M, N, K, lda, ldb, ldc, strideA1,
          strideA2, strideB1, strideB2, strideC1, strideC2, n1, n2, alpha,
          beta, out_kind, splitk, stream, M, force[1] == '2' ? 256 : 128);
    }
    if (force[0] == '8') {
      extern int ob_gemm_bf16_nt_8ph(
          const void* A, const void* B, void* C, const void* bias,
          const void* residual, int64_t M, int64_t N, int64_t K, int64_t lda,
          int64_t ldb, int64_t ldc, int64_t sA1, int64_t sA2, int64_t sB1,
          int64_t sB2, int64_t sC1, int64_t sC2, int64_t n1, int64_t n2,
          float alpha, float beta, int out_kind, int splitk, void* stream,
          int64_t Mr);
      return ob_gemm_bf16_nt_8ph(A, B, C, bias, residual, M, N, K, lda, ldb,
                                 ldc, strideA1, strideA2, strideB1, strideB2,
                                 strideC1, strideC2, n1, n2, alpha, beta,
                                 out_kind, splitk, stream, M);
    }
  }
  // plain GEMMs (no bias/residual/beta, unbatched, bf16 out) go to
  // hipBLASLt — the dX family, the bias-free lm_head family (measured
  // 815-1166 TF vs 436-692 for the hand-written kernels on those
  // shapes, tools/blas_probe.py).  Fused epilogues stay on ours.
  static const bool no_lt = [] {
    const char* e = getenv("OB_NO_BLASLT");
    return e && e[0] == '1';
  }();
  if (!no_lt && beta == 0.f && n1 == 1 && n2 == 1 && splitk == 1 &&
      out_kind == BF_OUT_BF16 && M * N >= 512 * 512) {
    // bias rides the BIAS epilogue; a residual is folded as the C-INPUT
    // operand of D = alpha*AB + beta*C (C != D) — no copy (the round-1
    // path paid a 12.6 MB DtoD per forward projection GEMM)
    int r;
    if (residual) {
      r = ob_gemm_lt_bias_res(transA, transB, M, N, K, alpha, A, lda, B,
                              ldb, C, ldc, 0, bias, residual, stream);
    } else {
      r = ob_gemm_lt_bias(transA, transB, M, N, K, alpha, A, lda, B, ldb,
                          0.f, C, ldc, 0, bias, stream);
    }
    if (r >= 0) return r;  // -1: no algo, fall through to our kernels
  }
  const int BN = (N <= 64) ? 64 : 128;  // narrow tiles for head_dim GEMMs
  const int nbm = (int)((M + BF_BM - 1) / BF_BM);
  const int nbn = (int)((N + BN - 1) / BN);
  const bool edge = (M % BF_BM) || (N % BN) || (K % BF_BK);
  dim3 grid(nbm * nbn, splitk, (unsigned)(n1 * n2));
  dim3 block(256);
#define OB_BFG_L2(TA_, TB_, OUT_, ED_, BN_)                                  \
  k_gemm_bf16<TA_, TB_, OUT_, ED_, BN_><<<grid, block, 0, S(stream)>>>(      \
      (const __bf16*)A, (const __bf16*)B, C, (const float*)bias,             \
      (const __bf16*)residual, (int)M, (int)N, (int)K, lda, ldb, ldc,        \
      strideA1, strideA2, strideB1, strideB2, strideC1, strideC2, (int)n2,   \
      alpha, beta, nbn)
#define OB_BFG_L(TA_, TB_, OUT_, ED_)                                        \
  do {                                                                       \
    if (BN == 64) OB_BFG_L2(TA_, TB_, OUT_, ED_, 64);                        \
    else OB_BFG_L2(TA_, TB_, OUT_, ED_, 128);                                \
  } while (0)
#define OB_BFG_OUT(TA_, TB_)                                                 \
  do {                                                                       \
    if (out_kind == BF_OUT_BF16) {                                           \
      if (edge) OB_BFG_L(TA_, TB_, BF_OUT_BF16, true);                       \
      else OB_BFG_L(TA_, TB_, BF_OUT_BF16, false);                           \
    } else if (out_kind == BF_OUT_F32) {                                     \
      if (edge) OB_BFG_L(TA_, TB_, BF_OUT_F32, true);                        \
      else OB_BFG_L(TA_, TB_, BF_OUT_F32, false);                            \
    } else {                                                                 \
      if (edge) OB_BFG_L(TA_, TB_, BF_OUT_F32_ATOMIC, true);                 \
      else OB_BFG_L(TA_, TB_, BF_OUT_F32_ATOMIC, false);                     \
    }                                                                        \
  } while (0)
  // fast NT path: async glds staging (ob_internal.h declares the
  // dispatch; interior 128-aligned shapes with 16-byte-aligned
  // bases/strides only)
  static const bool no_glds = [] {
    const char* e = getenv("OB_BF16_NOGLDS");
    return e && e[0] == '1';
  }();
  const bool aligned16 =
      ((reinterpret_cast<uintptr_t>(A) | reinterpret_cast<uintptr_t>(B)) %
           16 == 0) &&
      ((strideA1 | strideA2 | strideB1 | strideB2) % 8 == 0);
  if (!no_glds && transA == 0 && transB == 1 && !edge && aligned16 &&
      K >= 2 * BF_BK) {
    extern int ob_gemm_bf16_nt256_dispatch(
        const void* A, const void* B, void* C, const void* bias,
        const void* residual, int64_t M, int64_t N, int64_t K, int64_t lda,
        int64_t ldb, int64_t ldc, int64_t sA1, int64_t sA2, int64_t sB1,
        int64_t sB2, int64_t sC1, int64_t sC2, int64_t n1, int64_t n2,
        float alpha, float beta, int out_kind, int splitk, void* stream,
        int64_t Mr, int BN);
    // 8-phase 256^2 kernel: wins when its grid fills whole scheduling
    // rounds (measured: lmhead 692 vs 618 TF, 4096^3 1058 vs 995; loses
    // on partial rounds — fc 1.5 rounds: 529 vs 554).  tiles >= 192 and
    // round-efficiency >= 0.9 required.  For atomic outputs (weight
    // grads accumulate, so any K split is valid) the split is CHOSEN to
    // fill rounds rather than taken from the caller.
    if (M % 256 == 0 && N % 256 == 0 && K % 64 == 0) {
      extern int ob_gemm_bf16_nt_8ph(
          const void* A, const void* B, void* C, const void* bias,
          const void* residual, int64_t M, int64_t N, int64_t K, int64_t lda,
          int64_t ldb, int64_t ldc, int64_t sA1, int64_t sA2, int64_t sB1,
          int64_t sB2, int64_t sC1, int64_t sC2, int64_t n1, int64_t n2,
          float alpha, float beta, int out_kind, int splitk, void* stream,
          int64_t Mr);
      // NOTE: an auto-splitk search for atomic outputs was tried here
      // (round-filling sk up to 16) and REGRESSED the step 360k->347k
      // tok/s: split-k multiplies the f32 atomic output traffic by sk,
      // which dominates on the (W1, W2, BS) weight-grad shapes.
      const int sk = splitk < 1 ? 1 : splitk;
      const int64_t tiles = (M / 256) * (N / 256) * n1 * n2 * sk;
      const double eff = (double)tiles / (((tiles + 255) / 256) * 256);
      if (tiles >= 192 && eff >= 0.9)
        return ob_gemm_bf16_nt_8ph(A, B, C, bias, residual, M, N, K, lda,
                                   ldb, ldc, strideA1, strideA2, strideB1,
                                   strideB2, strideC1, strideC2, n1, n2,
                                   alpha, beta, out_kind, sk, stream, M);
    }
    static const bool no256 = [] {
      const char* e = getenv("OB_BF16_NO256");
      return e && e[0] == '1';
    }();
    if (!no256 && M % 256 == 0 && K % 64 == 0 && K >= 1024) {
      // pick BN by scheduling-round efficiency (blocks / ceil-to-256):
      // e.g. fc (M=8192,N=3072): BN=256 -> 384 blocks = 1.5 rounds (75%),
      // BN=128 -> 768 blocks = 3 full rounds (100%)
      int bn2 = 0;
      double best_eff = 0.0;
      const int64_t zb = n1 * n2 * (splitk < 1 ? 1 : splitk);
      for (int cand : {256, 128}) {
        if (N % cand) continue;
        const int64_t blocks = (M / 256) * (N / cand) * zb;
        const double eff =
            (double)blocks / (((blocks + 255) / 256) * 256);
        if (eff > best_eff + 1e-9) {
          best_eff = eff;
          bn2 = cand;
        }
      }
      if (bn2) {
        const int64_t tiles = (M / 256) * (N / bn2) * zb;
        if (tiles >= 192)
          return ob_gemm_bf16_nt256_dispatch(
              A, B, C, bias, residual, M, N, K, lda, ldb, ldc, strideA1,
              strideA2, strideB1, strideB2, strideC1, strideC2, n1, n2, alpha,
              beta, out_kind, splitk, stream, M, bn2);
      }
    }
    return ob_gemm_bf16_nt_dispatch(A, B, C, bias, residual, M, N, K, lda,
                                    ldb, ldc, strideA1, strideA2, strideB1,
                                    strideB2, strideC1, strideC2, n1, n2,
                                    alpha, beta, out_kind, splitk, stream, M);
  }
  const int sel = (transA ? 2 : 0) | (transB ? 1 : 0);
  switch (sel) {
    case 0: OB_BFG_OUT(false, false); break;
    case 1: OB_BFG_OUT(false, true); break;
    case 2: OB_BFG_OUT(true, false); break;
    case 3: OB_BFG_OUT(true, true); break;
  }
#undef OB_BFG_OUT
#undef OB_BFG_L
#undef OB_BFG_L2
  OB_LAUNCH_CHECK();
  return 0;
}

// ---------------------------------------------------------------------------
// f32 <-> bf16 helpers for the mixed-precision path
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void k_f32_to_bf16(const float* __restrict__ x,
                                                     __bf16* __restrict__ y,
                                                     int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * 256)
    y[i] = (__bf16)x[i];
}

// transposed cast: y[c][r] = x[r][c] for x [rows, cols]; y row stride
// out_ld >= rows (pad entries beyond rows are left untouched — zero the
// buffer once before the first call when out_ld > rows).
__global__ __launch_bounds__(256) void k_f32_to_bf16_t(const float* __restrict__ x,
                                                       __bf16* __restrict__ y,
                                                       int64_t rows,
                                                       int64_t cols,
                                                       int64_t out_ld) {
  for (int64_t idx = (int64_t)blockIdx.x * 256 + threadIdx.x; idx < rows * cols;
       idx += (int64_t)gridDim.x * 256) {
    const int64_t r = idx / cols, c = idx % cols;
    y[c * out_ld + r] = (__bf16)x[r * cols + c];
  }
}

__global__ __launch_bounds__(256) void k_bf16_to_f32(const __bf16* __restrict__ x,
                                                     float* __restrict__ y,
                                                     int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * 256)
    y[i] = (float)x[i];
}

__host__ __device__ static inline int64_t bmin64(int64_t a, int64_t b) { return a < b ? a : b; }

extern "C" int ob_f32_to_bf16(const void* x, void* y, int64_t n, void* stream) {
  k_f32_to_bf16<<<(int)bmin64((n + 255) / 256, 4096), 256, 0, S(stream)>>>(
      (const float*)x, (__bf16*)y, n);
  OB_LAUNCH_CHECK();
  return 0;
}
extern "C" int ob_f32_to_bf16_t(const void* x, void* y, int64_t rows,
                                int64_t cols, void* stream) {
  k_f32_to_bf16_t<<<(int)bmin64((rows * cols + 255) / 256, 4096), 256, 0,
                    S(stream)>>>((const float*)x, (__bf16*)y, rows, cols,
                                 rows);
  OB_LAUNCH_CHECK();
  return 0;
}
extern "C" int ob_f32_to_bf16_t_ld(const void* x, void* y, int64_t rows,
                                   int64_t cols, int64_t out_ld,
                                   void* stream) {
  k_f32_to_bf16_t<<<(int)bmin64((rows * cols + 255) / 256, 4096), 256, 0,
                    S(stream)>>>((const float*)x, (__bf16*)y, rows, cols,
                                 out_ld);
  OB_LAUNCH_CHECK();
  return 0;
}
extern "C" int ob_bf16_to_f32(const void* x, void* y, int64_t n, void* stream) {
  k_bf16_to_f32<<<(int)bmin64((n + 255) / 256, 4096), 256, 0, S(stream)>>>(
      (const __bf16*)x, (float*)y, n);
  OB_LAUNCH_CHECK();
  return 0;
}

// ---------------------------------------------------------------------------
// bf16-I/O elementwise / reduction kernels (fp32 math, fp32 params & grads).
// Mirrors the fp32 kernels in ob_kernels.hip; separate instantiations keep
// the proven fp32 path untouched.
// ---------------------------------------------------------------------------

__device__ __forceinline__ float bwave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;
}
__device__ __forceinline__ float bblock_sum256(float v, float* lds4) {
  v = bwave_sum(v);
  const int wid = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) lds4[wid] = v;
  __syncthreads();
  const float r = lds4[0] + lds4[1] + lds4[2] + lds4[3];
  __syncthreads();
  return r;
}
__device__ __forceinline__ float bblock_max256(float v, float* lds4) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, 64));
  const int wid = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) lds4[wid] = v;
  __syncthreads();
  const float r = fmaxf(fmaxf(lds4[0], lds4[1]), fmaxf(lds4[2], lds4[3]));
  __syncthreads();
  return r;
}

__global__ __launch_bounds__(256) void k_ln_fwd_bf16(
    const __bf16* __restrict__ x, const float* __restrict__ w,
    const float* __restrict__ b, __bf16* __restrict__ y,
    float* __restrict__ mean, float* __restrict__ rstd, int64_t rows, int H,
    float eps) {
  __shared__ float lds4[4];
  for (int64_t row = blockIdx.x; row < rows; row += gridDim.x) {
    const __bf16* xr = x + row * H;
    float s = 0.f, sq = 0.f;
    for (int c = threadIdx.x; c < H; c += 256) {
      const float v = bf2f(xr[c]);
      s += v;
      sq += v * v;
    }
    const float mu = bblock_sum256(s, lds4) / H;
    const float var = bblock_sum256(sq, lds4) / H - mu * mu;
    const float rs = rsqrtf(var + eps);
    if (threadIdx.x == 0) {
      mean[row] = mu;
      rstd[row] = rs;
    }
    __bf16* yr = y + row * H;
    for (int c = threadIdx.x; c < H; c += 256)
      yr[c] = (__bf16)((bf2f(xr[c]) - mu) * rs * w[c] + b[c]);
    __syncthreads();
  }
}

// wave-per-row ln forward (H in [512, 1024]): no block barriers, 16-B
// loads; row reductions are wave shfl ladders (the block-per-row kernel
// measured ~4x off the HBM roofline at H=768).
__global__ __launch_bounds__(256) void k_ln_fwd_bf16_w(
    const __bf16* __restrict__ x, const float* __restrict__ w,
    const float* __restrict__ b, __bf16* __restrict__ y,
    float* __restrict__ mean, float* __restrict__ rstd, int64_t rows, int H,
    float eps, int wrows) {
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  const int c0 = lane * 8, c1 = (lane + 64) * 8;
  const bool has1 = c1 < H;
  float wv0[8], bv0[8], wv1[8], bv1[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    wv0[j] = w[c0 + j];
    bv0[j] = b[c0 + j];
    wv1[j] = has1 ? w[c1 + j] : 0.f;
    bv1[j] = has1 ? b[c1 + j] : 0.f;
  }
  const int64_t r0 = (int64_t)blockIdx.x * (4 * wrows) + wid;
  // ROW PAIRS per iteration: the wave shfl reduction ladders of the two
  // rows interleave (ILP 4 on the shuffle latency) — the single-row
  // version measured 1.9 TB/s, shuffle-latency-bound, ~3x off roofline
  for (int i = 0; i < wrows; i += 2) {
    const int64_t rowa = r0 + (int64_t)i * 4;
    const int64_t rowb = r0 + (int64_t)(i + 1) * 4;
    if (rowa >= rows) return;
    const bool hb = rowb < rows;
    const __bf16* xa = x + rowa * H;
    const __bf16* xb = x + (hb ? rowb : rowa) * H;
    const uint4 xa0 = *reinterpret_cast<const uint4*>(xa + c0);
    const uint4 xb0 = *reinterpret_cast<const uint4*>(xb + c0);
    uint4 xa1 = {}, xb1 = {};
    if (has1) {
      xa1 = *reinterpret_cast<const uint4*>(xa + c1);
      xb1 = *reinterpret_cast<const uint4*>(xb + c1);
    }
    float va0[8], va1[8], vb0[8], vb1[8];
    float sa = 0.f, sqa = 0.f, sb = 0.f, sqb = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      va0[j] = bf2f(bf_extract(xa0, j));
      vb0[j] = bf2f(bf_extract(xb0, j));
      sa += va0[j];
      sqa += va0[j] * va0[j];
      sb += vb0[j];
      sqb += vb0[j] * vb0[j];
    }
    if (has1) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        va1[j] = bf2f(bf_extract(xa1, j));
        vb1[j] = bf2f(bf_extract(xb1, j));
        sa += va1[j];
        sqa += va1[j] * va1[j];
        sb += vb1[j];
        sqb += vb1[j] * vb1[j];
      }
    }
#pragma unroll
    for (int o = 32; o > 0; o >>= 1) {
      sa += __shfl_xor(sa, o, 64);
      sqa += __shfl_xor(sqa, o, 64);
      sb += __shfl_xor(sb, o, 64);
      sqb += __shfl_xor(sqb, o, 64);
    }
    const float mua = sa / H, mub = sb / H;
    const float rsa = rsqrtf(sqa / H - mua * mua + eps);
    const float rsb = rsqrtf(sqb / H - mub * mub + eps);
    if (lane == 0) {
      mean[rowa] = mua;
      rstd[rowa] = rsa;
      if (hb) {
        mean[rowb] = mub;
        rstd[rowb] = rsb;
      }
    }
    __bf16 oa[16], ob[16];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      oa[j] = (__bf16)((va0[j] - mua) * rsa * wv0[j] + bv0[j]);
      ob[j] = (__bf16)((vb0[j] - mub) * rsb * wv0[j] + bv0[j]);
    }
    if (has1) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        oa[8 + j] = (__bf16)((va1[j] - mua) * rsa * wv1[j] + bv1[j]);
        ob[8 + j] = (__bf16)((vb1[j] - mub) * rsb * wv1[j] + bv1[j]);
      }
    }
    *reinterpret_cast<uint4*>(y + rowa * H + c0) =
        *reinterpret_cast<const uint4*>(&oa[0]);
    if (has1)
      *reinterpret_cast<uint4*>(y + rowa * H + c1) =
          *reinterpret_cast<const uint4*>(&oa[8]);
    if (hb) {
      *reinterpret_cast<uint4*>(y + rowb * H + c0) =
          *reinterpret_cast<const uint4*>(&ob[0]);
      if (has1)
        *reinterpret_cast<uint4*>(y + rowb * H + c1) =
            *reinterpret_cast<const uint4*>(&ob[8]);
    }
  }
}

extern "C" int ob_layernorm_fwd_bf16(const void* x, const void* w,
                                     const void* b, void* y, void* mean,
                                     void* rstd, int64_t rows, int64_t H,
                                     float eps, void* stream) {
  if (H >= 512 && H <= 1024 && H % 8 == 0) {
    const int wrows = 4;
    const int grid = (int)((rows + 4 * wrows - 1) / (4 * wrows));
    k_ln_fwd_bf16_w<<<grid, 256, 0, S(stream)>>>(
        (const __bf16*)x, (const float*)w, (const float*)b, (__bf16*)y,
        (float*)mean, (float*)rstd, rows, (int)H, eps, wrows);
    OB_LAUNCH_CHECK();
    return 0;
  }
  const int grid = (int)bmin64(rows, 16384);
  k_ln_fwd_bf16<<<grid, 256, 0, S(stream)>>>(
      (const __bf16*)x, (const float*)w, (const float*)b, (__bf16*)y,
      (float*)mean, (float*)rstd, rows, (int)H, eps);
  OB_LAUNCH_CHECK();
  return 0;
}

#define BLN_CHUNK 16
#define BLN_WROWS 8
// wave-per-row ln backward (H <= 1024): each of the block's 4 waves owns
// its rows outright -- the row reductions are wave shfl ladders, no LDS
// and no per-row __syncthreads (the block-per-row kernel serialized two
// block reductions per row); 16-B vector loads throughout.  dw/db fold
// lane accumulators -> LDS (once) -> one global atomic per column per
// block.
template <bool DX_ACCUM>
__global__ __launch_bounds__(256) void k_ln_bwd_bf16_w(
    const __bf16* __restrict__ x, const float* __restrict__ w,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const __bf16* __restrict__ dy, __bf16* __restrict__ dx,
    float* __restrict__ dw, float* __restrict__ db, int64_t rows, int H) {
  __shared__ float redw[4][1024];
  __shared__ float redb[4][1024];
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  const int c0 = lane * 8, c1 = (lane + 64) * 8;
  const bool has1 = c1 < H;
  float accw[2][8] = {}, accb[2][8] = {};
  float wv0[8], wv1[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    wv0[j] = w[c0 + j];
    wv1[j] = has1 ? w[c1 + j] : 0.f;
  }
  const int64_t r0 = (int64_t)blockIdx.x * (4 * BLN_WROWS) + wid;
  // ROW PAIRS: interleave the two rows' shfl reduction ladders (ILP 4) —
  // single-row version measured 2.0 TB/s, shuffle-latency-bound
  for (int i = 0; i < BLN_WROWS; i += 2) {
    const int64_t rowa = r0 + (int64_t)i * 4;
    const int64_t rowb = r0 + (int64_t)(i + 1) * 4;
    if (rowa >= rows) break;
    const bool hb = rowb < rows;
    const __bf16* xra = x + rowa * H;
    const __bf16* dyra = dy + rowa * H;
    const __bf16* xrb = x + (hb ? rowb : rowa) * H;
    const __bf16* dyrb = dy + (hb ? rowb : rowa) * H;
    const float mua = mean[rowa], rsa = rstd[rowa];
    const float mub = hb ? mean[rowb] : 0.f, rsb = hb ? rstd[rowb] : 0.f;
    const uint4 xa0 = *reinterpret_cast<const uint4*>(xra + c0);
    const uint4 ya0 = *reinterpret_cast<const uint4*>(dyra + c0);
    const uint4 xb0 = *reinterpret_cast<const uint4*>(xrb + c0);
    const uint4 yb0 = *reinterpret_cast<const uint4*>(dyrb + c0);
    uint4 xa1 = {}, ya1 = {}, xb1 = {}, yb1 = {};
    if (has1) {
      xa1 = *reinterpret_cast<const uint4*>(xra + c1);
      ya1 = *reinterpret_cast<const uint4*>(dyra + c1);
      xb1 = *reinterpret_cast<const uint4*>(xrb + c1);
      yb1 = *reinterpret_cast<const uint4*>(dyrb + c1);
    }
    float s1a = 0.f, s2a = 0.f, s1b = 0.f, s2b = 0.f;
    float xha0[8], dya0[8], xha1[8], dya1[8];
    float xhb0[8], dyb0[8], xhb1[8], dyb1[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      xha0[j] = (bf2f(bf_extract(xa0, j)) - mua) * rsa;
      dya0[j] = bf2f(bf_extract(ya0, j));
      xhb0[j] = (bf2f(bf_extract(xb0, j)) - mub) * rsb;
      dyb0[j] = bf2f(bf_extract(yb0, j));
      s1a += dya0[j] * wv0[j] * xha0[j];
      s2a += dya0[j] * wv0[j];
      s1b += dyb0[j] * wv0[j] * xhb0[j];
      s2b += dyb0[j] * wv0[j];
      accw[0][j] += dya0[j] * xha0[j];
      accb[0][j] += dya0[j];
      if (hb) {
        accw[0][j] += dyb0[j] * xhb0[j];
        accb[0][j] += dyb0[j];
      }
    }
    if (has1) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        xha1[j] = (bf2f(bf_extract(xa1, j)) - mua) * rsa;
        dya1[j] = bf2f(bf_extract(ya1, j));
        xhb1[j] = (bf2f(bf_extract(xb1, j)) - mub) * rsb;
        dyb1[j] = bf2f(bf_extract(yb1, j));
        s1a += dya1[j] * wv1[j] * xha1[j];
        s2a += dya1[j] * wv1[j];
        s1b += dyb1[j] * wv1[j] * xhb1[j];
        s2b += dyb1[j] * wv1[j];
        accw[1][j] += dya1[j] * xha1[j];
        accb[1][j] += dya1[j];
        if (hb) {
          accw[1][j] += dyb1[j] * xhb1[j];
          accb[1][j] += dyb1[j];
        }
      }
    }
#pragma unroll
    for (int o = 32; o > 0; o >>= 1) {
      s1a += __shfl_xor(s1a, o, 64);
      s2a += __shfl_xor(s2a, o, 64);
      s1b += __shfl_xor(s1b, o, 64);
      s2b += __shfl_xor(s2b, o, 64);
    }
    const float m1a = s1a / H, m2a = s2a / H;
    const float m1b = s1b / H, m2b = s2b / H;
    __bf16 oxa[16], oxb[16];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float va = rsa * (dya0[j] * wv0[j] - m2a - xha0[j] * m1a);
      oxa[j] = (__bf16)(DX_ACCUM ? bf2f(dx[rowa * H + c0 + j]) + va : va);
      const float vb = rsb * (dyb0[j] * wv0[j] - m2b - xhb0[j] * m1b);
      oxb[j] =
          (__bf16)(DX_ACCUM && hb ? bf2f(dx[rowb * H + c0 + j]) + vb : vb);
    }
    if (has1) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float va = rsa * (dya1[j] * wv1[j] - m2a - xha1[j] * m1a);
        oxa[8 + j] =
            (__bf16)(DX_ACCUM ? bf2f(dx[rowa * H + c1 + j]) + va : va);
        const float vb = rsb * (dyb1[j] * wv1[j] - m2b - xhb1[j] * m1b);
        oxb[8 + j] =
            (__bf16)(DX_ACCUM && hb ? bf2f(dx[rowb * H + c1 + j]) + vb : vb);
      }
    }
    *reinterpret_cast<uint4*>(dx + rowa * H + c0) =
        *reinterpret_cast<const uint4*>(&oxa[0]);
    if (has1)
      *reinterpret_cast<uint4*>(dx + rowa * H + c1) =
          *reinterpret_cast<const uint4*>(&oxa[8]);
    if (hb) {
      *reinterpret_cast<uint4*>(dx + rowb * H + c0) =
          *reinterpret_cast<const uint4*>(&oxb[0]);
      if (has1)
        *reinterpret_cast<uint4*>(dx + rowb * H + c1) =
            *reinterpret_cast<const uint4*>(&oxb[8]);
    }
  }
  // dw/db: lane accs -> LDS (per wave) -> wave 0 folds -> one atomic/col
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    redw[wid][c0 + j] = accw[0][j];
    redb[wid][c0 + j] = accb[0][j];
    if (has1) {
      redw[wid][c1 + j] = accw[1][j];
      redb[wid][c1 + j] = accb[1][j];
    }
  }
  __syncthreads();
  for (int c = threadIdx.x; c < H; c += 256) {
    atomicAdd(&dw[c], redw[0][c] + redw[1][c] + redw[2][c] + redw[3][c]);
    atomicAdd(&db[c], redb[0][c] + redb[1][c] + redb[2][c] + redb[3][c]);
  }
}

template <bool DX_ACCUM>
__global__ __launch_bounds__(256) void k_ln_bwd_bf16(
    const __bf16* __restrict__ x, const float* __restrict__ w,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const __bf16* __restrict__ dy, __bf16* __restrict__ dx,
    float* __restrict__ dw, float* __restrict__ db, int64_t rows, int H) {
  __shared__ float lds4[4];
  const int64_t r0 = (int64_t)blockIdx.x * BLN_CHUNK;
  float accw[8] = {0}, accb[8] = {0};
  const int64_t rend = bmin64(rows, r0 + BLN_CHUNK);
  for (int64_t row = r0; row < rend; ++row) {
    const __bf16* xr = x + row * H;
    const __bf16* dyr = dy + row * H;
    const float mu = mean[row], rs = rstd[row];
    float s1 = 0.f, s2 = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int c = threadIdx.x + j * 256;
      if (c < H) {
        const float xhat = (bf2f(xr[c]) - mu) * rs;
        const float dyv = bf2f(dyr[c]);
        const float dyw = dyv * w[c];
        s1 += dyw * xhat;
        s2 += dyw;
        accw[j] += dyv * xhat;
        accb[j] += dyv;
      }
    }
    const float m1 = bblock_sum256(s1, lds4) / H;
    const float m2 = bblock_sum256(s2, lds4) / H;
    __bf16* dxr = dx + row * H;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int c = threadIdx.x + j * 256;
      if (c < H) {
        const float xhat = (bf2f(xr[c]) - mu) * rs;
        const float v = rs * (bf2f(dyr[c]) * w[c] - m2 - xhat * m1);
        dxr[c] = (__bf16)(DX_ACCUM ? bf2f(dxr[c]) + v : v);
      }
    }
    __syncthreads();
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int c = threadIdx.x + j * 256;
    if (c < H) {
      atomicAdd(&dw[c], accw[j]);
      atomicAdd(&db[c], accb[j]);
    }
  }
}

extern "C" int ob_layernorm_bwd_bf16(const void* x, const void* w,
                                     const void* mean, const void* rstd,
                                     const void* dy, void* dx, void* dw,
                                     void* db, int64_t rows, int64_t H,
                                     int dx_accum, void* stream) {
  if (H > 2048) return ob_fail("ln_bwd_bf16: H > 2048 unsupported");
  if (H % 8) return ob_fail("ln_bwd_bf16: H must be a multiple of 8");
  if (H >= 512 && H <= 1024) {
    const int grid = (int)((rows + 4 * BLN_WROWS - 1) / (4 * BLN_WROWS));
    if (dx_accum)
      k_ln_bwd_bf16_w<true><<<grid, 256, 0, S(stream)>>>(
          (const __bf16*)x, (const float*)w, (const float*)mean,
          (const float*)rstd, (const __bf16*)dy, (__bf16*)dx, (float*)dw,
          (float*)db, rows, (int)H);
    else
      k_ln_bwd_bf16_w<false><<<grid, 256, 0, S(stream)>>>(
          (const __bf16*)x, (const float*)w, (const float*)mean,
          (const float*)rstd, (const __bf16*)dy, (__bf16*)dx, (float*)dw,
          (float*)db, rows, (int)H);
    OB_LAUNCH_CHECK();
    return 0;
  }
  const int grid = (int)((rows + BLN_CHUNK - 1) / BLN_CHUNK);
  if (dx_accum)
    k_ln_bwd_bf16<true><<<grid, 256, 0, S(stream)>>>(
        (const __bf16*)x, (const float*)w, (const float*)mean,
        (const float*)rstd, (const __bf16*)dy, (__bf16*)dx, (float*)dw,
        (float*)db, rows, (int)H);
  else
    k_ln_bwd_bf16<false><<<grid, 256, 0, S(stream)>>>(
        (const __bf16*)x, (const float*)w, (const float*)mean,
        (const float*)rstd, (const __bf16*)dy, (__bf16*)dx, (float*)dw,
        (float*)db, rows, (int)H);
  OB_LAUNCH_CHECK();
  return 0;
}

__global__ __launch_bounds__(256) void k_softmax_fwd_bf16(
    __bf16* __restrict__ scores, int Sq, float scale) {
  __shared__ float lds4[4];
  const int64_t rid = blockIdx.x;
  const int row = (int)(rid % Sq);
  __bf16* p = scores + rid * Sq;
  const int valid = row + 1;
  float v[8];
  float mx = -INFINITY;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int c = threadIdx.x + j * 256;
    v[j] = (c < valid) ? bf2f(p[c]) * scale : -INFINITY;
    mx = fmaxf(mx, v[j]);
  }
  mx = bblock_max256(mx, lds4);
  float sum = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    v[j] = (v[j] == -INFINITY) ? 0.f : __expf(v[j] - mx);
    sum += v[j];
  }
  sum = bblock_sum256(sum, lds4);
  const float inv = 1.f / sum;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int c = threadIdx.x + j * 256;
    if (c < Sq) p[c] = (__bf16)(v[j] * inv);
  }
}



__device__ __forceinline__ uint4 bf_pack8_(const float* v) {
  return bf_pack8(v);
}

__global__ __launch_bounds__(256) void k_softmax_fwd_bf16_v8(
    __bf16* __restrict__ scores, int Sq, float scale) {
  __shared__ float lds4[4];
  const int64_t rid = blockIdx.x;
  const int row = (int)(rid % Sq);
  __bf16* p = scores + rid * Sq;
  const int valid = row + 1;
  const int c8 = threadIdx.x * 8;
  float v[8];
  uint4 in = {0, 0, 0, 0};
  if (c8 < Sq) in = *reinterpret_cast<const uint4*>(p + c8);
  float mx = -INFINITY;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    v[j] = (c8 + j < valid) ? bf2f(bf_extract(in, j)) * scale : -INFINITY;
    mx = fmaxf(mx, v[j]);
  }
  mx = bblock_max256(mx, lds4);
  float sum = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    v[j] = (v[j] == -INFINITY) ? 0.f : __expf(v[j] - mx);
    sum += v[j];
  }
  sum = bblock_sum256(sum, lds4);
  const float inv = 1.f / sum;
#pragma unroll
  for (int j = 0; j < 8; ++j) v[j] *= inv;
  if (c8 < Sq) *reinterpret_cast<uint4*>(p + c8) = bf_pack8(v);
}

extern "C" int ob_softmax_causal_fwd_bf16(void* scores, int64_t batch,
                                          int64_t Sq, float scale,
                                          void* stream) {
  if (Sq > 2048) return ob_fail("softmax_bf16: S > 2048 unsupported");
  if (Sq % 8 == 0)
    k_softmax_fwd_bf16_v8<<<(unsigned)(batch * Sq), 256, 0, S(stream)>>>(
        (__bf16*)scores, (int)Sq, scale);
  else
    k_softmax_fwd_bf16<<<(unsigned)(batch * Sq), 256, 0, S(stream)>>>(
        (__bf16*)scores, (int)Sq, scale);
  OB_LAUNCH_CHECK();
  return 0;
}

__global__ __launch_bounds__(256) void k_softmax_bwd_bf16(
    const __bf16* __restrict__ P, __bf16* __restrict__ dP, int Sq) {
  __shared__ float lds4[4];
  const int64_t rid = blockIdx.x;
  const int row = (int)(rid % Sq);
  const __bf16* pr = P + rid * Sq;
  __bf16* dr = dP + rid * Sq;
  const int valid = row + 1;
  float pv[8], dv[8];
  float t = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int c = threadIdx.x + j * 256;
    pv[j] = (c < valid) ? bf2f(pr[c]) : 0.f;
    dv[j] = (c < valid) ? bf2f(dr[c]) : 0.f;
    t += pv[j] * dv[j];
  }
  t = bblock_sum256(t, lds4);
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int c = threadIdx.x + j * 256;
    if (c < Sq) dr[c] = (__bf16)(pv[j] * (dv[j] - t));
  }
}

__global__ __launch_bounds__(256) void k_softmax_bwd_bf16_v8(
    const __bf16* __restrict__ P, __bf16* __restrict__ dP, int Sq) {
  __shared__ float lds4[4];
  const int64_t rid = blockIdx.x;
  const int row = (int)(rid % Sq);
  const __bf16* pr = P + rid * Sq;
  __bf16* dr = dP + rid * Sq;
  const int valid = row + 1;
  const int c8 = threadIdx.x * 8;
  uint4 pin = {0, 0, 0, 0}, din = {0, 0, 0, 0};
  if (c8 < Sq) {
    pin = *reinterpret_cast<const uint4*>(pr + c8);
    din = *reinterpret_cast<const uint4*>(dr + c8);
  }
  float pv[8], dv[8];
  float t = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    pv[j] = (c8 + j < valid) ? bf2f(bf_extract(pin, j)) : 0.f;
    dv[j] = (c8 + j < valid) ? bf2f(bf_extract(din, j)) : 0.f;
    t += pv[j] * dv[j];
  }
  t = bblock_sum256(t, lds4);
  float out[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) out[j] = pv[j] * (dv[j] - t);
  if (c8 < Sq) *reinterpret_cast<uint4*>(dr + c8) = bf_pack8(out);
}

extern "C" int ob_softmax_causal_bwd_bf16(const void* P, void* dP,
                                          int64_t batch, int64_t Sq,
                                          void* stream) {
  if (Sq > 2048) return ob_fail("softmax_bwd_bf16: S > 2048 unsupported");
  if (Sq % 8 == 0)
    k_softmax_bwd_bf16_v8<<<(unsigned)(batch * Sq), 256, 0, S(stream)>>>(
        (const __bf16*)P, (__bf16*)dP, (int)Sq);
  else
    k_softmax_bwd_bf16<<<(unsigned)(batch * Sq), 256, 0, S(stream)>>>(
        (const __bf16*)P, (__bf16*)dP, (int)Sq);
  OB_LAUNCH_CHECK();
  return 0;
}

#define BGELU_K 0.7978845608028654f
#define BGELU_C 0.044715f

// 8-wide (uint4) gelu: bf16 scalar loads cost ~2-2.5x (guide G13)
__global__ __launch_bounds__(256) void k_gelu_fwd_bf16(
    const __bf16* __restrict__ u, __bf16* __restrict__ g, int64_t n8) {
  for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < n8;
       i += (int64_t)gridDim.x * 256) {
    const uint4 in = *reinterpret_cast<const uint4*>(u + i * 8);
    float out[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float x = bf2f(bf_extract(in, j));
      const float t = tanhf(BGELU_K * (x + BGELU_C * x * x * x));
      out[j] = 0.5f * x * (1.f + t);
    }
    *reinterpret_cast<uint4*>(g + i * 8) = bf_pack8_(out);
  }
}
__global__ __launch_bounds__(256) void k_gelu_bwd_bf16(
    const __bf16* __restrict__ u, const __bf16* __restrict__ dg,
    __bf16* __restrict__ du, int64_t n8) {
  for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < n8;
       i += (int64_t)gridDim.x * 256) {
    const uint4 uin = *reinterpret_cast<const uint4*>(u + i * 8);
    const uint4 gin = *reinterpret_cast<const uint4*>(dg + i * 8);
    float out[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float x = bf2f(bf_extract(uin, j));
      const float t = tanhf(BGELU_K * (x + BGELU_C * x * x * x));
      const float d = 0.5f * (1.f + t) +
                      0.5f * x * (1.f - t * t) * BGELU_K *
                          (1.f + 3.f * BGELU_C * x * x);
      out[j] = bf2f(bf_extract(gin, j)) * d;
    }
    *reinterpret_cast<uint4*>(du + i * 8) = bf_pack8_(out);
  }
}
extern "C" int ob_gelu_fwd_bf16(const void* u, void* g, int64_t n,
                                void* stream) {
  if (n % 8) return ob_fail("gelu_bf16: n must be a multiple of 8");
  k_gelu_fwd_bf16<<<(int)bmin64((n / 8 + 255) / 256, 2048), 256, 0,
                    S(stream)>>>((const __bf16*)u, (__bf16*)g, n / 8);
  OB_LAUNCH_CHECK();
  return 0;
}
extern "C" int ob_gelu_bwd_bf16(const void* u, const void* dg, void* du,
                                int64_t n, void* stream) {
  if (n % 8) return ob_fail("gelu_bf16: n must be a multiple of 8");
  k_gelu_bwd_bf16<<<(int)bmin64((n / 8 + 255) / 256, 2048), 256, 0,
                    S(stream)>>>((const __bf16*)u, (const __bf16*)dg,
                                 (__bf16*)du, n / 8);
  OB_LAUNCH_CHECK();
  return 0;
}

__global__ __launch_bounds__(256) void k_colsum_bf16(
    const __bf16* __restrict__ X, float* __restrict__ db, int64_t M,
    int64_t N) {
  const int64_t c = (int64_t)blockIdx.x * 256 + threadIdx.x;
  if (c >= N) return;
  const int64_t r0 = (int64_t)blockIdx.y * 256;
  const int64_t r1 = bmin64(M, r0 + 256);
  float acc = 0.f;
  for (int64_t r = r0; r < r1; ++r) acc += bf2f(X[r * N + c]);
  atomicAdd(&db[c], acc);
}
// vectorized: each thread owns 8 contiguous columns (one uint4 per row)
__global__ __launch_bounds__(256) void k_colsum_bf16_v8(
    const __bf16* __restrict__ X, float* __restrict__ db, int64_t M,
    int64_t N) {
  const int64_t c8 = ((int64_t)blockIdx.x * 256 + threadIdx.x) * 8;
  if (c8 >= N) return;
  const int64_t r0 = (int64_t)blockIdx.y * 64;
  const int64_t r1 = bmin64(M, r0 + 64);
  float acc[8] = {0};
  for (int64_t r = r0; r < r1; ++r) {
    const uint4 v = *reinterpret_cast<const uint4*>(X + r * N + c8);
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[j] += bf2f(bf_extract(v, j));
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) atomicAdd(&db[c8 + j], acc[j]);
}

// two-level column sum (v9): block = 8 col-threads x 32 row-threads over
// a 64-col x 128-row tile; each thread keeps 4 independent uint4 loads
// in flight (the v8 kernel's single serial load stream measured ~640
// GB/s); wave shfl + LDS tree reduction, ONE global atomic per column
// per block.
__global__ __launch_bounds__(256) void k_colsum_bf16_v9(
    const __bf16* __restrict__ X, float* __restrict__ db, int64_t M,
    int64_t N) {
  __shared__ float red[4][64];
  const int tid = threadIdx.x;
  const int cg = tid & 7, rt = tid >> 3;       // col-thread, row-thread
  const int lane = tid & 63, w = tid >> 6;
  const int64_t c0 = (int64_t)blockIdx.x * 64 + cg * 8;
  const int64_t r0 = (int64_t)blockIdx.y * 128 + rt;
  float acc[8] = {0};
  if (c0 + 7 < N) {
    const __bf16* Xp = X + r0 * N + c0;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int64_t r = r0 + i * 32;
      if (r < M) {
        const uint4 v = *reinterpret_cast<const uint4*>(Xp + (int64_t)i * 32 * N);
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[j] += bf2f(bf_extract(v, j));
      }
    }
  } else if (c0 < N) {
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int64_t r = r0 + i * 32;
      if (r < M)
        for (int j = 0; j < (int)(N - c0); ++j)
          acc[j] += bf2f(X[r * N + c0 + j]);
    }
  }
  // wave: fold the 8 row-slots (lanes cg+8*s, s=0..7)
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    acc[j] += __shfl_down(acc[j], 32, 64);
    acc[j] += __shfl_down(acc[j], 16, 64);
    acc[j] += __shfl_down(acc[j], 8, 64);
  }
  if (lane < 8) {
#pragma unroll
    for (int j = 0; j < 8; ++j) red[w][lane * 8 + j] = acc[j];
  }
  __syncthreads();
  // first 64 threads: fold 4 waves, one atomic per column
  if (tid < 64) {
    const int64_t c = (int64_t)blockIdx.x * 64 + tid;
    if (c < N)
      atomicAdd(&db[c],
                red[0][tid] + red[1][tid] + red[2][tid] + red[3][tid]);
  }
}

extern "C" int ob_colsum_bf16(const void* X, void* db, int64_t M, int64_t N,
                              void* stream) {
  if (N % 8 == 0) {
    dim3 grid((unsigned)((N + 63) / 64), (unsigned)((M + 127) / 128));
    k_colsum_bf16_v9<<<grid, 256, 0, S(stream)>>>((const __bf16*)X,
                                                  (float*)db, M, N);
  } else {
    dim3 grid((unsigned)((N + 255) / 256), (unsigned)((M + 255) / 256));
    k_colsum_bf16<<<grid, 256, 0, S(stream)>>>((const __bf16*)X, (float*)db,
                                               M, N);
  }
  OB_LAUNCH_CHECK();
  return 0;
}

__global__ __launch_bounds__(256) void k_embed_fwd_bf16(
    const int64_t* __restrict__ ids, const float* __restrict__ wte,
    const float* __restrict__ wpe, __bf16* __restrict__ out, int64_t BS,
    int Sq, int H) {
  for (int64_t row = blockIdx.x; row < BS; row += gridDim.x) {
    const int64_t id = ids[row];
    const int s = (int)(row % Sq);
    const float* te = wte + id * H;
    const float* pe = wpe + (int64_t)s * H;
    __bf16* o = out + row * H;
    for (int c = threadIdx.x; c < H; c += 256) o[c] = (__bf16)(te[c] + pe[c]);
  }
}
extern "C" int ob_embed_fwd_bf16(const void* ids, const void* wte,
                                 const void* wpe, void* out, int64_t B,
                                 int64_t Sq, int64_t H, void* stream) {
  k_embed_fwd_bf16<<<(int)bmin64(B * Sq, 16384), 256, 0, S(stream)>>>(
      (const int64_t*)ids, (const float*)wte, (const float*)wpe, (__bf16*)out,
      B * Sq, (int)Sq, (int)H);
  OB_LAUNCH_CHECK();
  return 0;
}

__global__ __launch_bounds__(256) void k_embed_bwd_bf16(
    const int64_t* __restrict__ ids, const __bf16* __restrict__ dout,
    float* __restrict__ dwte, float* __restrict__ dwpe, int64_t BS, int Sq,
    int H) {
  for (int64_t row = blockIdx.x; row < BS; row += gridDim.x) {
    const int64_t id = ids[row];
    const int s = (int)(row % Sq);
    const __bf16* d = dout + row * H;
    for (int c = threadIdx.x; c < H; c += 256) {
      const float v = bf2f(d[c]);
      atomicAdd(&dwte[id * H + c], v);
      atomicAdd(&dwpe[(int64_t)s * H + c], v);
    }
  }
}
extern "C" int ob_embed_bwd_bf16(const void* ids, const void* dout, void* dwte,
                                 void* dwpe, int64_t B, int64_t Sq, int64_t H,
                                 void* stream) {
  k_embed_bwd_bf16<<<(int)bmin64(B * Sq, 16384), 256, 0, S(stream)>>>(
      (const int64_t*)ids, (const __bf16*)dout, (float*)dwte, (float*)dwpe,
      B * Sq, (int)Sq, (int)H);
  OB_LAUNCH_CHECK();
  return 0;
}

// CE with bf16 logits at row stride ld (padded to 8-half alignment); the
// shift/mean semantics match the fp32 kernels.
__global__ __launch_bounds__(256) void k_ce_fwd_bf16(
    const __bf16* __restrict__ logits, const int64_t* __restrict__ labels,
    float* __restrict__ lse, float* __restrict__ loss, int64_t BS, int Sq,
    int V, int64_t ld) {
  __shared__ float lmax[4], lsum[4];
  for (int64_t row = blockIdx.x; row < BS; row += gridDim.x) {
    const __bf16* lr = logits + row * ld;
    float m = -INFINITY, s = 0.f;
    // 16-B loads (scalar bf16 loads measured 3.6x off the HBM roofline);
    // pad columns (c >= V) masked out of the online max/sum
    for (int64_t c8 = (int64_t)threadIdx.x * 8; c8 < ld; c8 += 256 * 8) {
      const uint4 in = *reinterpret_cast<const uint4*>(lr + c8);
      float x[8];
      float m8 = -INFINITY;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        x[j] = (c8 + j < V) ? bf2f(bf_extract(in, j)) : -INFINITY;
        m8 = fmaxf(m8, x[j]);
      }
      if (m8 > -INFINITY) {
        const float nm = fmaxf(m, m8);
        float add = 0.f;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          add += (x[j] > -INFINITY) ? __expf(x[j] - nm) : 0.f;
        s = (m > -INFINITY ? s * __expf(m - nm) : 0.f) + add;
        m = nm;
      }
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      const float om = __shfl_down(m, off, 64);
      const float os = __shfl_down(s, off, 64);
      const float nm = fmaxf(m, om);
      const float t1 = (m > -INFINITY) ? s * __expf(m - nm) : 0.f;
      const float t2 = (om > -INFINITY) ? os * __expf(om - nm) : 0.f;
      s = t1 + t2;
      m = nm;
    }
    const int wid = threadIdx.x >> 6;
    if ((threadIdx.x & 63) == 0) {
      lmax[wid] = m;
      lsum[wid] = s;
    }
    __syncthreads();
    const float bm = fmaxf(fmaxf(lmax[0], lmax[1]), fmaxf(lmax[2], lmax[3]));
    float bs = 0.f;
#pragma unroll
    for (int wi = 0; wi < 4; ++wi)
      bs += (lmax[wi] > -INFINITY) ? lsum[wi] * __expf(lmax[wi] - bm) : 0.f;
    const float l = bm + __logf(bs);
    const int s_pos = (int)(row % Sq);
    if (threadIdx.x == 0) {
      lse[row] = l;
      if (s_pos < Sq - 1) {
        const int64_t lab = labels[row + 1];
        const int64_t B = BS / Sq;
        const float inv = 1.f / (float)(B * (Sq - 1));
        atomicAdd(loss, (l - bf2f(lr[lab])) * inv);
      }
    }
    __syncthreads();
  }
}
extern "C" int ob_ce_fwd_bf16(const void* logits, const void* labels,
                              void* lse, void* loss, int64_t B, int64_t Sq,
                              int64_t V, int64_t ld, void* stream) {
  k_ce_fwd_bf16<<<(int)bmin64(B * Sq, 16384), 256, 0, S(stream)>>>(
      (const __bf16*)logits, (const int64_t*)labels, (float*)lse,
      (float*)loss, B * Sq, (int)Sq, (int)V, ld);
  OB_LAUNCH_CHECK();
  return 0;
}

__global__ __launch_bounds__(256) void k_ce_bwd_bf16(
    __bf16* __restrict__ logits, const int64_t* __restrict__ labels,
    const float* __restrict__ lse, const float* __restrict__ dloss,
    int64_t BS, int Sq, int V, int64_t ld) {
  const float dl = dloss ? *dloss : 1.f;
  const int64_t B = BS / Sq;
  const float scale = dl / (float)(B * (Sq - 1));
  for (int64_t row = blockIdx.y; row < BS; row += gridDim.y) {
    __bf16* lr = logits + row * ld;
    const int s_pos = (int)(row % Sq);
    const float l = lse[row];
    const bool valid = s_pos < Sq - 1;
    const int64_t lab = valid ? labels[row + 1] : -1;
    for (int64_t c8 = ((int64_t)blockIdx.x * 256 + threadIdx.x) * 8;
         c8 < ld; c8 += (int64_t)gridDim.x * 256 * 8) {
      const uint4 in = *reinterpret_cast<const uint4*>(lr + c8);
      float out[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int64_t c = c8 + j;
        out[j] = (valid && c < V)
                     ? scale * (__expf(bf2f(bf_extract(in, j)) - l) -
                                (c == lab ? 1.f : 0.f))
                     : 0.f;
      }
      *reinterpret_cast<uint4*>(lr + c8) = bf_pack8(out);
    }
  }
}
extern "C" int ob_ce_bwd_bf16(void* logits, const void* labels,
                              const void* lse, const void* dloss, int64_t B,
                              int64_t Sq, int64_t V, int64_t ld,
                              void* stream) {
  if (ld % 8) return ob_fail("ce_bwd_bf16: ld must be a multiple of 8");
  dim3 grid((unsigned)bmin64((ld / 8 + 255) / 256, 256),
            (unsigned)bmin64(B * Sq, 16384));
  k_ce_bwd_bf16<<<grid, 256, 0, S(stream)>>>(
      (__bf16*)logits, (const int64_t*)labels, (const float*)lse,
      (const float*)dloss, B * Sq, (int)Sq, (int)V, ld);
  OB_LAUNCH_CHECK();
  return 0;
}

// ---------------------------------------------------------------------------
// Fast NT path (A [M,K] direct, B [N,K] direct — the fwd/dX hot shapes fed
// by the transposed weight shadows): async global_load_lds width-16 staging
// into lane-linear [row][32-half] images with a SOURCE-side slot swizzle
// (rule 21: glds writes base+lane*16, so the bank swizzle moves to the
// per-lane global address and the matching XOR on the read).  Swizzle key
// (row&3)^((row>>2)&3) makes the 16-lane ds_read_b128 fragment groups hit
// 16 distinct bank quads.  2-phase: issue next tile's glds, MFMA current,
// one __syncthreads per tile (its implicit vmcnt(0) drains the DMA —
// exactly the guide's minimum 2-phase glds recipe).  No staging registers
// -> 4 blocks/CU (the reg-staged kernel measured 62% SQ_WAIT_ANY: parked
// on HBM latency that 256-cycle bf16 MFMA segments cannot hide).
// ---------------------------------------------------------------------------

__device__ __forceinline__ int bf_xcd_swz(int bid, int nwg) {
  // T1: give each XCD a contiguous run of tiles so shared operand panels
  // stay in its private L2 (bijective variant for nwg % 8 != 0).
  const int q = nwg / 8, r = nwg % 8;
  const int xcd = bid % 8, idx = bid / 8;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

__device__ __forceinline__ int bf_swz_key(int row) {
  return (row & 3) ^ ((row >> 2) & 3);
}

template <int OUT>
__global__ __launch_bounds__(256, 3) void k_gemm_bf16_nt_glds(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ Cv, const float* __restrict__ bias,
    const __bf16* __restrict__ R, int M, int N, int K, int64_t lda,
    int64_t ldb, int64_t ldc, int64_t sA1, int64_t sA2, int64_t sB1,
    int64_t sB2, int64_t sC1, int64_t sC2, int n2, float alpha, float beta,
    int nbn, int Mr) {
  // 8 KB per operand per buffer, lane-linear; 3 buffers so two tiles'
  // DMAs stay in flight across the (raw) barrier — counted vmcnt(8)
  // waits only for the current tile's pieces (guide: 3-buf span +83%
  // over serial vs +40% for drain-per-tile).
  __shared__ __bf16 As[3][128 * 32];
  __shared__ __bf16 Bs[3][128 * 32];

  const int tile = bf_xcd_swz(blockIdx.x, gridDim.x);
  const int bm = tile / nbn, bn = tile % nbn;
  const int m0 = bm * 128, n0 = bn * 128;

  const int z = blockIdx.z;
  const int i1 = z / n2, i2 = z % n2;
  A += (int64_t)i1 * sA1 + (int64_t)i2 * sA2;
  B += (int64_t)i1 * sB1 + (int64_t)i2 * sB2;
  float* Cf = reinterpret_cast<float*>(Cv);
  __bf16* Cb = reinterpret_cast<__bf16*>(Cv);
  const int64_t coff = (int64_t)i1 * sC1 + (int64_t)i2 * sC2;
  Cf += coff;
  Cb += coff;
  if (R) R += coff;

  const int splitk = gridDim.y;
  const int kchunk = ((K + splitk * BF_BK - 1) / (splitk * BF_BK)) * BF_BK;
  const int kbeg = blockIdx.y * kchunk;
  const int kend = min(K, kbeg + kchunk);
  if (kbeg >= kend) return;

  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const int wr = w >> 1, wc = w & 1;
  const int il = lane & 31, kh = lane >> 5;

  // glds geometry: one 1-KiB wave instruction covers 16 rows x 64 B.
  // wave w stages rows [w*32, w*32+32) of each operand = 2 instructions.
  // lane l within an instruction: row16 = l/4, slot = l%4; the element
  // block written at [row][slot] must come from source half-range
  // (slot ^ key(row)) * 8.
  const int g_row16 = lane >> 2;        // 0..15
  const int g_slot = lane & 3;          // 16-B slot within the 64-B row
#define OB_NT_GLDS(BUF, KT)                                                   \
  {                                                                           \
    _Pragma("unroll") for (int half = 0; half < 2; ++half) {                  \
      const int row = w * 32 + half * 16 + g_row16;                           \
      const int srchalf = (g_slot ^ bf_swz_key(row)) * 8;                     \
      const __bf16* asrc = A + (int64_t)(m0 + row) * lda + (KT) + srchalf;    \
      const __bf16* bsrc = B + (int64_t)(n0 + row) * ldb + (KT) + srchalf;    \
      auto albase = (__attribute__((address_space(3))) void*)                 \
          (&As[BUF][(w * 32 + half * 16) * 32]);                              \
      auto blbase = (__attribute__((address_space(3))) void*)                 \
          (&Bs[BUF][(w * 32 + half * 16) * 32]);                              \
      __builtin_amdgcn_global_load_lds(                                       \
          (const __attribute__((address_space(1))) void*)asrc, albase, 16, 0, \
          0);                                                                 \
      __builtin_amdgcn_global_load_lds(                                       \
          (const __attribute__((address_space(1))) void*)bsrc, blbase, 16, 0, \
          0);                                                                 \
    }                                                                         \
  }

  f32x16 acc00 = {}, acc01 = {}, acc10 = {}, acc11 = {};

  // fragment read with the matching slot XOR
#define OB_NT_FRAG(IMG, ROW, SLOT)                                            \
  (*reinterpret_cast<const bf16x8*>(                                          \
      IMG + (ROW)*32 + (((SLOT) ^ bf_swz_key(ROW)) * 8)))

#define OB_NT_MFMA(BUF)                                                       \
  __builtin_amdgcn_s_setprio(1);                                              \
  _Pragma("unroll") for (int ks = 0; ks < 2; ++ks) {                          \
    const int slot = ks * 2 + kh;                                             \
    const int ra0 = wr * 64 + il, ra1 = wr * 64 + 32 + il;                    \
    const int rb0 = wc * 64 + il, rb1 = wc * 64 + 32 + il;                    \
    const bf16x8 a0 = OB_NT_FRAG(As[BUF], ra0, slot);                         \
    const bf16x8 a1 = OB_NT_FRAG(As[BUF], ra1, slot);                         \
    const bf16x8 b0 = OB_NT_FRAG(Bs[BUF], rb0, slot);                         \
    const bf16x8 b1 = OB_NT_FRAG(Bs[BUF], rb1, slot);                         \
    acc00 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b0, acc00, 0, 0, 0);  \
    acc01 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b1, acc01, 0, 0, 0);  \
    acc10 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b0, acc10, 0, 0, 0);  \
    acc11 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b1, acc11, 0, 0, 0);  \
  }

  // Prologue: two tiles in flight.  Each wave issues 4 glds per tile
  // (2 ops x 2 halves), so vmcnt(8) = "my tile-t pieces landed, tiles
  // t+1/t+2 still flying"; the raw barrier then guarantees every wave's
  // tile-t pieces landed (each waited its own count before arriving).
  OB_NT_GLDS(0, kbeg)
  const bool has1 = kbeg + BF_BK < kend;
  if (has1) OB_NT_GLDS(1, kbeg + BF_BK)
  int cur = 0;
  int kt = kbeg;
  for (; kt + 2 * BF_BK < kend; kt += BF_BK) {
    OB_NT_GLDS((cur + 2) % 3, kt + 2 * BF_BK)
    asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");
    OB_NT_MFMA(cur)
    __builtin_amdgcn_s_barrier();
    cur = (cur + 1) % 3;
  }
  // epilogue: drain the last one or two tiles
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
  OB_NT_MFMA(cur)
  if (has1) {
    __builtin_amdgcn_s_barrier();
    cur = (cur + 1) % 3;
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");
    OB_NT_MFMA(cur)
  }
#undef OB_NT_GLDS
#undef OB_NT_FRAG
#undef OB_NT_MFMA

  const int mw = m0 + wr * 64, nw = n0 + wc * 64;
#define OB_NT_EPI(ACC, TI, TJ)                                                \
  {                                                                           \
    const int nn = nw + (TJ)*32 + il;                                         \
    _Pragma("unroll") for (int r = 0; r < 16; ++r) {                          \
      const int mm = mw + (TI)*32 + (r & 3) + 8 * (r >> 2) + 4 * kh;          \
      if (mm >= Mr) continue;                                                 \
      float v = alpha * ACC[r];                                               \
      if (OUT == BF_OUT_F32_ATOMIC) {                                         \
        atomicAdd(&Cf[(int64_t)mm * ldc + nn], v);                            \
      } else {                                                                \
        if (bias) v += bias[nn];                                              \
        if (R) v += bf2f(R[(int64_t)mm * ldc + nn]);                          \
        if (OUT == BF_OUT_F32) {                                              \
          if (beta != 0.f) v += beta * Cf[(int64_t)mm * ldc + nn];            \
          Cf[(int64_t)mm * ldc + nn] = v;                                     \
        } else {                                                              \
          if (beta != 0.f) v += beta * bf2f(Cb[(int64_t)mm * ldc + nn]);      \
          Cb[(int64_t)mm * ldc + nn] = (__bf16)v;                             \
        }                                                                     \
      }                                                                       \
    }                                                                         \
  }
  OB_NT_EPI(acc00, 0, 0)
  OB_NT_EPI(acc01, 0, 1)
  OB_NT_EPI(acc10, 1, 0)
  OB_NT_EPI(acc11, 1, 1)
#undef OB_NT_EPI
}

int ob_gemm_bf16_nt_dispatch(const void* A, const void* B, void* C,
                             const void* bias, const void* residual,
                             int64_t M, int64_t N, int64_t K, int64_t lda,
                             int64_t ldb, int64_t ldc, int64_t sA1,
                             int64_t sA2, int64_t sB1, int64_t sB2,
                             int64_t sC1, int64_t sC2, int64_t n1, int64_t n2,
                             float alpha, float beta, int out_kind, int splitk,
                             void* stream, int64_t Mr) {
  const int nbm = (int)(M / 128), nbn = (int)(N / 128);
  dim3 grid(nbm * nbn, splitk, (unsigned)(n1 * n2));
  dim3 block(256);
#define OB_NTG(OUT_)                                                         \
  k_gemm_bf16_nt_glds<OUT_><<<grid, block, 0, S(stream)>>>(                  \
      (const __bf16*)A, (const __bf16*)B, C, (const float*)bias,             \
      (const __bf16*)residual, (int)M, (int)N, (int)K, lda, ldb, ldc, sA1,   \
      sA2, sB1, sB2, sC1, sC2, (int)n2, alpha, beta, nbn, (int)Mr)
  if (out_kind == BF_OUT_BF16) OB_NTG(BF_OUT_BF16);
  else if (out_kind == BF_OUT_F32) OB_NTG(BF_OUT_F32);
  else OB_NTG(BF_OUT_F32_ATOMIC);
#undef OB_NTG
  OB_LAUNCH_CHECK();
  return 0;
}

// ---------------------------------------------------------------------------
// bf16 matrix transpose (out[c][r] = in[r][c]) — feeds the weight-grad
// GEMMs: materializing X^T / dY^T turns the transpose-staged TN case
// (scattered 2-byte LDS writes, measured ~150 TF) into the fast NT glds
// path, at ~2 HBM passes over the activations (~12 ms/step total).
// 64x64 tiles through LDS ([64][65]-half padding de-conflicts the
// column-gather reads).
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void k_transpose_bf16(
    const __bf16* __restrict__ in, __bf16* __restrict__ out, int64_t R,
    int64_t C) {
  __shared__ __bf16 tile[64 * 65];
  const int64_t tr = blockIdx.y;  // 64-row tile index
  const int64_t tc = blockIdx.x;  // 64-col tile index
  const int64_t r0 = tr * 64, c0 = tc * 64;
  // load: thread t covers rows t/8 (+32), col-octet (t%8)*8
  {
    const int rr = threadIdx.x >> 3;
    const int c8 = (threadIdx.x & 7) * 8;
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int r = rr + it * 32;
      uint4 v = {0, 0, 0, 0};
      if (r0 + r < R && c0 + c8 + 7 < C) {
        v = *reinterpret_cast<const uint4*>(in + (r0 + r) * C + c0 + c8);
      } else if (r0 + r < R) {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (c0 + c8 + j < C)
            reinterpret_cast<unsigned short*>(&v)[j] = __builtin_bit_cast(
                unsigned short, in[(r0 + r) * C + c0 + c8 + j]);
      }
#pragma unroll
      for (int j = 0; j < 8; ++j)
        tile[r * 65 + c8 + j] = bf_extract(v, j);
    }
  }
  __syncthreads();
  // store: thread t covers out-rows (= in-cols) t/8 (+32), r-octet (t%8)*8
  {
    const int cc = threadIdx.x >> 3;
    const int r8 = (threadIdx.x & 7) * 8;
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int c = cc + it * 32;
      if (c0 + c >= C) continue;
      float dummy[1];
      (void)dummy;
      unsigned e[8];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        e[j] = bf_bits(tile[(r8 + j) * 65 + c]);
      uint4 v;
      v.x = e[0] | (e[1] << 16);
      v.y = e[2] | (e[3] << 16);
      v.z = e[4] | (e[5] << 16);
      v.w = e[6] | (e[7] << 16);
      if (r0 + r8 + 7 < R) {
        *reinterpret_cast<uint4*>(out + (c0 + c) * R + r0 + r8) = v;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (r0 + r8 + j < R)
            out[(c0 + c) * R + r0 + r8 + j] =
                bf_extract(v, j);
      }
    }
  }
}

extern "C" int ob_transpose_bf16(const void* in, void* out, int64_t R,
                                 int64_t C, void* stream) {
  dim3 grid((unsigned)((C + 63) / 64), (unsigned)((R + 63) / 64));
  k_transpose_bf16<<<grid, 256, 0, S(stream)>>>((const __bf16*)in,
                                                (__bf16*)out, R, C);
  OB_LAUNCH_CHECK();
  return 0;
}

// ---------------------------------------------------------------------------
// 256-row NT tile (BM=256, BN in {128,256}, BK=64, 8 waves): halves the
// per-operand re-read factor that bounds the 128² tile (PMC: 110 MB
// fetched vs 37.7 MB algorithmic on the fc shape) and quadruples the MFMA
// run between barriers (32 per wave per tile).  glds staging with the
// row-keyed slot swizzle (key = (row>>1)&7 over 128-B rows: the 16-lane
// ds_read_b128 groups land on 16 distinct bank quads).  2 LDS buffers,
// loads for tile t+1 in flight under tile t's MFMAs.
// ---------------------------------------------------------------------------

__device__ __forceinline__ int bf_swz_key8(int row) { return (row >> 1) & 7; }

template <int OUT, int BN>
__global__ __launch_bounds__(512, 2) void k_gemm_bf16_nt_256(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ Cv, const float* __restrict__ bias,
    const __bf16* __restrict__ R, int M, int N, int K, int64_t lda,
    int64_t ldb, int64_t ldc, int64_t sA1, int64_t sA2, int64_t sB1,
    int64_t sB2, int64_t sC1, int64_t sC2, int n2, float alpha, float beta,
    int nbn, int Mr) {
  constexpr int FN = BN / 128;  // N fragments per wave (1 or 2)
  __shared__ __bf16 As[2][256 * 64];
  __shared__ __bf16 Bs[2][BN * 64];

  const int tile = bf_xcd_swz(blockIdx.x, gridDim.x);
  const int bm = tile / nbn, bn = tile % nbn;
  const int m0 = bm * 256, n0 = bn * BN;

  const int z = blockIdx.z;
  const int i1 = z / n2, i2 = z % n2;
  A += (int64_t)i1 * sA1 + (int64_t)i2 * sA2;
  B += (int64_t)i1 * sB1 + (int64_t)i2 * sB2;
  float* Cf = reinterpret_cast<float*>(Cv);
  __bf16* Cb = reinterpret_cast<__bf16*>(Cv);
  const int64_t coff = (int64_t)i1 * sC1 + (int64_t)i2 * sC2;
  Cf += coff;
  Cb += coff;
  if (R) R += coff;

  const int splitk = gridDim.y;
  constexpr int BK = 64;
  const int kchunk = ((K + splitk * BK - 1) / (splitk * BK)) * BK;
  const int kbeg = blockIdx.y * kchunk;
  const int kend = min(K, kbeg + kchunk);
  if (kbeg >= kend) return;

  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;     // 0..7
  const int wr = w >> 2, wc = w & 3;  // 2 x 4 wave grid
  const int il = lane & 31, kh = lane >> 5;

  // glds: one instr = 8 rows x 128 B; lane l: row l/8, slot l%8 within it
  const int g_row8 = lane >> 3;
  const int g_slot = lane & 7;
#define OB_N2_GLDS(BUF, KT)                                                   \
  {                                                                           \
    _Pragma("unroll") for (int i = 0; i < 4; ++i) {                           \
      const int row = w * 32 + i * 8 + g_row8;                                \
      const int srch = (g_slot ^ bf_swz_key8(row)) * 8;                       \
      const __bf16* asrc = A + (int64_t)(m0 + row) * lda + (KT) + srch;       \
      auto albase = (__attribute__((address_space(3))) void*)                 \
          (&As[BUF][(w * 32 + i * 8) * 64]);                                  \
      __builtin_amdgcn_global_load_lds(                                       \
          (const __attribute__((address_space(1))) void*)asrc, albase, 16, 0, \
          0);                                                                 \
    }                                                                         \
    _Pragma("unroll") for (int i = 0; i < BN / 128; ++i) {                    \
      const int row = w * (BN / 8) + i * 8 + g_row8;                          \
      const int srch = (g_slot ^ bf_swz_key8(row)) * 8;                       \
      const __bf16* bsrc = B + (int64_t)(n0 + row) * ldb + (KT) + srch;       \
      auto blbase = (__attribute__((address_space(3))) void*)                 \
          (&Bs[BUF][(w * (BN / 8) + i * 8) * 64]);                            \
      __builtin_amdgcn_global_load_lds(                                       \
          (const __attribute__((address_space(1))) void*)bsrc, blbase, 16, 0, \
          0);                                                                 \
    }                                                                         \
  }

  f32x16 acc[4][FN];
#pragma unroll
  for (int mi = 0; mi < 4; ++mi)
#pragma unroll
    for (int ni = 0; ni < FN; ++ni) acc[mi][ni] = (f32x16){};

#define OB_N2_FRAG(IMG, ROW, SLOT)                                            \
  (*reinterpret_cast<const bf16x8*>(                                          \
      IMG + (ROW)*64 + (((SLOT) ^ bf_swz_key8(ROW)) * 8)))

#define OB_N2_MFMA(BUF)                                                       \
  __builtin_amdgcn_s_setprio(1);                                              \
  _Pragma("unroll") for (int ks = 0; ks < 4; ++ks) {                          \
    const int slot = ks * 2 + kh;                                             \
    bf16x8 bfr[FN];                                                           \
    _Pragma("unroll") for (int ni = 0; ni < FN; ++ni) bfr[ni] =               \
        OB_N2_FRAG(Bs[BUF], wc * (BN / 4) + ni * 32 + il, slot);              \
    _Pragma("unroll") for (int mi = 0; mi < 4; ++mi) {                        \
      const bf16x8 af =                                                       \
          OB_N2_FRAG(As[BUF], wr * 128 + mi * 32 + il, slot);                 \
      _Pragma("unroll") for (int ni = 0; ni < FN; ++ni) acc[mi][ni] =         \
          __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bfr[ni], acc[mi][ni],   \
                                                  0, 0, 0);                   \
    }                                                                         \
  }                                                                           \
  __builtin_amdgcn_s_setprio(0);

  // counted vmcnt + raw barriers: at 1 block/CU there is no co-resident
  // block to hide the drain, so never wait the in-flight next tile.
  // per-wave glds per tile: 4 (A) + BN/128*... = 8 (BN=256) or 6 (BN=128).
  OB_N2_GLDS(0, kbeg)
  int cur = 0;
  for (int kt = kbeg; kt + BK < kend; kt += BK) {
    OB_N2_GLDS(cur ^ 1, kt + BK)
    if (BN == 256)
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    asm volatile("" ::: "memory");
    OB_N2_MFMA(cur)
    __builtin_amdgcn_s_barrier();
    cur ^= 1;
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
  OB_N2_MFMA(cur)
#undef OB_N2_GLDS
#undef OB_N2_FRAG
#undef OB_N2_MFMA

  const int mw = m0 + wr * 128, nw = n0 + wc * (BN / 4);
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int ni = 0; ni < FN; ++ni) {
      const int nn = nw + ni * 32 + il;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int mm = mw + mi * 32 + (r & 3) + 8 * (r >> 2) + 4 * kh;
        if (mm >= Mr) continue;
        float v = alpha * acc[mi][ni][r];
        if (OUT == BF_OUT_F32_ATOMIC) {
          atomicAdd(&Cf[(int64_t)mm * ldc + nn], v);
        } else {
          if (bias) v += bias[nn];
          if (R) v += bf2f(R[(int64_t)mm * ldc + nn]);
          if (OUT == BF_OUT_F32) {
            if (beta != 0.f) v += beta * Cf[(int64_t)mm * ldc + nn];
            Cf[(int64_t)mm * ldc + nn] = v;
          } else {
            if (beta != 0.f) v += beta * bf2f(Cb[(int64_t)mm * ldc + nn]);
            Cb[(int64_t)mm * ldc + nn] = (__bf16)v;
          }
        }
      }
    }
  }
}

int ob_gemm_bf16_nt256_dispatch(const void* A, const void* B, void* C,
                                const void* bias, const void* residual,
                                int64_t M, int64_t N, int64_t K, int64_t lda,
                                int64_t ldb, int64_t ldc, int64_t sA1,
                                int64_t sA2, int64_t sB1, int64_t sB2,
                                int64_t sC1, int64_t sC2, int64_t n1,
                                int64_t n2, float alpha, float beta,
                                int out_kind, int splitk, void* stream,
                                int64_t Mr, int BN) {
  const int nbm = (int)((M + 255) / 256), nbn = (int)((N + BN - 1) / BN);
  dim3 grid(nbm * nbn, splitk, (unsigned)(n1 * n2));
  dim3 block(512);
#define OB_N2G(OUT_, BN_)                                                    \
  k_gemm_bf16_nt_256<OUT_, BN_><<<grid, block, 0, S(stream)>>>(              \
      (const __bf16*)A, (const __bf16*)B, C, (const float*)bias,             \
      (const __bf16*)residual, (int)M, (int)N, (int)K, lda, ldb, ldc, sA1,   \
      sA2, sB1, sB2, sC1, sC2, (int)n2, alpha, beta, nbn, (int)Mr)
  if (BN == 256) {
    if (out_kind == BF_OUT_BF16) OB_N2G(BF_OUT_BF16, 256);
    else if (out_kind == BF_OUT_F32) OB_N2G(BF_OUT_F32, 256);
    else OB_N2G(BF_OUT_F32_ATOMIC, 256);
  } else {
    if (out_kind == BF_OUT_BF16) OB_N2G(BF_OUT_BF16, 128);
    else if (out_kind == BF_OUT_F32) OB_N2G(BF_OUT_F32, 128);
    else OB_N2G(BF_OUT_F32_ATOMIC, 128);
  }
#undef OB_N2G
  OB_LAUNCH_CHECK();
  return 0;
}

// ---------------------------------------------------------------------------
// Flash-style fused causal attention, bf16, head_dim = 64 (§7 perf queue
// item 3; SURVEY.md "hard part (i)").  Forward: per (b, h, 128-row q-tile)
// block of 4 waves; each wave owns 32 q rows.  QK^T is computed SWAPPED
// (mfma(A=K, B=Q)) so each lane holds the scores of ONE q column (q =
// lane&31, 16 kv rows per accumulator) — the online-softmax row reduce is
// lane-local + one shfl_xor(32) partner combine.  P converts to the PV
// A-fragment with 8 pack-to-bf16 ops + 2 permlane32_swap per k-step (the
// kh halves exchange kv{4..7}/{8..11} exactly as the fragment k-runs
// need).  V is consumed through a materialized V^T ([z][64][S], cheap
// batched transpose) so the PV B-fragment reads are k-contiguous.  O
// accumulates in fp32; per-tile rescale factors broadcast wave-locally
// through LDS (no barriers anywhere in the loop).  Writes O (bf16) and
// the per-row LSE (fp32) for the backward recompute.
// ---------------------------------------------------------------------------

__device__ __forceinline__ unsigned bf_pk2(float a, float b) {
  return bf_bits((__bf16)a) | (bf_bits((__bf16)b) << 16);
}

// pack-and-exchange: acc-held values (rows pattern) -> A-fragment k-runs
__device__ __forceinline__ bf16x8 bf_dance(const float* pv8) {
  unsigned x0 = bf_pk2(pv8[0], pv8[1]);
  unsigned x1 = bf_pk2(pv8[2], pv8[3]);
  unsigned y0 = bf_pk2(pv8[4], pv8[5]);
  unsigned y1 = bf_pk2(pv8[6], pv8[7]);
  {
    auto r2 = __builtin_amdgcn_permlane32_swap(x0, y0, false, false);
    x0 = r2[0];
    y0 = r2[1];
  }
  {
    auto r2 = __builtin_amdgcn_permlane32_swap(x1, y1, false, false);
    x1 = r2[0];
    y1 = r2[1];
  }
  uint4 fr;
  fr.x = x0;
  fr.y = x1;
  fr.z = y0;
  fr.w = y1;
  return __builtin_bit_cast(bf16x8, fr);
}

// ---------------------------------------------------------------------------
// Software-pipelined flash forward (round 2): the kernel above serializes
// per tile (K loads -> S MFMAs -> softmax -> V^T loads -> PV MFMAs, nothing
// in flight across tiles — measured 15x off the MFMA floor, and occupancy
// alone did not move it: the stall is load latency on the critical path).
// This variant double-buffers the K fragments (prefetch tile t+1 during
// tile t's compute) and hoists the V^T fragment loads to the top of the
// tile (consumed last, ~full-tile latency cover).  Per-wave diagonal loop
// bound replaces the wave-uniform `continue`.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256, 2) void k_flash_fwd_bf16_pf(
    const __bf16* __restrict__ qkv, const __bf16* __restrict__ VT,
    __bf16* __restrict__ Obase, float* __restrict__ lse, int Sq, int H,
    int nh, float scale) {
  __shared__ float bcast[128];
  const int z = blockIdx.z;
  const int b = z / nh, h = z % nh;
  const int64_t qoff = (int64_t)b * Sq * 3 * H + h * 64;
  const __bf16* Qp = qkv + qoff;
  const __bf16* Kp = Qp + H;
  const __bf16* VTp = VT + (int64_t)z * 64 * Sq;
  __bf16* Op = Obase + (int64_t)b * Sq * H + h * 64;
  float* lsep = lse + (int64_t)z * Sq;

  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const int il = lane & 31, kh = lane >> 5;
  const int q0 = blockIdx.x * 128 + w * 32;
  const int myq = q0 + il;

  bf16x8 qf[4];
#pragma unroll
  for (int s = 0; s < 4; ++s)
    qf[s] = *reinterpret_cast<const bf16x8*>(
        Qp + (int64_t)myq * 3 * H + s * 16 + kh * 8);

  f32x16 o0 = {}, o1 = {};
  float m = -INFINITY, l = 0.f;
  const int my_ntiles = (q0 + 32) / 32;  // per-wave causal bound

  bf16x8 kfA[4], kfB[4];
#pragma unroll
  for (int s = 0; s < 4; ++s)
    kfA[s] = *reinterpret_cast<const bf16x8*>(
        Kp + (int64_t)il * 3 * H + s * 16 + kh * 8);

  auto tile = [&](int kvt, bf16x8 (&kcur)[4], bf16x8 (&knxt)[4]) {
    const int kv0 = kvt * 32;
    // V^T fragments for THIS tile first (consumed last)
    const bf16x8 v00 = *reinterpret_cast<const bf16x8*>(
        VTp + (int64_t)il * Sq + kv0 + kh * 8);
    const bf16x8 v01 = *reinterpret_cast<const bf16x8*>(
        VTp + (int64_t)(32 + il) * Sq + kv0 + kh * 8);
    const bf16x8 v10 = *reinterpret_cast<const bf16x8*>(
        VTp + (int64_t)il * Sq + kv0 + 16 + kh * 8);
    const bf16x8 v11 = *reinterpret_cast<const bf16x8*>(
        VTp + (int64_t)(32 + il) * Sq + kv0 + 16 + kh * 8);
    // prefetch NEXT tile's K fragments
    if (kvt + 1 < my_ntiles) {
#pragma unroll
      for (int s = 0; s < 4; ++s)
        knxt[s] = *reinterpret_cast<const bf16x8*>(
            Kp + (int64_t)(kv0 + 32 + il) * 3 * H + s * 16 + kh * 8);
    }
    f32x16 sacc = {};
#pragma unroll
    for (int s = 0; s < 4; ++s)
      sacc =
          __builtin_amdgcn_mfma_f32_32x32x16_bf16(kcur[s], qf[s], sacc, 0,
                                                  0, 0);
    float sv[16];
    float mt = -INFINITY;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int kv = kv0 + (r & 3) + 8 * (r >> 2) + 4 * kh;
      sv[r] = (kv <= myq) ? sacc[r] * scale : -INFINITY;
      mt = fmaxf(mt, sv[r]);
    }
    mt = fmaxf(mt, __shfl_xor(mt, 32, 64));
    const float mnew = fmaxf(m, mt);
    const float af = __expf(m - mnew);
    m = mnew;
    float psum = 0.f;
    float pv[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      pv[r] = (sv[r] == -INFINITY) ? 0.f : __expf(sv[r] - mnew);
      psum += pv[r];
    }
    psum += __shfl_xor(psum, 32, 64);
    l = l * af + psum;
    if (kh == 0) bcast[w * 32 + il] = af;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const float a = bcast[w * 32 + (r & 3) + 8 * (r >> 2) + 4 * kh];
      o0[r] *= a;
      o1[r] *= a;
    }
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      const bf16x8 pa = bf_dance(pv + t * 8);
      const bf16x8 v0 = t ? v10 : v00;
      const bf16x8 v1 = t ? v11 : v01;
      o0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa, v0, o0, 0, 0, 0);
      o1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa, v1, o1, 0, 0, 0);
    }
  };

  for (int kvt = 0; kvt < my_ntiles; kvt += 2) {
    tile(kvt, kfA, kfB);
    if (kvt + 1 < my_ntiles) tile(kvt + 1, kfB, kfA);
  }

  // epilogue: O /= l ; lse = m + log(l)  (identical to the kernel below:
  // acc rows = q offsets, cols = il = d)
  if (kh == 0) {
    bcast[w * 32 + il] = 1.f / l;
    lsep[myq] = m + __logf(l);
  }
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * kh;
    const float inv = bcast[w * 32 + row];
    Op[(int64_t)(q0 + row) * H + il] = (__bf16)(o0[r] * inv);
    Op[(int64_t)(q0 + row) * H + 32 + il] = (__bf16)(o1[r] * inv);
  }
}

// ---------------------------------------------------------------------------
// Flash forward v3 — the guide's 8-wave ladder structure
// (cdna_hip_programming.md §"Fused attention prefill", plain-HIP ladder):
//   * 8 waves x QBLK=32 q-rows = 256 q-rows per block; KVBLK=64.
//   * K tiles STAGED IN LDS, double-buffered, shared by all 8 waves (the
//     ladder's single biggest lever: the per-wave redundant global K
//     fragment loads become one coalesced 16 B/thread load + ds_write),
//     XOR-swizzled image (byte ^= (row&7)<<4) for ~conflict-free
//     ds_read_b128 A-fragments.
//   * V^T fragments stay GLOBAL (the VT materialization already exists;
//     8-wave + all q-blocks of a (b,h) reuse them through L2), hoisted to
//     the tile top.
//   * two independent S chains (kv sub-tiles) per tile + tree reductions
//     replace the serial fmax/sum chains (SQ_WAIT_INST_ANY was 51% after
//     register prefetch alone — dependency-bound).
//   * one barrier per tile (stage t+1 into the buffer whose readers
//     finished at the previous barrier).
// ---------------------------------------------------------------------------
// v4: the v3 structure at 4 waves / 128 q-rows per block — halves the
// intra-block causal diagonal spread the per-tile barrier serializes on
// (waves idle while the block's longest diagonal finishes), doubles the
// block count.  Same LDS staging, shared by 4 waves; each thread stages
// two 16 B chunks.
__global__ __launch_bounds__(256, 3) void k_flash_fwd_bf16_v4(
    const __bf16* __restrict__ qkv, const __bf16* __restrict__ VT,
    __bf16* __restrict__ Obase, float* __restrict__ lse, int Sq, int H,
    int nh, float scale) {
  __shared__ __bf16 kbuf[2][64 * 64];
  __shared__ __bf16 vbuf[2][64 * 64];
  __shared__ float bcast[128];
  const int z = blockIdx.z;
  const int b = z / nh, h = z % nh;
  const int64_t qoff = (int64_t)b * Sq * 3 * H + h * 64;
  const __bf16* Qp = qkv + qoff;
  const __bf16* Kp = Qp + H;
  const __bf16* VTp = VT + (int64_t)z * 64 * Sq;
  __bf16* Op = Obase + (int64_t)b * Sq * H + h * 64;
  float* lsep = lse + (int64_t)z * Sq;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const int il = lane & 31, kh = lane >> 5;
  const int q0b = blockIdx.x * 128;
  const int q0 = q0b + w * 32;
  const int myq = q0 + il;

  // staging: thread covers chunks tid and tid+256 of the 512-slot tile
  const int sr0 = tid >> 3, sc0 = tid & 7;
  const int sr1 = (tid + 256) >> 3, sc1 = (tid + 256) & 7;
  const int sw0 = sc0 ^ (sr0 & 7), sw1 = sc1 ^ (sr1 & 7);
  auto stage = [&](int buf, int kv0) {
    *reinterpret_cast<bf16x8*>(&kbuf[buf][sr0 * 64 + sw0 * 8]) =
        *reinterpret_cast<const bf16x8*>(
            Kp + (int64_t)(kv0 + sr0) * 3 * H + sc0 * 8);
    *reinterpret_cast<bf16x8*>(&kbuf[buf][sr1 * 64 + sw1 * 8]) =
        *reinterpret_cast<const bf16x8*>(
            Kp + (int64_t)(kv0 + sr1) * 3 * H + sc1 * 8);
    *reinterpret_cast<bf16x8*>(&vbuf[buf][sr0 * 64 + sw0 * 8]) =
        *reinterpret_cast<const bf16x8*>(
            VTp + (int64_t)sr0 * Sq + kv0 + sc0 * 8);
    *reinterpret_cast<bf16x8*>(&vbuf[buf][sr1 * 64 + sw1 * 8]) =
        *reinterpret_cast<const bf16x8*>(
            VTp + (int64_t)sr1 * Sq + kv0 + sc1 * 8);
  };

  bf16x8 qf[4];
#pragma unroll
  for (int s = 0; s < 4; ++s)
    qf[s] = *reinterpret_cast<const bf16x8*>(
        Qp + (int64_t)myq * 3 * H + s * 16 + kh * 8);

  const int ntiles = (q0b + 128) / 64;
  stage(0, 0);
  __syncthreads();

  f32x16 o0 = {}, o1 = {};
  float m = -INFINITY, l = 0.f;
  for (int kvt = 0; kvt < ntiles; ++kvt) {
    const int kv0 = kvt * 64;
    if (kvt + 1 < ntiles) stage((kvt + 1) & 1, kv0 + 64);
    const bool active = kv0 <= q0 + 31;
    if (active) {
      const __bf16* kb = kbuf[kvt & 1];
      const __bf16* vb = vbuf[kvt & 1];
      f32x16 sacc0 = {}, sacc1 = {};
#pragma unroll
      for (int s = 0; s < 4; ++s) {
        const int ch = (s * 2 + kh);
        const bf16x8 ka = *reinterpret_cast<const bf16x8*>(
            &kb[il * 64 + (ch ^ (il & 7)) * 8]);
        const bf16x8 kb2 = *reinterpret_cast<const bf16x8*>(
            &kb[(32 + il) * 64 + (ch ^ ((32 + il) & 7)) * 8]);
        sacc0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka, qf[s], sacc0, 0,
                                                        0, 0);
        sacc1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kb2, qf[s], sacc1,
                                                        0, 0, 0);
      }
      float sv[32];
      if (kv0 + 63 < q0) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          sv[r] = sacc0[r] * scale;
          sv[16 + r] = sacc1[r] * scale;
        }
      } else {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int kvr = kv0 + (r & 3) + 8 * (r >> 2) + 4 * kh;
          sv[r] = (kvr <= myq) ? sacc0[r] * scale : -INFINITY;
          sv[16 + r] = (kvr + 32 <= myq) ? sacc1[r] * scale : -INFINITY;
        }
      }
      float mx[8];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        mx[j] = fmaxf(fmaxf(sv[j], sv[8 + j]),
                      fmaxf(sv[16 + j], sv[24 + j]));
      float mt = fmaxf(fmaxf(fmaxf(mx[0], mx[1]), fmaxf(mx[2], mx[3])),
                       fmaxf(fmaxf(mx[4], mx[5]), fmaxf(mx[6], mx[7])));
      mt = fmaxf(mt, __shfl_xor(mt, 32, 64));
      const bool rescale = !__all(mt - m <= 8.f);
      if (rescale) {
        const float mnew = fmaxf(m, mt);
        const float af = __expf(m - mnew);
        m = mnew;
        if (kh == 0) bcast[w * 32 + il] = af;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const float a = bcast[w * 32 + (r & 3) + 8 * (r >> 2) + 4 * kh];
          o0[r] *= a;
          o1[r] *= a;
        }
        l *= af;
      }
      float ps[8] = {};
#pragma unroll
      for (int r = 0; r < 32; ++r) {
        sv[r] = (sv[r] == -INFINITY) ? 0.f : __expf(sv[r] - m);
        ps[r & 7] += sv[r];
      }
      float psum = ((ps[0] + ps[1]) + (ps[2] + ps[3])) +
                   ((ps[4] + ps[5]) + (ps[6] + ps[7]));
      psum += __shfl_xor(psum, 32, 64);
      l += psum;
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        const bf16x8 pa = bf_dance(sv + t * 8);
        const int ch = t * 2 + kh;
        const bf16x8 v0f = *reinterpret_cast<const bf16x8*>(
            &vb[il * 64 + (ch ^ (il & 7)) * 8]);
        const bf16x8 v1f = *reinterpret_cast<const bf16x8*>(
            &vb[(32 + il) * 64 + (ch ^ ((32 + il) & 7)) * 8]);
        o0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa, v0f, o0, 0, 0, 0);
        o1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa, v1f, o1, 0, 0, 0);
      }
    }
    __syncthreads();
  }

  if (kh == 0) {
    bcast[w * 32 + il] = 1.f / l;
    lsep[myq] = m + __logf(l);
  }
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * kh;
    const float inv = bcast[w * 32 + row];
    Op[(int64_t)(q0 + row) * H + il] = (__bf16)(o0[r] * inv);
    Op[(int64_t)(q0 + row) * H + 32 + il] = (__bf16)(o1[r] * inv);
  }
}

__global__ __launch_bounds__(512, 2) void k_flash_fwd_bf16_v3(
    const __bf16* __restrict__ qkv, const __bf16* __restrict__ VT,
    __bf16* __restrict__ Obase, float* __restrict__ lse, int Sq, int H,
    int nh, float scale) {
  __shared__ __bf16 kbuf[2][64 * 64];  // [buffer][row 64][col 64], swizzled
  __shared__ __bf16 vbuf[2][64 * 64];  // V^T tile ([d 64][kv 64]), swizzled
  __shared__ float bcast[256];
  const int z = blockIdx.z;
  const int b = z / nh, h = z % nh;
  const int64_t qoff = (int64_t)b * Sq * 3 * H + h * 64;
  const __bf16* Qp = qkv + qoff;
  const __bf16* Kp = Qp + H;
  const __bf16* VTp = VT + (int64_t)z * 64 * Sq;
  __bf16* Op = Obase + (int64_t)b * Sq * H + h * 64;
  float* lsep = lse + (int64_t)z * Sq;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const int il = lane & 31, kh = lane >> 5;
  const int q0b = blockIdx.x * 256;
  const int q0 = q0b + w * 32;
  const int myq = q0 + il;

  // staging map: thread t loads K row (t>>3), 16 B chunk (t&7) of the
  // tile; LDS write at chunk ^ (row&7) (read side XORs the same key)
  const int srow = tid >> 3, schunk = tid & 7;
  const int swz_chunk = schunk ^ (srow & 7);
  __bf16* swr = &kbuf[0][srow * 64 + swz_chunk * 8];
  __bf16* swr1 = &kbuf[1][srow * 64 + swz_chunk * 8];
  __bf16* svr = &vbuf[0][srow * 64 + swz_chunk * 8];
  __bf16* svr1 = &vbuf[1][srow * 64 + swz_chunk * 8];

  bf16x8 qf[4];
#pragma unroll
  for (int s = 0; s < 4; ++s)
    qf[s] = *reinterpret_cast<const bf16x8*>(
        Qp + (int64_t)myq * 3 * H + s * 16 + kh * 8);

  const int ntiles = (q0b + 256) / 64;  // block-uniform (barrier safety)
  // prologue: stage tile 0 (K + V^T), load tile 1 into registers
  bf16x8 g1, g1v;
  {
    const bf16x8 g0 = *reinterpret_cast<const bf16x8*>(
        Kp + (int64_t)srow * 3 * H + schunk * 8);
    const bf16x8 g0v = *reinterpret_cast<const bf16x8*>(
        VTp + (int64_t)srow * Sq + schunk * 8);
    *reinterpret_cast<bf16x8*>(swr) = g0;
    *reinterpret_cast<bf16x8*>(svr) = g0v;
    if (1 < ntiles) {
      g1 = *reinterpret_cast<const bf16x8*>(
          Kp + (int64_t)(64 + srow) * 3 * H + schunk * 8);
      g1v = *reinterpret_cast<const bf16x8*>(
          VTp + (int64_t)srow * Sq + 64 + schunk * 8);
    }
    __syncthreads();
  }

  f32x16 o0 = {}, o1 = {};
  float m = -INFINITY, l = 0.f;

  for (int kvt = 0; kvt < ntiles; ++kvt) {
    const int kv0 = kvt * 64;
    // stage tile t+1 (register -> LDS; its readers finished at the
    // previous barrier), then load tile t+2 into the register
    if (kvt + 1 < ntiles) {
      *reinterpret_cast<bf16x8*>((kvt & 1) ? swr : swr1) = g1;
      *reinterpret_cast<bf16x8*>((kvt & 1) ? svr : svr1) = g1v;
      if (kvt + 2 < ntiles) {
        g1 = *reinterpret_cast<const bf16x8*>(
            Kp + (int64_t)(kv0 + 128 + srow) * 3 * H + schunk * 8);
        g1v = *reinterpret_cast<const bf16x8*>(
            VTp + (int64_t)srow * Sq + kv0 + 128 + schunk * 8);
      }
    }
    const bool active = kv0 <= q0 + 31;  // wave-uniform causal skip
    if (active) {
      // two independent S chains from the swizzled LDS image
      const __bf16* kb = kbuf[kvt & 1];
      const __bf16* vb = vbuf[kvt & 1];
      f32x16 sacc0 = {}, sacc1 = {};
#pragma unroll
      for (int s = 0; s < 4; ++s) {
        const int ch = (s * 2 + kh);
        const bf16x8 ka = *reinterpret_cast<const bf16x8*>(
            &kb[il * 64 + (ch ^ (il & 7)) * 8]);
        const bf16x8 kb2 = *reinterpret_cast<const bf16x8*>(
            &kb[(32 + il) * 64 + (ch ^ ((32 + il) & 7)) * 8]);
        sacc0 =
            __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka, qf[s], sacc0, 0, 0,
                                                    0);
        sacc1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kb2, qf[s], sacc1,
                                                        0, 0, 0);
      }
      // masked scale + tree max over the 32 kv entries of this lane;
      // interior tiles (kv0+63 < q0 <= every myq) skip the masking
      float sv[32];
      if (kv0 + 63 < q0) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          sv[r] = sacc0[r] * scale;
          sv[16 + r] = sacc1[r] * scale;
        }
      } else {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int kvr = kv0 + (r & 3) + 8 * (r >> 2) + 4 * kh;
          sv[r] = (kvr <= myq) ? sacc0[r] * scale : -INFINITY;
          sv[16 + r] = (kvr + 32 <= myq) ? sacc1[r] * scale : -INFINITY;
        }
      }
      float mx[8];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        mx[j] = fmaxf(fmaxf(sv[j], sv[8 + j]),
                      fmaxf(sv[16 + j], sv[24 + j]));
      float mt = fmaxf(fmaxf(fmaxf(mx[0], mx[1]), fmaxf(mx[2], mx[3])),
                       fmaxf(fmaxf(mx[4], mx[5]), fmaxf(mx[6], mx[7])));
      mt = fmaxf(mt, __shfl_xor(mt, 32, 64));
      // defer-max (guide T13): skip the O-rescale (and its LDS
      // round-trip) when no row's max grew by more than 8 — P is then
      // bounded by e^8, which the f32 accumulators absorb; ~3x max-abs
      // error vs always-rescaling (documented, within the stated bf16
      // test tolerances).  Decision BEFORE exponentiation; wave-uniform.
      const bool rescale = !__all(mt - m <= 8.f);
      if (rescale) {
        const float mnew = fmaxf(m, mt);
        const float af = __expf(m - mnew);
        m = mnew;
        if (kh == 0) bcast[w * 32 + il] = af;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const float a = bcast[w * 32 + (r & 3) + 8 * (r >> 2) + 4 * kh];
          o0[r] *= a;
          o1[r] *= a;
        }
        l *= af;
      }
      // exponentiate IN PLACE (sv becomes P: -32 VGPR vs a second array)
      float ps[8] = {};
#pragma unroll
      for (int r = 0; r < 32; ++r) {
        sv[r] = (sv[r] == -INFINITY) ? 0.f : __expf(sv[r] - m);
        ps[r & 7] += sv[r];
      }
      float psum = ((ps[0] + ps[1]) + (ps[2] + ps[3])) +
                   ((ps[4] + ps[5]) + (ps[6] + ps[7]));
      psum += __shfl_xor(psum, 32, 64);
      l += psum;
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        const bf16x8 pa = bf_dance(sv + t * 8);
        const int ch = t * 2 + kh;
        const bf16x8 v0f = *reinterpret_cast<const bf16x8*>(
            &vb[il * 64 + (ch ^ (il & 7)) * 8]);
        const bf16x8 v1f = *reinterpret_cast<const bf16x8*>(
            &vb[(32 + il) * 64 + (ch ^ ((32 + il) & 7)) * 8]);
        o0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa, v0f, o0, 0, 0, 0);
        o1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa, v1f, o1, 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // epilogue (acc rows = q offsets, cols = il = d)
  if (kh == 0) {
    bcast[w * 32 + il] = 1.f / l;
    lsep[myq] = m + __logf(l);
  }
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * kh;
    const float inv = bcast[w * 32 + row];
    Op[(int64_t)(q0 + row) * H + il] = (__bf16)(o0[r] * inv);
    Op[(int64_t)(q0 + row) * H + 32 + il] = (__bf16)(o1[r] * inv);
  }
}

__global__ __launch_bounds__(256, 2) void k_flash_fwd_bf16(
    const __bf16* __restrict__ qkv, const __bf16* __restrict__ VT,
    __bf16* __restrict__ Obase, float* __restrict__ lse, int Sq, int H,
    int nh, float scale) {
  __shared__ float bcast[128];
  const int z = blockIdx.z;
  const int b = z / nh, h = z % nh;
  const int64_t qoff = (int64_t)b * Sq * 3 * H + h * 64;
  const __bf16* Qp = qkv + qoff;
  const __bf16* Kp = Qp + H;
  const __bf16* VTp = VT + (int64_t)z * 64 * Sq;
  __bf16* Op = Obase + (int64_t)b * Sq * H + h * 64;
  float* lsep = lse + (int64_t)z * Sq;

  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const int il = lane & 31, kh = lane >> 5;
  const int q0 = blockIdx.x * 128 + w * 32;
  const int myq = q0 + il;

  // Q fragments, resident for the whole block: B[k=d-run][j=q]
  bf16x8 qf[4];
#pragma unroll
  for (int s = 0; s < 4; ++s)
    qf[s] = *reinterpret_cast<const bf16x8*>(
        Qp + (int64_t)myq * 3 * H + s * 16 + kh * 8);

  f32x16 o0 = {}, o1 = {};
  float m = -INFINITY, l = 0.f;

  const int ntiles = (blockIdx.x * 128 + 128) / 32;
  for (int kvt = 0; kvt < ntiles; ++kvt) {
    const int kv0 = kvt * 32;
    if (kv0 > q0 + 31) continue;  // wave-uniform: fully masked for this wave

    // K fragments: A[i=kv][k=d-run]
    f32x16 sacc = {};
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      const bf16x8 kf = *reinterpret_cast<const bf16x8*>(
          Kp + (int64_t)(kv0 + il) * 3 * H + s * 16 + kh * 8);
      sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[s], sacc, 0, 0, 0);
    }

    // online softmax over this lane's 16 kv entries (column q = myq)
    float sv[16];
    float mt = -INFINITY;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int kv = kv0 + (r & 3) + 8 * (r >> 2) + 4 * kh;
      sv[r] = (kv <= myq) ? sacc[r] * scale : -INFINITY;
      mt = fmaxf(mt, sv[r]);
    }
    mt = fmaxf(mt, __shfl_xor(mt, 32, 64));
    const float mnew = fmaxf(m, mt);
    const float af = __expf(m - mnew);  // exp(-inf - x) = 0 on the 1st tile
    m = mnew;
    float psum = 0.f;
    float pv[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      pv[r] = (sv[r] == -INFINITY) ? 0.f : __expf(sv[r] - mnew);
      psum += pv[r];
    }
    psum += __shfl_xor(psum, 32, 64);
    l = l * af + psum;

    // rescale O by alpha per q-row (wave-local LDS broadcast, no barrier:
    // within-wave ds ordering is by lgkmcnt)
    if (kh == 0) bcast[w * 32 + il] = af;
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const float a = bcast[w * 32 + (r & 3) + 8 * (r >> 2) + 4 * kh];
      o0[r] *= a;
      o1[r] *= a;
    }

    // P -> A-fragments: pack + kh-half exchange (see header comment)
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      unsigned x0 = bf_pk2(pv[t * 8 + 0], pv[t * 8 + 1]);
      unsigned x1 = bf_pk2(pv[t * 8 + 2], pv[t * 8 + 3]);
      unsigned y0 = bf_pk2(pv[t * 8 + 4], pv[t * 8 + 5]);
      unsigned y1 = bf_pk2(pv[t * 8 + 6], pv[t * 8 + 7]);
      {
        auto r2 = __builtin_amdgcn_permlane32_swap(x0, y0, false, false);
        x0 = r2[0];
        y0 = r2[1];
      }
      {
        auto r2 = __builtin_amdgcn_permlane32_swap(x1, y1, false, false);
        x1 = r2[0];
        y1 = r2[1];
      }
      uint4 fr;
      fr.x = x0;
      fr.y = x1;
      fr.z = y0;
      fr.w = y1;
      const bf16x8 pa = __builtin_bit_cast(bf16x8, fr);
      // PV: B[k=kv-run][j=d] from V^T rows
      const bf16x8 v0 = *reinterpret_cast<const bf16x8*>(
          VTp + (int64_t)il * Sq + kv0 + t * 16 + kh * 8);
      const bf16x8 v1 = *reinterpret_cast<const bf16x8*>(
          VTp + (int64_t)(32 + il) * Sq + kv0 + t * 16 + kh * 8);
      o0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa, v0, o0, 0, 0, 0);
      o1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa, v1, o1, 0, 0, 0);
    }
  }

  // epilogue: O /= l ; lse = m + log(l)
  if (kh == 0) {
    bcast[w * 32 + il] = 1.f / l;
    lsep[myq] = m + __logf(l);
  }
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * kh;
    const float inv = bcast[w * 32 + row];
    Op[(int64_t)(q0 + row) * H + il] = (__bf16)(o0[r] * inv);
    Op[(int64_t)(q0 + row) * H + 32 + il] = (__bf16)(o1[r] * inv);
  }
}

extern "C" int ob_flash_fwd_bf16(const void* qkv, const void* VT, void* O,
                                 void* lse, int64_t B, int64_t Sq, int64_t H,
                                 int64_t nh, float scale, void* stream) {
  if (H / nh != 64) return ob_fail("flash_fwd: head_dim must be 64");
  if (Sq % 128) return ob_fail("flash_fwd: S must be a multiple of 128");
  dim3 grid((unsigned)(Sq / 128), 1, (unsigned)(B * nh));
  // kernel choice via OB_FLASH_FWD: "v3" (default; the 8-wave LDS-staged
  // ladder, needs Sq % 256 == 0 — other shapes fall back to v1), "pf"
  // (register-pipelined), "v1" (round 1)
  const char* fsel = getenv("OB_FLASH_FWD");
  // default v4: the 4-wave LDS ladder (57 us vs v3's 71 at the step
  // shape — halving the per-block causal-diagonal spread halves the
  // per-tile barrier straggling); "3"/"pf"/other pick the alternatives
  const char sel = fsel ? fsel[0] : '4';
  if (sel == '4' && Sq % 128 == 0) {
    dim3 grid4((unsigned)(Sq / 128), 1, (unsigned)(B * nh));
    k_flash_fwd_bf16_v4<<<grid4, 256, 0, S(stream)>>>(
        (const __bf16*)qkv, (const __bf16*)VT, (__bf16*)O, (float*)lse,
        (int)Sq, (int)H, (int)nh, scale);
    OB_LAUNCH_CHECK();
    return 0;
  }
  if (sel == '3' && Sq % 256 == 0) {
    dim3 grid3((unsigned)(Sq / 256), 1, (unsigned)(B * nh));
    k_flash_fwd_bf16_v3<<<grid3, 512, 0, S(stream)>>>(
        (const __bf16*)qkv, (const __bf16*)VT, (__bf16*)O, (float*)lse,
        (int)Sq, (int)H, (int)nh, scale);
    OB_LAUNCH_CHECK();
    return 0;
  }
  if (sel == 'p')
    k_flash_fwd_bf16_pf<<<grid, 256, 0, S(stream)>>>(
        (const __bf16*)qkv, (const __bf16*)VT, (__bf16*)O, (float*)lse,
        (int)Sq, (int)H, (int)nh, scale);
  else
    k_flash_fwd_bf16<<<grid, 256, 0, S(stream)>>>(
        (const __bf16*)qkv, (const __bf16*)VT, (__bf16*)O, (float*)lse,
        (int)Sq, (int)H, (int)nh, scale);
  OB_LAUNCH_CHECK();
  return 0;
}

// batched strided transpose: out[z][C][R] = in[z-slice][R][C] where the
// input slice z starts at in + z1*(z/n2? no) — two-level (b,h) strides like
// the GEMMs.  Used to materialize V^T (and the backward's Q^T/K^T/dO^T).
__global__ __launch_bounds__(256) void k_transpose_bf16_b(
    const __bf16* __restrict__ in, __bf16* __restrict__ out, int64_t R,
    int64_t C, int64_t sIn1, int64_t sIn2, int64_t ldin, int n2) {
  __shared__ __bf16 tile[64 * 65];
  const int z = blockIdx.z;
  const int i1 = z / n2, i2 = z % n2;
  const __bf16* src = in + (int64_t)i1 * sIn1 + (int64_t)i2 * sIn2;
  __bf16* dst = out + (int64_t)z * R * C;
  const int64_t r0 = (int64_t)blockIdx.y * 64, c0 = (int64_t)blockIdx.x * 64;
  {
    const int rr = threadIdx.x >> 3;
    const int c8 = (threadIdx.x & 7) * 8;
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int r = rr + it * 32;
      uint4 v = {0, 0, 0, 0};
      if (r0 + r < R && c0 + c8 + 7 < C)
        v = *reinterpret_cast<const uint4*>(src + (r0 + r) * ldin + c0 + c8);
#pragma unroll
      for (int j = 0; j < 8; ++j) tile[r * 65 + c8 + j] = bf_extract(v, j);
    }
  }
  __syncthreads();
  {
    const int cc = threadIdx.x >> 3;
    const int r8 = (threadIdx.x & 7) * 8;
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int c = cc + it * 32;
      if (c0 + c >= C) continue;
      float tmp[8];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        tmp[j] = bf2f(tile[(r8 + j) * 65 + c]);
      if (r0 + r8 + 7 < R)
        *reinterpret_cast<uint4*>(dst + (c0 + c) * R + r0 + r8) =
            bf_pack8(tmp);
    }
  }
}

extern "C" int ob_transpose_bf16_b(const void* in, void* out, int64_t R,
                                   int64_t C, int64_t sIn1, int64_t sIn2,
                                   int64_t ldin, int64_t n1, int64_t n2,
                                   void* stream) {
  if (R % 64 || C % 64) return ob_fail("transpose_b: R,C must be 64-multiples");
  dim3 grid((unsigned)((C + 63) / 64), (unsigned)((R + 63) / 64),
            (unsigned)(n1 * n2));
  k_transpose_bf16_b<<<grid, 256, 0, S(stream)>>>(
      (const __bf16*)in, (__bf16*)out, R, C, sIn1, sIn2, ldin, (int)n2);
  OB_LAUNCH_CHECK();
  return 0;
}

// ---------------------------------------------------------------------------
// Flash attention backward (recompute from Q, K, LSE; no stored P):
//   D[z][q]   = sum_d dO[q,d] * O[q,d]                    (k_flash_dsum)
//   dKdV      : blocks own 128 kv rows, loop q-tiles:
//               S = QK^T (acc col = kv), P = exp(scale*S - lse[q]),
//               dP = dO V^T-free (mfma(dO, V)), dS = P*(dP - D[q]),
//               dV += P^T dO (via dO^T), dK += scale * dS^T Q (via Q^T)
//   dQ        : blocks own 128 q rows, loop kv-tiles (fwd orientation):
//               S^T, P^T, dP^T (mfma(V, dO)), dS^T, dQ += scale*dS K (via K^T)
// Q^T/K^T/dO^T are materialized batched transposes ([z][64][S]).
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void k_flash_dsum(
    const __bf16* __restrict__ O, const __bf16* __restrict__ dO,
    float* __restrict__ D, int Sq, int H, int nh) {
  // 8 lanes x 8 elements (uint4) per row, 32 rows per block — the
  // one-lane-per-element version's 2-B loads measured 3x off roofline
  const int z = blockIdx.z;
  const int b = z / nh, h = z % nh;
  const int row = blockIdx.x * 32 + (threadIdx.x >> 3);
  if (row >= Sq) return;
  const int l8 = threadIdx.x & 7;
  const int64_t off =
      (int64_t)b * Sq * H + (int64_t)row * H + h * 64 + l8 * 8;
  const uint4 ov = *reinterpret_cast<const uint4*>(O + off);
  const uint4 dv = *reinterpret_cast<const uint4*>(dO + off);
  float s = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j)
    s += bf2f(bf_extract(ov, j)) * bf2f(bf_extract(dv, j));
#pragma unroll
  for (int o = 4; o > 0; o >>= 1) s += __shfl_down(s, o, 64);
  if (l8 == 0) D[(int64_t)z * Sq + row] = s;
}

extern "C" int ob_flash_dsum_bf16(const void* O, const void* dO, void* D,
                                  int64_t B, int64_t Sq, int64_t H, int64_t nh,
                                  void* stream) {
  dim3 grid((unsigned)((Sq + 31) / 32), 1, (unsigned)(B * nh));
  k_flash_dsum<<<grid, 256, 0, S(stream)>>>(
      (const __bf16*)O, (const __bf16*)dO, (float*)D, (int)Sq, (int)H,
      (int)nh);
  OB_LAUNCH_CHECK();
  return 0;
}

template <int MAXB>
__global__ __launch_bounds__(256, MAXB) void k_flash_bwd_dkdv(
    const __bf16* __restrict__ qkv, const __bf16* __restrict__ QT,
    const __bf16* __restrict__ dOT, const __bf16* __restrict__ dO,
    const float* __restrict__ lse, const float* __restrict__ D,
    __bf16* __restrict__ dqkv, int Sq, int H, int nh, float scale) {
  const int z = blockIdx.z;
  const int b = z / nh, h = z % nh;
  const int64_t base = (int64_t)b * Sq * 3 * H + h * 64;
  const __bf16* Qp = qkv + base;
  const __bf16* Kp = Qp + H;
  const __bf16* Vp = Qp + 2 * H;
  const __bf16* dOp = dO + (int64_t)b * Sq * H + h * 64;
  const __bf16* QTp = QT + (int64_t)z * 64 * Sq;
  const __bf16* dOTp = dOT + (int64_t)z * 64 * Sq;
  const float* lsep = lse + (int64_t)z * Sq;
  const float* Dp = D + (int64_t)z * Sq;

  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const int il = lane & 31, kh = lane >> 5;
  const int kv0 = blockIdx.x * 128 + w * 32;
  const int mykv = kv0 + il;

  // resident K and V fragments (j = kv columns for S and dP)
  bf16x8 kf[4], vf[4];
#pragma unroll
  for (int s = 0; s < 4; ++s) {
    kf[s] = *reinterpret_cast<const bf16x8*>(
        Kp + (int64_t)mykv * 3 * H + s * 16 + kh * 8);
    vf[s] = *reinterpret_cast<const bf16x8*>(
        Vp + (int64_t)mykv * 3 * H + s * 16 + kh * 8);
  }

  f32x16 dv0 = {}, dv1 = {}, dk0 = {}, dk1 = {};

  const int qt0 = (blockIdx.x * 128) / 32;
  const int nqt = Sq / 32;
  for (int qt = qt0; qt < nqt; ++qt) {
    const int q0 = qt * 32;
    if (q0 + 31 < kv0) continue;  // wave-uniform fully-masked
    // lse/D for this q-tile: one coalesced load per half-wave (lane il
    // holds row q0+il), fanned out by bpermute below — NOT per-element
    // global loads (32 scalar loads/tile measured as the kernel's
    // dominant stall).
    const float lse_t = lsep[q0 + il];
    const float d_t = Dp[q0 + il];
    // S-acc: mfma(A=Q(i=q), B=K(j=kv))  -> col = kv, rows = q
    f32x16 sacc = {}, dpacc = {};
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      const bf16x8 af = *reinterpret_cast<const bf16x8*>(
          Qp + (int64_t)(q0 + il) * 3 * H + s * 16 + kh * 8);
      const bf16x8 df = *reinterpret_cast<const bf16x8*>(
          dOp + (int64_t)(q0 + il) * H + s * 16 + kh * 8);
      sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, kf[s], sacc, 0, 0, 0);
      dpacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(df, vf[s], dpacc, 0, 0,
                                                      0);
    }
    // P and dS per element (one 8-run at a time: halves the live pv/dsv
    // registers so the kernel fits 3 waves/SIMD); rows are q
    // dV += P^T dO   (A = P^T frag: i = kv = lane col; k = q-run)
    // dK += scale * dS^T Q
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      float pv[8], dsv[8];
#pragma unroll
      for (int r8 = 0; r8 < 8; ++r8) {
        const int r = t * 8 + r8;
        const int qoff = (r & 3) + 8 * (r >> 2) + 4 * kh;
        const int q = q0 + qoff;
        const bool ok = q >= mykv;
        const float lse_q = __shfl(lse_t, qoff, 64);
        const float d_q = __shfl(d_t, qoff, 64);
        const float p = ok ? __expf(sacc[r] * scale - lse_q) : 0.f;
        pv[r8] = p;
        dsv[r8] = ok ? p * (dpacc[r] - d_q) : 0.f;
      }
      const bf16x8 pa = bf_dance(pv);
      const bf16x8 da = bf_dance(dsv);
      const bf16x8 do0 = *reinterpret_cast<const bf16x8*>(
          dOTp + (int64_t)il * Sq + q0 + t * 16 + kh * 8);
      const bf16x8 do1 = *reinterpret_cast<const bf16x8*>(
          dOTp + (int64_t)(32 + il) * Sq + q0 + t * 16 + kh * 8);
      const bf16x8 qt0f = *reinterpret_cast<const bf16x8*>(
          QTp + (int64_t)il * Sq + q0 + t * 16 + kh * 8);
      const bf16x8 qt1f = *reinterpret_cast<const bf16x8*>(
          QTp + (int64_t)(32 + il) * Sq + q0 + t * 16 + kh * 8);
      dv0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa, do0, dv0, 0, 0, 0);
      dv1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa, do1, dv1, 0, 0, 0);
      dk0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da, qt0f, dk0, 0, 0, 0);
      dk1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da, qt1f, dk1, 0, 0, 0);
    }
  }

  // write dK (scaled) and dV into dqkv slices: lane col = d, rows = kv
  __bf16* dKp = dqkv + base + H;
  __bf16* dVp = dqkv + base + 2 * H;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int kv = kv0 + (r & 3) + 8 * (r >> 2) + 4 * kh;
    dVp[(int64_t)kv * 3 * H + il] = (__bf16)dv0[r];
    dVp[(int64_t)kv * 3 * H + 32 + il] = (__bf16)dv1[r];
    dKp[(int64_t)kv * 3 * H + il] = (__bf16)(scale * dk0[r]);
    dKp[(int64_t)kv * 3 * H + 32 + il] = (__bf16)(scale * dk1[r]);
  }
}

// ---------------------------------------------------------------------------
// Paired-wave dKdV (the round-2 register diet): the
// 2-wave-occupancy kernel above holds K+V residents (32 VGPR), BOTH the
// S and dP accumulators (32) and BOTH d-halves of dV/dK (64) per wave —
// 217 VGPRs, 2 waves/SIMD.  Here a WAVE PAIR shares one 32-kv-row tile:
//   role 0: K resident, computes the S tile (Q K^T), owns d-half [0,32)
//   role 1: V resident, computes the dP tile (dO V^T), owns d-half [32,64)
// The S/dP tiles are exchanged through double-buffered LDS (one
// __syncthreads per q-tile), so each wave holds HALF of everything:
// ~140 VGPRs -> 3 waves/SIMD, same MFMA count per kv row, 2x the blocks
// (64 kv rows per block).  The causal q-loop start is block-uniform; the
// sub-diagonal tile pair-1 wastes is masked per element (p = 0).
// ---------------------------------------------------------------------------

__device__ __forceinline__ int acc_row(int r, int kh) {
  return (r & 3) + 8 * (r >> 2) + 4 * kh;
}

// ---------------------------------------------------------------------------
// dKdV v3: the paired-wave kernel below + LDS staging of the per-q-tile
// operands (Q, dO tiles [32][64]; Q^T, dO^T tiles [64][32]), shared by
// the block's 2 wave pairs — one coalesced 16 B/thread load replaces the
// per-wave global fragment reads.  Two buffers, TWO barriers per q-tile
// (the S/dP exchange splits reads across a barrier, so the single-barrier
// fwd-v3 scheme would race stage-writes with B-fragment reads).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256, 3) void k_flash_bwd_dkdv_v3(
    const __bf16* __restrict__ qkv, const __bf16* __restrict__ QT,
    const __bf16* __restrict__ dOT, const __bf16* __restrict__ dO,
    const float* __restrict__ lse, const float* __restrict__ D,
    __bf16* __restrict__ dqkv, int Sq, int H, int nh, float scale) {
  __shared__ float xch[2][2][2][32 * 32];
  __shared__ __bf16 qbuf[2][32 * 64];
  __shared__ __bf16 obuf[2][32 * 64];   // dO tile
  __shared__ __bf16 qtb[2][64 * 32];    // Q^T tile
  __shared__ __bf16 otb[2][64 * 32];    // dO^T tile
  const int z = blockIdx.z;
  const int b = z / nh, h = z % nh;
  const int64_t base = (int64_t)b * Sq * 3 * H + h * 64;
  const __bf16* Qp = qkv + base;
  const __bf16* Kp = Qp + H;
  const __bf16* Vp = Qp + 2 * H;
  const __bf16* dOp = dO + (int64_t)b * Sq * H + h * 64;
  const __bf16* QTp = QT + (int64_t)z * 64 * Sq;
  const __bf16* dOTp = dOT + (int64_t)z * 64 * Sq;
  const float* lsep = lse + (int64_t)z * Sq;
  const float* Dp = D + (int64_t)z * Sq;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const int pair = w >> 1, role = w & 1;
  const int il = lane & 31, kh = lane >> 5;
  const int kv0 = blockIdx.x * 64 + pair * 32;
  const int mykv = kv0 + il;

  const __bf16* KV = role ? Vp : Kp;
  bf16x8 of[4];
#pragma unroll
  for (int s = 0; s < 4; ++s)
    of[s] = *reinterpret_cast<const bf16x8*>(
        KV + (int64_t)mykv * 3 * H + s * 16 + kh * 8);

  // staging maps
  const int srow = tid >> 3, schunk = tid & 7;    // Q/dO: 32 rows x 8 chunks
  const int swzq = schunk ^ (srow & 7);
  const int trow = tid >> 2, tchunk = tid & 3;    // QT/dOT: 64 rows x 4 chunks
  const int swzt = tchunk ^ (trow & 3);
  auto stage = [&](int buf, int q0) {
    *reinterpret_cast<bf16x8*>(&qbuf[buf][srow * 64 + swzq * 8]) =
        *reinterpret_cast<const bf16x8*>(
            Qp + (int64_t)(q0 + srow) * 3 * H + schunk * 8);
    *reinterpret_cast<bf16x8*>(&obuf[buf][srow * 64 + swzq * 8]) =
        *reinterpret_cast<const bf16x8*>(
            dOp + (int64_t)(q0 + srow) * H + schunk * 8);
    *reinterpret_cast<bf16x8*>(&qtb[buf][trow * 32 + swzt * 8]) =
        *reinterpret_cast<const bf16x8*>(
            QTp + (int64_t)trow * Sq + q0 + tchunk * 8);
    *reinterpret_cast<bf16x8*>(&otb[buf][trow * 32 + swzt * 8]) =
        *reinterpret_cast<const bf16x8*>(
            dOTp + (int64_t)trow * Sq + q0 + tchunk * 8);
  };

  const int qt0 = (blockIdx.x * 64) / 32;
  const int nqt = Sq / 32;
  stage(qt0 & 1, qt0 * 32);
  __syncthreads();

  f32x16 dvh = {}, dkh = {};
  for (int qt = qt0; qt < nqt; ++qt) {
    const int q0 = qt * 32;
    if (qt + 1 < nqt) stage((qt + 1) & 1, q0 + 32);
    const float lse_t = lsep[q0 + il];
    const float d_t = Dp[q0 + il];
    // own tile: S (role 0, A = Q rows) or dP (role 1, A = dO rows)
    const __bf16* ab = role ? obuf[qt & 1] : qbuf[qt & 1];
    f32x16 own = {};
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      const int ch = s * 2 + kh;
      const bf16x8 af = *reinterpret_cast<const bf16x8*>(
          &ab[il * 64 + (ch ^ (il & 7)) * 8]);
      own = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, of[s], own, 0, 0, 0);
    }
    float* mine = xch[qt & 1][pair][role];
#pragma unroll
    for (int r = 0; r < 16; ++r) mine[acc_row(r, kh) * 32 + il] = own[r];
    __syncthreads();
    const float* theirs = xch[qt & 1][pair][role ^ 1];
    float other[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) other[r] = theirs[acc_row(r, kh) * 32 + il];

    const __bf16* dob = otb[qt & 1];
    const __bf16* qtbuf = qtb[qt & 1];
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      float pv[8], dsv[8];
#pragma unroll
      for (int r8 = 0; r8 < 8; ++r8) {
        const int r = t * 8 + r8;
        const int qoff = acc_row(r, kh);
        const bool ok = q0 + qoff >= mykv;
        const float sv = role == 0 ? own[r] : other[r];
        const float dpv = role == 0 ? other[r] : own[r];
        const float lse_q = __shfl(lse_t, qoff, 64);
        const float d_q = __shfl(d_t, qoff, 64);
        const float p = ok ? __expf(sv * scale - lse_q) : 0.f;
        pv[r8] = p;
        dsv[r8] = ok ? p * (dpv - d_q) : 0.f;
      }
      const bf16x8 pa = bf_dance(pv);
      const bf16x8 da = bf_dance(dsv);
      const int row = role * 32 + il;
      const int ch = t * 2 + kh;
      const bf16x8 dof = *reinterpret_cast<const bf16x8*>(
          &dob[row * 32 + ((ch ^ (row & 3)) * 8)]);
      const bf16x8 qtf = *reinterpret_cast<const bf16x8*>(
          &qtbuf[row * 32 + ((ch ^ (row & 3)) * 8)]);
      dvh = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa, dof, dvh, 0, 0, 0);
      dkh = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da, qtf, dkh, 0, 0, 0);
    }
    __syncthreads();
  }

  __bf16* dKp = dqkv + base + H;
  __bf16* dVp = dqkv + base + 2 * H;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int kv = kv0 + acc_row(r, kh);
    dVp[(int64_t)kv * 3 * H + role * 32 + il] = (__bf16)dvh[r];
    dKp[(int64_t)kv * 3 * H + role * 32 + il] = (__bf16)(scale * dkh[r]);
  }
}

// Split-duty paired dkdv (round-2d): the paired kernel below exchanges
// BOTH tiles (S and dP) and every wave runs the full exp + lse/D fan-out.
// Here role 0 owns the whole P chain (S -> P -> BOTH d-halves of dV) and
// role 1 owns the whole dS chain (dP -> dS -> BOTH d-halves of dK):
// only P crosses the pair (one-way, layout-preserving: lane L element r
// of role 0 is exactly what lane L element r of role 1 multiplies), role
// 1 skips the exp + causal mask entirely (masked P is already 0), and
// each wave runs HALF the lse/D shuffle fan-out.
__global__ __launch_bounds__(256, 3) void k_flash_bwd_dkdv_s(
    const __bf16* __restrict__ qkv, const __bf16* __restrict__ QT,
    const __bf16* __restrict__ dOT, const __bf16* __restrict__ dO,
    const float* __restrict__ lse, const float* __restrict__ D,
    __bf16* __restrict__ dqkv, int Sq, int H, int nh, float scale) {
  __shared__ float xch[2][2][32 * 32];  // [buffer][pair][P tile]
  __shared__ __bf16 qbuf[2][32 * 64];
  __shared__ __bf16 obuf[2][32 * 64];
  const int z = blockIdx.z;
  const int b = z / nh, h = z % nh;
  const int64_t base = (int64_t)b * Sq * 3 * H + h * 64;
  const __bf16* Qp = qkv + base;
  const __bf16* Kp = Qp + H;
  const __bf16* Vp = Qp + 2 * H;
  const __bf16* dOp = dO + (int64_t)b * Sq * H + h * 64;
  const __bf16* QTp = QT + (int64_t)z * 64 * Sq;
  const __bf16* dOTp = dOT + (int64_t)z * 64 * Sq;
  const float* lsep = lse + (int64_t)z * Sq;
  const float* Dp = D + (int64_t)z * Sq;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const int pair = w >> 1, role = w & 1;
  const int il = lane & 31, kh = lane >> 5;
  const int kv0 = blockIdx.x * 64 + pair * 32;
  const int mykv = kv0 + il;

  const __bf16* KV = role ? Vp : Kp;
  bf16x8 of[4];
#pragma unroll
  for (int s = 0; s < 4; ++s)
    of[s] = *reinterpret_cast<const bf16x8*>(
        KV + (int64_t)mykv * 3 * H + s * 16 + kh * 8);

  const int srow = tid >> 3, schunk = tid & 7;
  const int swzq = schunk ^ (srow & 7);
  auto stage = [&](int buf, int q0s) {
    *reinterpret_cast<bf16x8*>(&qbuf[buf][srow * 64 + swzq * 8]) =
        *reinterpret_cast<const bf16x8*>(
            Qp + (int64_t)(q0s + srow) * 3 * H + schunk * 8);
    *reinterpret_cast<bf16x8*>(&obuf[buf][srow * 64 + swzq * 8]) =
        *reinterpret_cast<const bf16x8*>(
            dOp + (int64_t)(q0s + srow) * H + schunk * 8);
  };

  f32x16 acc0 = {}, acc1 = {};  // role 0: dV d-halves; role 1: dK d-halves
  const int nqt = Sq / 32;
  const int qt0 = (blockIdx.x * 64) / 32;
  stage(qt0 & 1, qt0 * 32);
  __syncthreads();

  for (int qt = qt0; qt < nqt; ++qt) {
    const int q0 = qt * 32;
    if (qt + 1 < nqt) stage((qt + 1) & 1, q0 + 32);
    // B-operands early: BOTH d-halves of this role's output operand
    const __bf16* Bp = role ? QTp : dOTp;
    const bf16x8 b00 = *reinterpret_cast<const bf16x8*>(
        Bp + (int64_t)il * Sq + q0 + kh * 8);
    const bf16x8 b01 = *reinterpret_cast<const bf16x8*>(
        Bp + (int64_t)(32 + il) * Sq + q0 + kh * 8);
    const bf16x8 b10 = *reinterpret_cast<const bf16x8*>(
        Bp + (int64_t)il * Sq + q0 + 16 + kh * 8);
    const bf16x8 b11 = *reinterpret_cast<const bf16x8*>(
        Bp + (int64_t)(32 + il) * Sq + q0 + 16 + kh * 8);
    // own tile: S (role 0, A = Q rows) or dP (role 1, A = dO rows)
    const __bf16* ab = role ? obuf[qt & 1] : qbuf[qt & 1];
    f32x16 own = {};
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      const int ch = s * 2 + kh;
      const bf16x8 af = *reinterpret_cast<const bf16x8*>(
          &ab[il * 64 + (ch ^ (il & 7)) * 8]);
      own = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, of[s], own, 0, 0,
                                                    0);
    }
    float pv[16];
    float* ptile = xch[qt & 1][pair];
    if (role == 0) {
      const float lse_t = lsep[q0 + il];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qoff = acc_row(r, kh);
        const bool ok = q0 + qoff >= mykv;
        const float lse_q = __shfl(lse_t, qoff, 64);
        pv[r] = ok ? __expf(own[r] * scale - lse_q) : 0.f;
        ptile[qoff * 32 + il] = pv[r];
      }
    }
    __syncthreads();
    if (role == 1) {
      const float d_t = Dp[q0 + il];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qoff = acc_row(r, kh);
        const float d_q = __shfl(d_t, qoff, 64);
        // dS = P * (dP - D); masked entries have P == 0 already
        pv[r] = ptile[qoff * 32 + il] * (own[r] - d_q);
      }
    }
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      const bf16x8 a = bf_dance(pv + t * 8);
      const bf16x8 bl = t ? b10 : b00;
      const bf16x8 bh = t ? b11 : b01;
      acc0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, bl, acc0, 0, 0, 0);
      acc1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, bh, acc1, 0, 0, 0);
    }
  }

  __bf16* outp = dqkv + base + (role ? H : 2 * H);  // dK or dV slice
  const float os = role ? scale : 1.f;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int kv = kv0 + acc_row(r, kh);
    outp[(int64_t)kv * 3 * H + il] = (__bf16)(os * acc0[r]);
    outp[(int64_t)kv * 3 * H + 32 + il] = (__bf16)(os * acc1[r]);
  }
}

__global__ __launch_bounds__(256, 3) void k_flash_bwd_dkdv_p(
    const __bf16* __restrict__ qkv, const __bf16* __restrict__ QT,
    const __bf16* __restrict__ dOT, const __bf16* __restrict__ dO,
    const float* __restrict__ lse, const float* __restrict__ D,
    __bf16* __restrict__ dqkv, int Sq, int H, int nh, float scale) {
  // [buffer][pair][role][32 rows x 32 cols]
  __shared__ float xch[2][2][2][32 * 32];
  // hybrid LDS staging (round-2c): the q-tile's Q and dO rows — the
  // A-fragments, read BEFORE the exchange barrier, so the single-barrier
  // fwd-v3 buffering scheme applies — staged once per block instead of
  // per wave pair.  B-fragments (Q^T/dO^T) stay register-pipelined from
  // global: LDS-staging them too needs a second barrier, which measured
  // slower (dkdv_v3, profiles/r02_flash_probe.log).
  __shared__ __bf16 qbuf[2][32 * 64];
  __shared__ __bf16 obuf[2][32 * 64];
  const int z = blockIdx.z;
  const int b = z / nh, h = z % nh;
  const int64_t base = (int64_t)b * Sq * 3 * H + h * 64;
  const __bf16* Qp = qkv + base;
  const __bf16* Kp = Qp + H;
  const __bf16* Vp = Qp + 2 * H;
  const __bf16* dOp = dO + (int64_t)b * Sq * H + h * 64;
  const __bf16* QTp = QT + (int64_t)z * 64 * Sq;
  const __bf16* dOTp = dOT + (int64_t)z * 64 * Sq;
  const float* lsep = lse + (int64_t)z * Sq;
  const float* Dp = D + (int64_t)z * Sq;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const int pair = w >> 1, role = w & 1;
  const int il = lane & 31, kh = lane >> 5;
  const int kv0 = blockIdx.x * 64 + pair * 32;
  const int mykv = kv0 + il;
  const int srow = tid >> 3, schunk = tid & 7;
  const int swzq = schunk ^ (srow & 7);
  auto stage = [&](int buf, int q0s) {
    *reinterpret_cast<bf16x8*>(&qbuf[buf][srow * 64 + swzq * 8]) =
        *reinterpret_cast<const bf16x8*>(
            Qp + (int64_t)(q0s + srow) * 3 * H + schunk * 8);
    *reinterpret_cast<bf16x8*>(&obuf[buf][srow * 64 + swzq * 8]) =
        *reinterpret_cast<const bf16x8*>(
            dOp + (int64_t)(q0s + srow) * H + schunk * 8);
  };

  // one resident operand per role (K for S, V for dP)
  const __bf16* KV = role ? Vp : Kp;
  bf16x8 of[4];
#pragma unroll
  for (int s = 0; s < 4; ++s)
    of[s] = *reinterpret_cast<const bf16x8*>(
        KV + (int64_t)mykv * 3 * H + s * 16 + kh * 8);

  f32x16 dvh = {}, dkh = {};
  const int nqt = Sq / 32;
  const int qt0 = (blockIdx.x * 64) / 32;
  stage(qt0 & 1, qt0 * 32);
  __syncthreads();
  float lseA = lsep[qt0 * 32 + il], dA = Dp[qt0 * 32 + il];
  float lseB, dB;

  auto tile = [&](int qt, float lse_t, float d_t, float& lse_n,
                  float& d_n) {
    const int q0 = qt * 32;
    // stage tile t+1's Q/dO rows (single-barrier scheme: this buffer's
    // readers finished before the PREVIOUS tile's barrier)
    if (qt + 1 < nqt) stage((qt + 1) & 1, q0 + 32);
    // early: B-operands for this tile's dv/dk (consumed after the barrier)
    const bf16x8 dof0 = *reinterpret_cast<const bf16x8*>(
        dOTp + (int64_t)(role * 32 + il) * Sq + q0 + kh * 8);
    const bf16x8 qtf0 = *reinterpret_cast<const bf16x8*>(
        QTp + (int64_t)(role * 32 + il) * Sq + q0 + kh * 8);
    const bf16x8 dof1 = *reinterpret_cast<const bf16x8*>(
        dOTp + (int64_t)(role * 32 + il) * Sq + q0 + 16 + kh * 8);
    const bf16x8 qtf1 = *reinterpret_cast<const bf16x8*>(
        QTp + (int64_t)(role * 32 + il) * Sq + q0 + 16 + kh * 8);
    // prefetch tile t+1's lse/D
    if (qt + 1 < nqt) {
      lse_n = lsep[q0 + 32 + il];
      d_n = Dp[q0 + 32 + il];
    }
    // own tile: S (role 0, A = Q rows) or dP (role 1, A = dO rows),
    // A-fragments from the staged LDS image
    const __bf16* ab = role ? obuf[qt & 1] : qbuf[qt & 1];
    f32x16 own = {};
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      const int ch = s * 2 + kh;
      const bf16x8 af = *reinterpret_cast<const bf16x8*>(
          &ab[il * 64 + (ch ^ (il & 7)) * 8]);
      own = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, of[s], own, 0, 0,
                                                    0);
    }
    // exchange: write own tile, barrier, read the partner's
    float* mine = xch[qt & 1][pair][role];
#pragma unroll
    for (int r = 0; r < 16; ++r) mine[acc_row(r, kh) * 32 + il] = own[r];
    __syncthreads();
    const float* theirs = xch[qt & 1][pair][role ^ 1];
    float other[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) other[r] = theirs[acc_row(r, kh) * 32 + il];

#pragma unroll
    for (int t = 0; t < 2; ++t) {
      float pv[8], dsv[8];
#pragma unroll
      for (int r8 = 0; r8 < 8; ++r8) {
        const int r = t * 8 + r8;
        const int qoff = acc_row(r, kh);
        const bool ok = q0 + qoff >= mykv;
        const float sv = role == 0 ? own[r] : other[r];
        const float dpv = role == 0 ? other[r] : own[r];
        const float lse_q = __shfl(lse_t, qoff, 64);
        const float d_q = __shfl(d_t, qoff, 64);
        const float p = ok ? __expf(sv * scale - lse_q) : 0.f;
        pv[r8] = p;
        dsv[r8] = ok ? p * (dpv - d_q) : 0.f;
      }
      const bf16x8 pa = bf_dance(pv);
      const bf16x8 da = bf_dance(dsv);
      const bf16x8 dof = t ? dof1 : dof0;
      const bf16x8 qtf = t ? qtf1 : qtf0;
      dvh = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa, dof, dvh, 0, 0, 0);
      dkh = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da, qtf, dkh, 0, 0, 0);
    }
  };

  for (int qt = qt0; qt < nqt; qt += 2) {
    tile(qt, lseA, dA, lseB, dB);
    if (qt + 1 < nqt) tile(qt + 1, lseB, dB, lseA, dA);
  }

  __bf16* dKp = dqkv + base + H;
  __bf16* dVp = dqkv + base + 2 * H;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int kv = kv0 + acc_row(r, kh);
    dVp[(int64_t)kv * 3 * H + role * 32 + il] = (__bf16)dvh[r];
    dKp[(int64_t)kv * 3 * H + role * 32 + il] = (__bf16)(scale * dkh[r]);
  }
}

// ---------------------------------------------------------------------------
// dQ v3: the fwd-v3 ladder structure applied to the dQ loop — K, V and
// K^T tiles STAGED IN LDS (double-buffered, XOR-swizzled, shared by the
// block's 4 waves: one coalesced 16 B/thread load replaces each wave's
// redundant global fragment reads), KVBLK=64 with two independent
// S/dP chain pairs, tree-free in-lane masking as before.  One barrier
// per kv tile; stage-writes target the buffer whose readers finished
// before the previous barrier (the fwd-v3 scheme).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256, 2) void k_flash_bwd_dq_v3(
    const __bf16* __restrict__ qkv, const __bf16* __restrict__ KT,
    const __bf16* __restrict__ dO, const float* __restrict__ lse,
    const float* __restrict__ D, __bf16* __restrict__ dqkv, int Sq, int H,
    int nh, float scale) {
  __shared__ __bf16 kbuf[2][64 * 64];
  __shared__ __bf16 vbuf[2][64 * 64];
  __shared__ __bf16 tbuf[2][64 * 64];  // K^T tile ([d 64][kv 64])
  const int z = blockIdx.z;
  const int b = z / nh, h = z % nh;
  const int64_t base = (int64_t)b * Sq * 3 * H + h * 64;
  const __bf16* Qp = qkv + base;
  const __bf16* Kp = Qp + H;
  const __bf16* Vp = Qp + 2 * H;
  const __bf16* dOp = dO + (int64_t)b * Sq * H + h * 64;
  const __bf16* KTp = KT + (int64_t)z * 64 * Sq;
  const float* lsep = lse + (int64_t)z * Sq;
  const float* Dp = D + (int64_t)z * Sq;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int w = tid >> 6;
  const int il = lane & 31, kh = lane >> 5;
  const int q0 = blockIdx.x * 128 + w * 32;
  const int myq = q0 + il;
  const float mylse = lsep[myq];
  const float myD = Dp[myq];

  bf16x8 qf[4], df[4];
#pragma unroll
  for (int s = 0; s < 4; ++s) {
    qf[s] = *reinterpret_cast<const bf16x8*>(
        Qp + (int64_t)myq * 3 * H + s * 16 + kh * 8);
    df[s] = *reinterpret_cast<const bf16x8*>(
        dOp + (int64_t)myq * H + s * 16 + kh * 8);
  }

  // staging map: 256 threads x 2 rows each (row r and r+32), 16 B chunks
  const int srow = tid >> 3, schunk = tid & 7;  // rows 0..31
  const int swz0 = schunk ^ (srow & 7);
  const int swz1 = schunk ^ ((srow + 32) & 7);
  auto stage = [&](int buf, int kv0) {
    // K and V rows kv0+srow / kv0+srow+32; K^T rows d=srow / d=srow+32
    *reinterpret_cast<bf16x8*>(&kbuf[buf][srow * 64 + swz0 * 8]) =
        *reinterpret_cast<const bf16x8*>(
            Kp + (int64_t)(kv0 + srow) * 3 * H + schunk * 8);
    *reinterpret_cast<bf16x8*>(&kbuf[buf][(srow + 32) * 64 + swz1 * 8]) =
        *reinterpret_cast<const bf16x8*>(
            Kp + (int64_t)(kv0 + srow + 32) * 3 * H + schunk * 8);
    *reinterpret_cast<bf16x8*>(&vbuf[buf][srow * 64 + swz0 * 8]) =
        *reinterpret_cast<const bf16x8*>(
            Vp + (int64_t)(kv0 + srow) * 3 * H + schunk * 8);
    *reinterpret_cast<bf16x8*>(&vbuf[buf][(srow + 32) * 64 + swz1 * 8]) =
        *reinterpret_cast<const bf16x8*>(
            Vp + (int64_t)(kv0 + srow + 32) * 3 * H + schunk * 8);
    *reinterpret_cast<bf16x8*>(&tbuf[buf][srow * 64 + swz0 * 8]) =
        *reinterpret_cast<const bf16x8*>(
            KTp + (int64_t)srow * Sq + kv0 + schunk * 8);
    *reinterpret_cast<bf16x8*>(&tbuf[buf][(srow + 32) * 64 + swz1 * 8]) =
        *reinterpret_cast<const bf16x8*>(
            KTp + (int64_t)(srow + 32) * Sq + kv0 + schunk * 8);
  };

  const int ntiles = (blockIdx.x * 128 + 128) / 64;  // block-uniform
  stage(0, 0);
  __syncthreads();

  f32x16 dq0 = {}, dq1 = {};
  for (int kvt = 0; kvt < ntiles; ++kvt) {
    const int kv0 = kvt * 64;
    if (kvt + 1 < ntiles) stage((kvt + 1) & 1, kv0 + 64);
    const bool active = kv0 <= q0 + 31;  // wave-uniform causal skip
    if (active) {
      const __bf16* kb = kbuf[kvt & 1];
      const __bf16* vb = vbuf[kvt & 1];
      const __bf16* tb = tbuf[kvt & 1];
      // two kv sub-tiles: independent S and dP chains
      f32x16 s0 = {}, s1 = {}, p0 = {}, p1 = {};
#pragma unroll
      for (int s = 0; s < 4; ++s) {
        const int ch = s * 2 + kh;
        const bf16x8 ka = *reinterpret_cast<const bf16x8*>(
            &kb[il * 64 + (ch ^ (il & 7)) * 8]);
        const bf16x8 kb2 = *reinterpret_cast<const bf16x8*>(
            &kb[(32 + il) * 64 + (ch ^ ((32 + il) & 7)) * 8]);
        const bf16x8 va = *reinterpret_cast<const bf16x8*>(
            &vb[il * 64 + (ch ^ (il & 7)) * 8]);
        const bf16x8 vb2 = *reinterpret_cast<const bf16x8*>(
            &vb[(32 + il) * 64 + (ch ^ ((32 + il) & 7)) * 8]);
        s0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka, qf[s], s0, 0, 0, 0);
        s1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kb2, qf[s], s1, 0, 0,
                                                     0);
        p0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(va, df[s], p0, 0, 0, 0);
        p1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vb2, df[s], p1, 0, 0,
                                                     0);
      }
      float dsv[32];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kv = kv0 + (r & 3) + 8 * (r >> 2) + 4 * kh;
        const bool ok0 = kv <= myq;
        const bool ok1 = kv + 32 <= myq;
        const float pa = ok0 ? __expf(s0[r] * scale - mylse) : 0.f;
        const float pb = ok1 ? __expf(s1[r] * scale - mylse) : 0.f;
        dsv[r] = ok0 ? pa * (p0[r] - myD) : 0.f;
        dsv[16 + r] = ok1 ? pb * (p1[r] - myD) : 0.f;
      }
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        const bf16x8 da = bf_dance(dsv + t * 8);
        const int ch = t * 2 + kh;
        const bf16x8 kt0 = *reinterpret_cast<const bf16x8*>(
            &tb[il * 64 + (ch ^ (il & 7)) * 8]);
        const bf16x8 kt1 = *reinterpret_cast<const bf16x8*>(
            &tb[(32 + il) * 64 + (ch ^ ((32 + il) & 7)) * 8]);
        dq0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da, kt0, dq0, 0, 0, 0);
        dq1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da, kt1, dq1, 0, 0, 0);
      }
    }
    __syncthreads();
  }
  __bf16* dQp = dqkv + base;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int q = q0 + (r & 3) + 8 * (r >> 2) + 4 * kh;
    dQp[(int64_t)q * 3 * H + il] = (__bf16)(scale * dq0[r]);
    dQp[(int64_t)q * 3 * H + 32 + il] = (__bf16)(scale * dq1[r]);
  }
}

__global__ __launch_bounds__(256, 2) void k_flash_bwd_dq(
    const __bf16* __restrict__ qkv, const __bf16* __restrict__ KT,
    const __bf16* __restrict__ dO, const float* __restrict__ lse,
    const float* __restrict__ D, __bf16* __restrict__ dqkv, int Sq, int H,
    int nh, float scale) {
  const int z = blockIdx.z;
  const int b = z / nh, h = z % nh;
  const int64_t base = (int64_t)b * Sq * 3 * H + h * 64;
  const __bf16* Qp = qkv + base;
  const __bf16* Kp = Qp + H;
  const __bf16* Vp = Qp + 2 * H;
  const __bf16* dOp = dO + (int64_t)b * Sq * H + h * 64;
  const __bf16* KTp = KT + (int64_t)z * 64 * Sq;
  const float* lsep = lse + (int64_t)z * Sq;
  const float* Dp = D + (int64_t)z * Sq;

  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;
  const int il = lane & 31, kh = lane >> 5;
  const int q0 = blockIdx.x * 128 + w * 32;
  const int myq = q0 + il;
  const float mylse = lsep[myq];
  const float myD = Dp[myq];

  bf16x8 qf[4], df[4];
#pragma unroll
  for (int s = 0; s < 4; ++s) {
    qf[s] = *reinterpret_cast<const bf16x8*>(
        Qp + (int64_t)myq * 3 * H + s * 16 + kh * 8);
    df[s] = *reinterpret_cast<const bf16x8*>(
        dOp + (int64_t)myq * H + s * 16 + kh * 8);
  }

  f32x16 dq0 = {}, dq1 = {};
  const int my_nt = (q0 + 32) / 32;  // per-wave causal bound (was continue)
  // software pipeline: K/V fragments double-buffered across kv-tiles;
  // K^T fragments hoisted to the tile top (consumed last)
  bf16x8 kfA[4], vfA[4], kfB[4], vfB[4];
#pragma unroll
  for (int s = 0; s < 4; ++s) {
    kfA[s] = *reinterpret_cast<const bf16x8*>(
        Kp + (int64_t)il * 3 * H + s * 16 + kh * 8);
    vfA[s] = *reinterpret_cast<const bf16x8*>(
        Vp + (int64_t)il * 3 * H + s * 16 + kh * 8);
  }

  auto tile = [&](int kvt, bf16x8 (&kc)[4], bf16x8 (&vc)[4],
                  bf16x8 (&kn)[4], bf16x8 (&vn)[4]) {
    const int kv0 = kvt * 32;
    const bf16x8 kt00 = *reinterpret_cast<const bf16x8*>(
        KTp + (int64_t)il * Sq + kv0 + kh * 8);
    const bf16x8 kt01 = *reinterpret_cast<const bf16x8*>(
        KTp + (int64_t)(32 + il) * Sq + kv0 + kh * 8);
    const bf16x8 kt10 = *reinterpret_cast<const bf16x8*>(
        KTp + (int64_t)il * Sq + kv0 + 16 + kh * 8);
    const bf16x8 kt11 = *reinterpret_cast<const bf16x8*>(
        KTp + (int64_t)(32 + il) * Sq + kv0 + 16 + kh * 8);
    if (kvt + 1 < my_nt) {
#pragma unroll
      for (int s = 0; s < 4; ++s) {
        kn[s] = *reinterpret_cast<const bf16x8*>(
            Kp + (int64_t)(kv0 + 32 + il) * 3 * H + s * 16 + kh * 8);
        vn[s] = *reinterpret_cast<const bf16x8*>(
            Vp + (int64_t)(kv0 + 32 + il) * 3 * H + s * 16 + kh * 8);
      }
    }
    f32x16 sacc = {}, dpacc = {};
#pragma unroll
    for (int s = 0; s < 4; ++s) {
      sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kc[s], qf[s], sacc, 0,
                                                     0, 0);
      dpacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vc[s], df[s], dpacc, 0,
                                                      0, 0);
    }
    float dsv[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int kv = kv0 + (r & 3) + 8 * (r >> 2) + 4 * kh;
      const bool ok = kv <= myq;
      const float p = ok ? __expf(sacc[r] * scale - mylse) : 0.f;
      dsv[r] = ok ? p * (dpacc[r] - myD) : 0.f;
    }
#pragma unroll
    for (int t = 0; t < 2; ++t) {
      const bf16x8 da = bf_dance(dsv + t * 8);
      const bf16x8 kt0 = t ? kt10 : kt00;
      const bf16x8 kt1 = t ? kt11 : kt01;
      dq0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da, kt0, dq0, 0, 0, 0);
      dq1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da, kt1, dq1, 0, 0, 0);
    }
  };

  for (int kvt = 0; kvt < my_nt; kvt += 2) {
    tile(kvt, kfA, vfA, kfB, vfB);
    if (kvt + 1 < my_nt) tile(kvt + 1, kfB, vfB, kfA, vfA);
  }
  __bf16* dQp = dqkv + base;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int q = q0 + (r & 3) + 8 * (r >> 2) + 4 * kh;
    dQp[(int64_t)q * 3 * H + il] = (__bf16)(scale * dq0[r]);
    dQp[(int64_t)q * 3 * H + 32 + il] = (__bf16)(scale * dq1[r]);
  }
}

extern "C" int ob_flash_bwd_bf16(const void* qkv, const void* QT,
                                 const void* KT, const void* dOT,
                                 const void* dO, const void* lse,
                                 const void* D, void* dqkv, int64_t B,
                                 int64_t Sq, int64_t H, int64_t nh,
                                 float scale, void* stream) {
  if (H / nh != 64) return ob_fail("flash_bwd: head_dim must be 64");
  if (Sq % 128) return ob_fail("flash_bwd: S must be a multiple of 128");
  dim3 grid((unsigned)(Sq / 128), 1, (unsigned)(B * nh));
  // NOTE: the 3-wave build (168 VGPR + 52 B scratch) benched "faster"
  // but FAULTS — its dQ-phase never ran (parity tests caught it); it
  // stays opt-in (OB_DKDV3=1) for debugging only.
  static const bool occ3 = [] {
    const char* e = getenv("OB_DKDV3");
    return e && e[0] == '1';
  }();
  if (occ3) {
    k_flash_bwd_dkdv<3><<<grid, 256, 0, S(stream)>>>(
        (const __bf16*)qkv, (const __bf16*)QT, (const __bf16*)dOT,
        (const __bf16*)dO, (const float*)lse, (const float*)D,
        (__bf16*)dqkv, (int)Sq, (int)H, (int)nh, scale);
    OB_LAUNCH_CHECK();
    return 0;
  }
  // dkdv kernel choice: "p" (default) = register-pipelined paired-wave
  // (measured FASTER than the LDS-staged v3: the S/dP exchange already
  // forces one barrier; v3's second staging barrier costs more than the
  // shared loads save — 259 vs 302 us with dq-v3, profiles/
  // r02_flash_probe.log); "3" = LDS-staged v3; anything else = round 1.
  // Read per call so tools/flash_probe.py can A/B in one process.
  const char* pe = getenv("OB_FLASH_PAIR");
  // default "s": the split-duty pair (208.6 us whole-backward vs 231.2
  // for the two-way exchange, profiles/r02_flash_probe.log)
  const char dsel = pe ? pe[0] : 's';
  if (dsel == 's') {
    dim3 gridp((unsigned)(Sq / 64), 1, (unsigned)(B * nh));
    k_flash_bwd_dkdv_s<<<gridp, 256, 0, S(stream)>>>(
        (const __bf16*)qkv, (const __bf16*)QT, (const __bf16*)dOT,
        (const __bf16*)dO, (const float*)lse, (const float*)D,
        (__bf16*)dqkv, (int)Sq, (int)H, (int)nh, scale);
    OB_LAUNCH_CHECK();
  } else
  if (dsel == '3') {
    dim3 gridp((unsigned)(Sq / 64), 1, (unsigned)(B * nh));
    k_flash_bwd_dkdv_v3<<<gridp, 256, 0, S(stream)>>>(
        (const __bf16*)qkv, (const __bf16*)QT, (const __bf16*)dOT,
        (const __bf16*)dO, (const float*)lse, (const float*)D,
        (__bf16*)dqkv, (int)Sq, (int)H, (int)nh, scale);
    OB_LAUNCH_CHECK();
  } else if (dsel == 'p') {
    dim3 gridp((unsigned)(Sq / 64), 1, (unsigned)(B * nh));
    k_flash_bwd_dkdv_p<<<gridp, 256, 0, S(stream)>>>(
        (const __bf16*)qkv, (const __bf16*)QT, (const __bf16*)dOT,
        (const __bf16*)dO, (const float*)lse, (const float*)D,
        (__bf16*)dqkv, (int)Sq, (int)H, (int)nh, scale);
    OB_LAUNCH_CHECK();
  } else {
    k_flash_bwd_dkdv<2><<<grid, 256, 0, S(stream)>>>(
        (const __bf16*)qkv, (const __bf16*)QT, (const __bf16*)dOT,
        (const __bf16*)dO, (const float*)lse, (const float*)D, (__bf16*)dqkv,
        (int)Sq, (int)H, (int)nh, scale);
    OB_LAUNCH_CHECK();
  }
  // dq kernel choice: v3 LDS-staged ladder by default; OB_FLASH_DQ=1
  // reverts to the register-pipelined round-2 kernel
  const char* dqe = getenv("OB_FLASH_DQ");
  if (dqe && dqe[0] == '1')
    k_flash_bwd_dq<<<grid, 256, 0, S(stream)>>>(
        (const __bf16*)qkv, (const __bf16*)KT, (const __bf16*)dO,
        (const float*)lse, (const float*)D, (__bf16*)dqkv, (int)Sq, (int)H,
        (int)nh, scale);
  else
    k_flash_bwd_dq_v3<<<grid, 256, 0, S(stream)>>>(
        (const __bf16*)qkv, (const __bf16*)KT, (const __bf16*)dO,
        (const float*)lse, (const float*)D, (__bf16*)dqkv, (int)Sq, (int)H,
        (int)nh, scale);
  OB_LAUNCH_CHECK();
  return 0;
}

// ---------------------------------------------------------------------------
// 256x256 8-phase NT GEMM (the guide's fast template: glds staging,
// st_16x32 LDS swizzle, per-phase ds_read || glds || MFMA interleave,
// counted vmcnt once per K-tile, raw barriers, 1 block/CU).
//   tile 256x256, BK=64, 512 threads = 8 waves as 2(M) x 4(N);
//   per-wave output 128x64; MFMA 16x16x32 bf16 (fp32 acc).
//   LDS = 2 buffers x 4 images x [128 rows][64 k] bf16 = 128 KiB.
//   images: 0 = A rows 0-127, 1 = A rows 128-255, 2 = B rows 0-127,
//   3 = B rows 128-255 (B = [N,K] k-major, NT).
//   st_16x32 swizzle: image byte ^= ((byte>>9)&1)<<5, applied via the
//   SOURCE-side address on the lane-linear glds write and the matching
//   XOR on ds_read (glds cannot scatter).
//   K-tile = 4 phases, one output quadrant (4 M-frags x 2 N-frags) each;
//   staging runs 3 images ahead: phase 1 stages the next tile's last
//   image, phase 4 stages tile+2's first three (into the buffer whose
//   reads completed at phase 3's closing barrier), so the per-K-tile
//   wait is vmcnt(6) = "3 images still flying".
// ---------------------------------------------------------------------------

using f32x4 = __attribute__((ext_vector_type(4))) float;

// SWZ 0: within-row 8-slot XOR, key (row>>1)&7 (the nt256 kernel's
// measured-0-conflict pattern); SWZ 1: 16-slot XOR with the bit-6 flip.
template <int OUT, int SWZ = 0>
__global__ __launch_bounds__(512, 1) void k_gemm_bf16_nt_8ph(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ Cv, const float* __restrict__ bias,
    const __bf16* __restrict__ R, int M, int N, int K, int64_t lda,
    int64_t ldb, int64_t ldc, int64_t sA1, int64_t sA2, int64_t sB1,
    int64_t sB2, int64_t sC1, int64_t sC2, int n2, float alpha, float beta,
    int nbn, int Mr) {
  __shared__ __bf16 lds[2][4][128 * 64];

  const int tile = bf_xcd_swz(blockIdx.x, gridDim.x);
  const int bm = tile / nbn, bn = tile % nbn;
  const int m0 = bm * 256, n0 = bn * 256;

  const int z = blockIdx.z;
  const int i1 = z / n2, i2 = z % n2;
  A += (int64_t)i1 * sA1 + (int64_t)i2 * sA2;
  B += (int64_t)i1 * sB1 + (int64_t)i2 * sB2;
  float* Cf = reinterpret_cast<float*>(Cv);
  __bf16* Cb = reinterpret_cast<__bf16*>(Cv);
  const int64_t coff = (int64_t)i1 * sC1 + (int64_t)i2 * sC2;
  Cf += coff;
  Cb += coff;
  if (R) R += coff;

  const int splitk = gridDim.y;
  constexpr int BK = 64;
  const int kchunk = ((K + splitk * BK - 1) / (splitk * BK)) * BK;
  const int kbeg = blockIdx.y * kchunk;
  const int kend = min(K, kbeg + kchunk);
  if (kbeg >= kend) return;
  const int NT = (kend - kbeg) / BK;  // K % 64 == 0 gated at dispatch

  const int lane = threadIdx.x & 63;
  const int w = threadIdx.x >> 6;     // 0..7
  const int wr = w >> 2, wc = w & 3;  // 2(M) x 4(N)
  const int l16 = lane & 15, lg = lane >> 4;

  // ---- LDS swizzle (T2, conflict-free variant): element index
  //   idx = row*64 XOR (ke XOR ((row&15)<<3))
  // The (row&8)<<3 term crosses the 64-element row boundary (it flips
  // idx bit 6 = the bank-row half), so the 16 distinct rows of a
  // ds_read_b128 fragment group land on 16 distinct 16-B slots of the
  // 256-B bank row: conflict-free (the plain in-row XOR caps at 4-way).
  // glds writes are lane-linear, so the permutation moves to the SOURCE
  // address: lane l of block blk (8 image rows, 1 KiB) fetches
  //   row = blk*8 + (l>>4)*2 + (((l>>3)&1) ^ (blk&1))
  //   ce  = (l&7)*8 ^ ((row&7)<<3)
  // (the inverse of idx at a = blk*512 + l*8).
  // stage image IMG (0/1: A half, 2/3: B half) of K-tile at element KT
#define P8_GLDS1(BUF, IMG, KT)                                                \
  {                                                                           \
    const __bf16* const src0 = (IMG) < 2 ? A : B;                             \
    const int64_t ld = (IMG) < 2 ? lda : ldb;                                 \
    const int r0 = ((IMG)&1) * 128 + ((IMG) < 2 ? m0 : n0);                   \
    _Pragma("unroll") for (int v = 0; v < 2; ++v) {                           \
      const int blk = w * 2 + v;                                              \
      const int row =                                                         \
          SWZ ? blk * 8 + ((lane >> 4) << 1) + (((lane >> 3) & 1) ^ (blk & 1))\
              : blk * 8 + (lane >> 3);                                        \
      const int ce = SWZ ? ((lane & 7) * 8) ^ ((row & 7) << 3)                \
                         : ((lane & 7) * 8) ^ (((row >> 1) & 7) << 3);        \
      const __bf16* src = src0 + (int64_t)(r0 + row) * ld + (KT) + ce;        \
      auto lbase = (__attribute__((address_space(3))) void*)                  \
          (&lds[BUF][IMG][blk * 8 * 64]);                                     \
      __builtin_amdgcn_global_load_lds(                                       \
          (const __attribute__((address_space(1))) void*)src, lbase, 16, 0,   \
          0);                                                                 \
    }                                                                         \
  }

  // ---- swizzled fragment read: row r, k-run element ke (lane's 8-run)
#define P8_FRAG(BUF, IMG, ROW, KE)                                           \
  (*reinterpret_cast<const bf16x8*>(                                         \
      &lds[BUF][IMG][SWZ ? (((ROW)*64) ^ ((KE) ^ ((((ROW)&15)) << 3)))       \
                         : ((ROW)*64 + ((KE) ^ (((((ROW) >> 1)) & 7)         \
                                                << 3)))]))

  f32x4 acc[8][4];
#pragma unroll
  for (int mi = 0; mi < 8; ++mi)
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) acc[mi][ni] = (f32x4){};

  bf16x8 Ar[4][2];     // one A half (mi, kk): a=0 read at phase 1 (used
                       // p1, p2), overwritten by a=1 at phase 3
  bf16x8 Br[2][2][2];  // BOTH N-quarters (b, ni, kk): b=0 at phase 1,
                       // b=1 at phase 2, both live through phase 4 (the
                       // B image is then free for phase-3 staging)
  const int imA = wr;            // this wave's A image
  const int imB = 2 + (wc >> 1); // this wave's B image
  const int arow0 = l16;                    // + a*64 + mi*16
  const int brow0 = (wc & 1) * 64 + l16;    // + b*32 + ni*16

#define P8_RD_A(BUF, a)                                                      \
  _Pragma("unroll") for (int mi = 0; mi < 4; ++mi)                           \
      _Pragma("unroll") for (int kk = 0; kk < 2; ++kk) Ar[mi][kk] =          \
          P8_FRAG(BUF, imA, arow0 + (a)*64 + mi * 16, kk * 32 + lg * 8);
#define P8_RD_B(BUF, b)                                                      \
  _Pragma("unroll") for (int ni = 0; ni < 2; ++ni)                           \
      _Pragma("unroll") for (int kk = 0; kk < 2; ++kk) Br[b][ni][kk] =       \
          P8_FRAG(BUF, imB, brow0 + (b)*32 + ni * 16, kk * 32 + lg * 8);

  // kk outer: 8 independent MFMAs between the two updates of each acc
#define P8_MFMA(a, b)                                                        \
  __builtin_amdgcn_s_setprio(1);                                             \
  _Pragma("unroll") for (int kk = 0; kk < 2; ++kk)                           \
      _Pragma("unroll") for (int mi = 0; mi < 4; ++mi)                       \
          _Pragma("unroll") for (int ni = 0; ni < 2; ++ni)                   \
              acc[(a)*4 + mi][(b)*2 + ni] =                                  \
      __builtin_amdgcn_mfma_f32_16x16x32_bf16(                               \
          Ar[mi][kk], Br[b][ni][kk], acc[(a)*4 + mi][(b)*2 + ni], 0, 0,     \
          0);                                                                \
  __builtin_amdgcn_s_setprio(0);

#define P8_FENCE asm volatile("" ::: "memory")
#define P8_BAR                                                               \
  P8_FENCE;                                                                  \
  __builtin_amdgcn_s_barrier();                                              \
  P8_FENCE
#define P8_LGKM asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory")

  // ---- prologue: tile 0 fully, then tile 1's images in steady-state
  // issue order (img2=B0 as a phase-3 would, img0/img1=A as a phase-4)
  P8_GLDS1(0, 0, kbeg)
  P8_GLDS1(0, 1, kbeg)
  P8_GLDS1(0, 2, kbeg)
  P8_GLDS1(0, 3, kbeg)
  if (NT > 1) {
    P8_GLDS1(1, 2, kbeg + BK)
    P8_GLDS1(1, 0, kbeg + BK)
    P8_GLDS1(1, 1, kbeg + BK)
    asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  P8_BAR;

  for (int ti = 0; ti < NT; ++ti) {
    const int buf = ti & 1;
    const int kt = kbeg + ti * BK;
    // phase 1: quad (0,0); stage next tile's last image (other buffer)
    P8_RD_A(buf, 0)
    P8_RD_B(buf, 0)
    if (ti + 1 < NT) P8_GLDS1(buf ^ 1, 3, kt + BK)
    P8_BAR;
    P8_LGKM;
    P8_MFMA(0, 0)
    P8_BAR;
    // phase 2: quad (0,1) — last B reads of this tile's buffer
    P8_RD_B(buf, 1)
    P8_BAR;
    P8_LGKM;
    P8_MFMA(0, 1)
    P8_BAR;
    // phase 3: quad (1,0) — last A reads; stage tile+2's first B image
    // into this buffer (B reads completed at phase 2's closing barrier)
    P8_RD_A(buf, 1)
    if (ti + 2 < NT) P8_GLDS1(buf, 2, kt + 2 * BK)
    P8_BAR;
    P8_LGKM;
    P8_MFMA(1, 0)
    P8_BAR;
    // phase 4: quad (1,1); stage tile+2's A images (A reads ended at
    // phase 3's closing barrier)
    if (ti + 2 < NT) {
      P8_GLDS1(buf, 0, kt + 2 * BK)
      P8_GLDS1(buf, 1, kt + 2 * BK)
    }
    P8_MFMA(1, 1)
    if (ti + 1 < NT) {
      if (ti + 2 < NT)
        asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    P8_BAR;
  }
#undef P8_GLDS1
#undef P8_FRAG
#undef P8_RD_A
#undef P8_RD_B
#undef P8_MFMA

  // ---- epilogue: D frag (mi, ni): row = mi*16 + lg*4 + r, col = ni*16+l16
  const int mw = m0 + wr * 128, nw = n0 + wc * 64;
#pragma unroll
  for (int mi = 0; mi < 8; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      const int nn = nw + ni * 16 + l16;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int mm = mw + mi * 16 + lg * 4 + r;
        if (mm >= Mr) continue;
        float v = alpha * acc[mi][ni][r];
        if (OUT == BF_OUT_F32_ATOMIC) {
          atomicAdd(&Cf[(int64_t)mm * ldc + nn], v);
        } else {
          if (bias) v += bias[nn];
          if (R) v += bf2f(R[(int64_t)mm * ldc + nn]);
          if (OUT == BF_OUT_F32) {
            if (beta != 0.f) v += beta * Cf[(int64_t)mm * ldc + nn];
            Cf[(int64_t)mm * ldc + nn] = v;
          } else {
            if (beta != 0.f) v += beta * bf2f(Cb[(int64_t)mm * ldc + nn]);
            Cb[(int64_t)mm * ldc + nn] = (__bf16)v;
          }
        }
      }
    }
  }
}

extern "C" int ob_gemm_bf16_nt_8ph(const void* A, const void* B, void* C,
                                   const void* bias, const void* residual,
                                   int64_t M, int64_t N, int64_t K,
                                   int64_t lda, int64_t ldb, int64_t ldc,
                                   int64_t sA1, int64_t sA2, int64_t sB1,
                                   int64_t sB2, int64_t sC1, int64_t sC2,
                                   int64_t n1, int64_t n2, float alpha,
                                   float beta, int out_kind, int splitk,
                                   void* stream, int64_t Mr) {
  if (M % 256 || N % 256 || K % 64)
    return ob_fail("nt_8ph: M,N %% 256, K %% 64 required");
  const int nbm = (int)(M / 256), nbn = (int)(N / 256);
  dim3 grid(nbm * nbn, splitk < 1 ? 1 : splitk, (unsigned)(n1 * n2));
  dim3 block(512);
  static const int swz = [] {
    const char* e = getenv("OB_P8_SWZ");
    return e ? atoi(e) : 0;
  }();
#define OB_P8(OUT_, SWZ_)                                                    \
  k_gemm_bf16_nt_8ph<OUT_, SWZ_><<<grid, block, 0, S(stream)>>>(             \
      (const __bf16*)A, (const __bf16*)B, C, (const float*)bias,             \
      (const __bf16*)residual, (int)M, (int)N, (int)K, lda, ldb, ldc, sA1,   \
      sA2, sB1, sB2, sC1, sC2, (int)n2, alpha, beta, nbn, (int)Mr)
#define OB_P8S(OUT_)                                                         \
  do {                                                                       \
    if (swz == 1) OB_P8(OUT_, 1);                                            \
    else OB_P8(OUT_, 0);                                                     \
  } while (0)
  if (out_kind == BF_OUT_BF16) OB_P8S(BF_OUT_BF16);
  else if (out_kind == BF_OUT_F32) OB_P8S(BF_OUT_F32);
  else OB_P8S(BF_OUT_F32_ATOMIC);
#undef OB_P8S
#undef OB_P8
  OB_LAUNCH_CHECK();
  return 0;
}
