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

// direct staging: img[row][k] = src[row][k], k contiguous in memory.
// 256 threads; thread t covers row = t/2, k-halfs (t%2)*8 and +16.
template <bool EDGE>
__device__ __forceinline__ void bf_stage_direct(__bf16* img,
                                                const __bf16* src, int64_t ld,
                                                int rmax, int kmax) {
  const int r = threadIdx.x >> 1;  // 0..127
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    const int kb = (threadIdx.x & 1) * 8 + it * 16;
    uint4 v = {0, 0, 0, 0};
    if (!EDGE || (r < rmax && kb + 7 < kmax)) {
      v = *reinterpret_cast<const uint4*>(src + (int64_t)r * ld + kb);
    } else if (r < rmax && kb < kmax) {
      v = bf_load8<EDGE>(src + (int64_t)r * ld + kb, kmax - kb);
    }
    *reinterpret_cast<uint4*>(img + r * BF_LDS_K + kb) = v;
  }
}

// transpose staging: img[row][k] = src[k][row] (src k-major, ld = src row
// stride).  Reads 16 B along the row dim (coalesced), writes 8 scattered
// 2-byte LDS stores.
template <bool EDGE>
__device__ __forceinline__ void bf_stage_transpose(__bf16* img,
                                                   const __bf16* src,
                                                   int64_t ld, int rmax,
                                                   int kmax) {
  const int rb = (threadIdx.x & 15) * 8;
  const int k0 = threadIdx.x >> 4;  // 0..15
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    const int k = k0 + it * 16;
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

template <bool TA, bool TB, int OUT, bool EDGE>
__global__ __launch_bounds__(256, 2) void k_gemm_bf16(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    void* __restrict__ Cv, const float* __restrict__ bias,
    const __bf16* __restrict__ R, int M, int N, int K, int64_t lda,
    int64_t ldb, int64_t ldc, int64_t sA1, int64_t sA2, int64_t sB1,
    int64_t sB2, int64_t sC1, int64_t sC2, int n2, float alpha, float beta,
    int nbn) {
  __shared__ __bf16 As[2][BF_BM * BF_LDS_K];
  __shared__ __bf16 Bs[2][BF_BN * BF_LDS_K];

  const int tile = blockIdx.x;
  const int bm = tile / nbn, bn = tile % nbn;
  const int m0 = bm * BF_BM, n0 = bn * BF_BN;

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
      bf_stage_direct<EDGE>(Bs[BUF], B + (int64_t)n0 * ldb + (KT), ldb,       \
                            min(N - n0, BF_BN), kmax_);                       \
    else                                                                      \
      bf_stage_transpose<EDGE>(Bs[BUF], B + (int64_t)(KT)*ldb + n0, ldb,      \
                               min(N - n0, BF_BN), kmax_);                    \
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
        Bs[BUF] + (wc * 64 + il) * BF_LDS_K + kb);                            \
    const bf16x8 b1 = *reinterpret_cast<const bf16x8*>(                       \
        Bs[BUF] + (wc * 64 + 32 + il) * BF_LDS_K + kb);                       \
    acc00 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b0, acc00, 0, 0, 0);  \
    acc01 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b1, acc01, 0, 0, 0);  \
    acc10 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b0, acc10, 0, 0, 0);  \
    acc11 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b1, acc11, 0, 0, 0);  \
  }

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

  const int mw = m0 + wr * 64, nw = n0 + wc * 64;
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
  OB_BF_EPI(acc01, 0, 1)
  OB_BF_EPI(acc10, 1, 0)
  OB_BF_EPI(acc11, 1, 1)
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
  const int nbm = (int)((M + BF_BM - 1) / BF_BM);
  const int nbn = (int)((N + BF_BN - 1) / BF_BN);
  const bool edge = (M % BF_BM) || (N % BF_BN) || (K % BF_BK);
  dim3 grid(nbm * nbn, splitk, (unsigned)(n1 * n2));
  dim3 block(256);
#define OB_BFG_L(TA_, TB_, OUT_, ED_)                                        \
  k_gemm_bf16<TA_, TB_, OUT_, ED_><<<grid, block, 0, S(stream)>>>(           \
      (const __bf16*)A, (const __bf16*)B, C, (const float*)bias,             \
      (const __bf16*)residual, (int)M, (int)N, (int)K, lda, ldb, ldc,        \
      strideA1, strideA2, strideB1, strideB2, strideC1, strideC2, (int)n2,   \
      alpha, beta, nbn)
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
  const int sel = (transA ? 2 : 0) | (transB ? 1 : 0);
  switch (sel) {
    case 0: OB_BFG_OUT(false, false); break;
    case 1: OB_BFG_OUT(false, true); break;
    case 2: OB_BFG_OUT(true, false); break;
    case 3: OB_BFG_OUT(true, true); break;
  }
#undef OB_BFG_OUT
#undef OB_BFG_L
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

// transposed cast: y[o][i] = x[i][o] for x [rows=in, cols=out]
__global__ __launch_bounds__(256) void k_f32_to_bf16_t(const float* __restrict__ x,
                                                       __bf16* __restrict__ y,
                                                       int64_t rows,
                                                       int64_t cols) {
  for (int64_t idx = (int64_t)blockIdx.x * 256 + threadIdx.x; idx < rows * cols;
       idx += (int64_t)gridDim.x * 256) {
    const int64_t r = idx / cols, c = idx % cols;
    y[c * rows + r] = (__bf16)x[r * cols + c];
  }
}

__global__ __launch_bounds__(256) void k_bf16_to_f32(const __bf16* __restrict__ x,
                                                     float* __restrict__ y,
                                                     int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * 256 + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * 256)
    y[i] = (float)x[i];
}

static inline int64_t bmin64(int64_t a, int64_t b) { return a < b ? a : b; }

extern "C" int ob_f32_to_bf16(const void* x, void* y, int64_t n, void* stream) {
  k_f32_to_bf16<<<(int)bmin64((n + 255) / 256, 4096), 256, 0, S(stream)>>>(
      (const float*)x, (__bf16*)y, n);
  OB_LAUNCH_CHECK();
  return 0;
}
extern "C" int ob_f32_to_bf16_t(const void* x, void* y, int64_t rows,
                                int64_t cols, void* stream) {
  k_f32_to_bf16_t<<<(int)bmin64((rows * cols + 255) / 256, 4096), 256, 0,
                    S(stream)>>>((const float*)x, (__bf16*)y, rows, cols);
  OB_LAUNCH_CHECK();
  return 0;
}
extern "C" int ob_bf16_to_f32(const void* x, void* y, int64_t n, void* stream) {
  k_bf16_to_f32<<<(int)bmin64((n + 255) / 256, 4096), 256, 0, S(stream)>>>(
      (const __bf16*)x, (float*)y, n);
  OB_LAUNCH_CHECK();
  return 0;
}
