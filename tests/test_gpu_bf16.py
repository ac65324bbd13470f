# @gpu parity for the bf16 MFMA path (§8 f4): the bf16 GEMM against a
# torch fp32 reference computed from the SAME bf16-rounded inputs (our
# kernel accumulates in fp32, so the comparison tolerance covers only the
# bf16 input rounding + fp32 sum-order effects).
import ctypes

import pytest
import torch

from tests.gpu_helpers import ptr, stream

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs MI355X")
DEV = "cuda:0"


def gemm_bf16(A, Bm, C, *, transA=0, transB=0, M, N, K, alpha=1.0, beta=0.0,
              lda, ldb, ldc, sA=(0, 0), sB=(0, 0), sC=(0, 0), n1=1, n2=1,
              bias=None, residual=None, out_kind=0, splitk=1):
    from oobleck_amd._ext import check, get_ext
    check(get_ext().ob_gemm_bf16(
        transA, transB, M, N, K, alpha, ptr(A), lda, sA[0], sA[1], ptr(Bm),
        ldb, sB[0], sB[1], beta, ptr(C), ldc, sC[0], sC[1], n1, n2, ptr(bias),
        ptr(residual), out_kind, splitk, stream()), "gemm_bf16")


def rb(*shape, seed=0, scale=1.0):
    g = torch.Generator().manual_seed(seed)
    return (torch.randn(*shape, generator=g) * scale).to(DEV).bfloat16()


@requires_gpu
@pytest.mark.parametrize("tA,tB", [(0, 0), (0, 1), (1, 0), (1, 1)])
@pytest.mark.parametrize("M,N,K", [(128, 128, 32), (256, 384, 128),
                                   (200, 136, 72), (96, 64, 1024)])
def test_gemm_bf16_parity(tA, tB, M, N, K):
    # asymmetric operands (transpose-detecting, guide G9)
    A = rb(*(K, M) if tA else (M, K), seed=1)
    B = rb(*(N, K) if tB else (K, N), seed=2)
    lda, ldb = A.shape[1], B.shape[1]
    if lda % 8 or ldb % 8:
        pytest.skip("unaligned ld")
    C = torch.empty(M, N, device=DEV, dtype=torch.bfloat16)
    gemm_bf16(A, B, C, transA=tA, transB=tB, M=M, N=N, K=K,
              lda=lda, ldb=ldb, ldc=N)
    Aop = (A.t() if tA else A).float()
    Bop = (B.t() if tB else B).float()
    ref = Aop @ Bop
    torch.cuda.synchronize()
    torch.testing.assert_close(C.float(), ref, rtol=2e-2, atol=2e-2)
    # tighter check in fp32-out mode (removes the output rounding)
    C32 = torch.empty(M, N, device=DEV, dtype=torch.float32)
    gemm_bf16(A, B, C32, transA=tA, transB=tB, M=M, N=N, K=K,
              lda=lda, ldb=ldb, ldc=N, out_kind=1)
    torch.cuda.synchronize()
    torch.testing.assert_close(C32, ref, rtol=1e-3, atol=1e-3)


@requires_gpu
def test_gemm_bf16_bias_residual_beta():
    M, N, K = 256, 256, 128
    A, B = rb(M, K, seed=3), rb(K, N, seed=4)
    bias = torch.randn(N, device=DEV)
    R = rb(M, N, seed=5)
    C = rb(M, N, seed=6)
    C0 = C.clone()
    gemm_bf16(A, B, C, M=M, N=N, K=K, lda=K, ldb=N, ldc=N, alpha=0.5,
              beta=1.0, bias=bias, residual=R)
    ref = 0.5 * (A.float() @ B.float()) + bias + R.float() + C0.float()
    torch.cuda.synchronize()
    torch.testing.assert_close(C.float(), ref, rtol=2e-2, atol=2e-2)


@requires_gpu
def test_gemm_bf16_atomic_splitk_f32out():
    M, N, K = 128, 384, 4096
    A = rb(K, M, seed=7)  # TA (the dW pattern)
    B = rb(K, N, seed=8)
    C = torch.zeros(M, N, device=DEV, dtype=torch.float32)
    gemm_bf16(A, B, C, transA=1, M=M, N=N, K=K, lda=M, ldb=N, ldc=N,
              out_kind=2, splitk=4)
    ref = A.t().float() @ B.float()
    torch.cuda.synchronize()
    torch.testing.assert_close(C, ref, rtol=2e-3, atol=2e-2)


@requires_gpu
def test_casts_roundtrip():
    from oobleck_amd._ext import check, get_ext
    x = torch.randn(1000, device=DEV)
    y = torch.empty(1000, device=DEV, dtype=torch.bfloat16)
    check(get_ext().ob_f32_to_bf16(ptr(x), ptr(y), 1000, stream()), "cast")
    torch.cuda.synchronize()
    torch.testing.assert_close(y, x.bfloat16())
    z = torch.empty(1000, device=DEV)
    check(get_ext().ob_bf16_to_f32(ptr(y), ptr(z), 1000, stream()), "cast2")
    torch.cuda.synchronize()
    torch.testing.assert_close(z, y.float())
    # transposed cast
    w = torch.randn(96, 288, device=DEV)
    wt = torch.empty(288, 96, device=DEV, dtype=torch.bfloat16)
    check(get_ext().ob_f32_to_bf16_t(ptr(w), ptr(wt), 96, 288, stream()),
          "cast_t")
    torch.cuda.synchronize()
    torch.testing.assert_close(wt, w.t().contiguous().bfloat16())


@requires_gpu
def test_gemm_bf16_batched_strided():
    Bn, S, hd = 4, 128, 64
    q = rb(Bn, S, hd, seed=9)
    k = rb(Bn, S, hd, seed=10)
    out = torch.empty(Bn, S, S, device=DEV, dtype=torch.bfloat16)
    gemm_bf16(q, k, out, transB=1, M=S, N=S, K=hd, lda=hd, ldb=hd, ldc=S,
              sA=(S * hd, 0), sB=(S * hd, 0), sC=(S * S, 0), n1=Bn, n2=1)
    ref = torch.matmul(q.float(), k.float().transpose(-1, -2))
    torch.cuda.synchronize()
    torch.testing.assert_close(out.float(), ref, rtol=2e-2, atol=2e-2)
