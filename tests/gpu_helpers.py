# Thin torch-tensor wrappers over the standalone C-ABI kernel entry points,
# used by the @gpu parity tests and bench.py's roofline probe.
from __future__ import annotations

import ctypes

import torch


def ptr(t: torch.Tensor | None) -> ctypes.c_void_p:
    if t is None:
        return ctypes.c_void_p(0)
    assert t.is_contiguous()
    return ctypes.c_void_p(t.data_ptr())


def stream() -> ctypes.c_void_p:
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


def gemm(A: torch.Tensor, Bm: torch.Tensor, C: torch.Tensor, *, transA=0,
         transB=0, M, N, K, alpha=1.0, beta=0.0, lda, ldb, ldc,
         sA=(0, 0), sB=(0, 0), sC=(0, 0), n1=1, n2=1, bias=None,
         residual=None, atomic=0, splitk=1):
    from oobleck_amd._ext import check, get_ext
    check(get_ext().ob_gemm_f32(
        transA, transB, M, N, K, alpha, ptr(A), lda, sA[0], sA[1], ptr(Bm),
        ldb, sB[0], sB[1], beta, ptr(C), ldc, sC[0], sC[1], n1, n2, ptr(bias),
        ptr(residual), atomic, splitk, stream()), "gemm")


def layernorm_fwd(x, w, b, eps=1e-5):
    from oobleck_amd._ext import check, get_ext
    rows, H = x.shape
    y = torch.empty_like(x)
    mean = torch.empty(rows, device=x.device)
    rstd = torch.empty(rows, device=x.device)
    check(get_ext().ob_layernorm_fwd_f32(ptr(x), ptr(w), ptr(b), ptr(y),
                                         ptr(mean), ptr(rstd), rows, H, eps,
                                         stream()), "ln_fwd")
    return y, mean, rstd


def layernorm_bwd(x, w, mean, rstd, dy, dx, dw, db, dx_accum=0):
    from oobleck_amd._ext import check, get_ext
    rows, H = x.shape
    check(get_ext().ob_layernorm_bwd_f32(ptr(x), ptr(w), ptr(mean), ptr(rstd),
                                         ptr(dy), ptr(dx), ptr(dw), ptr(db),
                                         rows, H, dx_accum, stream()), "ln_bwd")


def softmax_causal_fwd(scores, scale):
    from oobleck_amd._ext import check, get_ext
    batch, S1, S2 = scores.shape
    assert S1 == S2
    check(get_ext().ob_softmax_causal_fwd_f32(ptr(scores), batch, S1, scale,
                                              stream()), "softmax_fwd")


def softmax_causal_bwd(P, dP):
    from oobleck_amd._ext import check, get_ext
    batch, S1, _ = P.shape
    check(get_ext().ob_softmax_causal_bwd_f32(ptr(P), ptr(dP), batch, S1,
                                              stream()), "softmax_bwd")


def gelu_fwd(u):
    from oobleck_amd._ext import check, get_ext
    g = torch.empty_like(u)
    check(get_ext().ob_gelu_fwd_f32(ptr(u), ptr(g), u.numel(), stream()),
          "gelu_fwd")
    return g


def gelu_bwd(u, dg):
    from oobleck_amd._ext import check, get_ext
    du = torch.empty_like(u)
    check(get_ext().ob_gelu_bwd_f32(ptr(u), ptr(dg), ptr(du), u.numel(),
                                    stream()), "gelu_bwd")
    return du


def colsum(X, db):
    from oobleck_amd._ext import check, get_ext
    M, N = X.shape
    check(get_ext().ob_colsum_f32(ptr(X), ptr(db), M, N, stream()), "colsum")


def adamw(p, g, m, v, step, lr, b1=0.9, b2=0.999, eps=1e-8, wd=0.0):
    from oobleck_amd._ext import check, get_ext
    check(get_ext().ob_adamw_step(ptr(p), ptr(g), ptr(m), ptr(v), p.numel(),
                                  step, lr, b1, b2, eps, wd, stream()),
          "adamw")
