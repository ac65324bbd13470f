# @gpu per-kernel parity: each HIP kernel against a plain torch fp32
# reference of the same op (torch-on-GPU is test infrastructure here; the
# product path never calls it).  Tolerances: fp32, stated per test;
# atomically-accumulated outputs (split-K dW, LN dw/db, embedding scatter)
# get a looser atol for summation-order nondeterminism.
import math

import pytest
import torch

from tests import gpu_helpers as gh

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs MI355X")

DEV = "cuda:0"


def rt(*shape, seed=0, scale=1.0):
    g = torch.Generator(device="cpu").manual_seed(seed)
    return (torch.randn(*shape, generator=g) * scale).to(DEV)


@requires_gpu
@pytest.mark.parametrize("tA,tB", [(0, 0), (0, 1), (1, 0), (1, 1)])
@pytest.mark.parametrize("M,N,K", [(128, 128, 32), (257, 130, 100),
                                   (512, 768, 768), (96, 64, 1024)])
def test_gemm_trans_variants(tA, tB, M, N, K):
    A = rt(*(K, M) if tA else (M, K), seed=1)
    B = rt(*(N, K) if tB else (K, N), seed=2)
    C = torch.empty(M, N, device=DEV)
    gh.gemm(A, B, C, transA=tA, transB=tB, M=M, N=N, K=K,
            lda=A.shape[1], ldb=B.shape[1], ldc=N)
    Aop = A.t() if tA else A
    Bop = B.t() if tB else B
    ref = Aop @ Bop
    torch.cuda.synchronize()
    torch.testing.assert_close(C, ref, rtol=1e-5, atol=1e-4)


@requires_gpu
def test_gemm_alpha_beta_bias_residual():
    M, N, K = 200, 192, 96
    A, B = rt(M, K, seed=3), rt(K, N, seed=4)
    bias, R = rt(N, seed=5), rt(M, N, seed=6)
    C = rt(M, N, seed=7)
    C0 = C.clone()
    gh.gemm(A, B, C, M=M, N=N, K=K, lda=K, ldb=N, ldc=N, alpha=0.5, beta=1.0,
            bias=bias, residual=R)
    ref = 0.5 * (A @ B) + bias + R + C0
    torch.cuda.synchronize()
    torch.testing.assert_close(C, ref, rtol=1e-5, atol=1e-4)


@requires_gpu
def test_gemm_batched_two_level_strides():
    # attention-shaped: z = (b, h), strided slices of a [B,S,3H] buffer
    Bn, nh, S, hd = 3, 4, 64, 32
    H = nh * hd
    qkv = rt(Bn, S, 3 * H, seed=8)
    q = qkv[..., :H].view(Bn, S, nh, hd).permute(0, 2, 1, 3)
    k = qkv[..., H:2 * H].view(Bn, S, nh, hd).permute(0, 2, 1, 3)
    qkv_k = qkv.flatten()[H:]  # base pointer at the K slice (stays a view)
    out = torch.empty(Bn * nh, S, S, device=DEV)
    gh.gemm(qkv, qkv_k, out, transB=1, M=S, N=S, K=hd, lda=3 * H, ldb=3 * H,
            ldc=S, sA=(S * 3 * H, hd), sB=(S * 3 * H, hd),
            sC=(nh * S * S, S * S), n1=Bn, n2=nh)
    ref = torch.matmul(q, k.transpose(-1, -2)).reshape(Bn * nh, S, S)
    torch.cuda.synchronize()
    torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-4)


@requires_gpu
def test_gemm_atomic_splitk():
    M, N, K = 96, 288, 2048
    A = rt(K, M, seed=9)   # stored [K,M]: transA (the dW pattern)
    B = rt(K, N, seed=10)
    C = torch.zeros(M, N, device=DEV)
    gh.gemm(A, B, C, transA=1, M=M, N=N, K=K, lda=M, ldb=N, ldc=N,
            atomic=1, splitk=4)
    ref = A.t() @ B
    torch.cuda.synchronize()
    torch.testing.assert_close(C, ref, rtol=1e-4, atol=5e-4)


@requires_gpu
@pytest.mark.parametrize("rows,H", [(512, 768), (333, 1600), (64, 96)])
def test_layernorm_fwd_bwd(rows, H):
    x = rt(rows, H, seed=11, scale=2.0)
    w = rt(H, seed=12) + 1.0
    b = rt(H, seed=13)
    y, mean, rstd = gh.layernorm_fwd(x, w, b)
    xr = x.detach().clone().requires_grad_(True)
    wr = w.detach().clone().requires_grad_(True)
    br = b.detach().clone().requires_grad_(True)
    yr = torch.nn.functional.layer_norm(xr, (H,), wr, br, eps=1e-5)
    torch.cuda.synchronize()
    torch.testing.assert_close(y, yr, rtol=1e-5, atol=1e-5)
    dy = rt(rows, H, seed=14)
    yr.backward(dy)
    dx = torch.empty_like(x)
    dw = torch.zeros(H, device=DEV)
    db = torch.zeros(H, device=DEV)
    gh.layernorm_bwd(x, w, mean, rstd, dy, dx, dw, db)
    torch.cuda.synchronize()
    torch.testing.assert_close(dx, xr.grad, rtol=1e-4, atol=1e-5)
    torch.testing.assert_close(dw, wr.grad, rtol=1e-4, atol=1e-3)
    torch.testing.assert_close(db, br.grad, rtol=1e-4, atol=1e-3)
    # accumulate flag
    dx2 = torch.ones_like(x)
    gh.layernorm_bwd(x, w, mean, rstd, dy, dx2, dw, db, dx_accum=1)
    torch.cuda.synchronize()
    torch.testing.assert_close(dx2, xr.grad + 1.0, rtol=1e-4, atol=1e-5)


@requires_gpu
@pytest.mark.parametrize("batch,S", [(8, 128), (3, 1024), (2, 1000)])
def test_softmax_causal_fwd_bwd(batch, S):
    scale = 1.0 / math.sqrt(64.0)
    scores = rt(batch, S, S, seed=15, scale=3.0)
    P = scores.clone()
    gh.softmax_causal_fwd(P, scale)
    mask = torch.tril(torch.ones(S, S, dtype=torch.bool, device=DEV))
    sr = scores.detach().clone().requires_grad_(True)
    w = torch.where(mask, sr * scale,
                    torch.tensor(torch.finfo(torch.float32).min, device=DEV))
    Pr = torch.softmax(w, dim=-1)
    torch.cuda.synchronize()
    torch.testing.assert_close(P, Pr, rtol=1e-5, atol=1e-6)
    dP = rt(batch, S, S, seed=16)
    Pr.backward(dP)
    dS = dP.clone()
    gh.softmax_causal_bwd(P, dS)
    # reference grad is w.r.t. scores*scale BEFORE our kernel's scale split:
    # our dS is d/d(w) (pre-softmax, post-scale+mask); torch's sr.grad is
    # d/d(scores) = scale * masked(dS).  Compare on the masked+scaled form.
    torch.cuda.synchronize()
    torch.testing.assert_close(dS * scale,
                               torch.where(mask, sr.grad, torch.zeros((), device=DEV)),
                               rtol=1e-4, atol=1e-6)


@requires_gpu
def test_gelu_fwd_bwd():
    u = rt(1 << 20, seed=17, scale=2.0)
    g = gh.gelu_fwd(u)
    ur = u.detach().clone().requires_grad_(True)
    gr = 0.5 * ur * (1.0 + torch.tanh(math.sqrt(2.0 / math.pi) *
                                      (ur + 0.044715 * ur ** 3)))
    torch.cuda.synchronize()
    torch.testing.assert_close(g, gr, rtol=1e-5, atol=1e-6)
    dg = rt(1 << 20, seed=18)
    gr.backward(dg)
    du = gh.gelu_bwd(u, dg)
    torch.cuda.synchronize()
    torch.testing.assert_close(du, ur.grad, rtol=1e-4, atol=1e-5)


@requires_gpu
def test_colsum():
    M, N = 4096, 768
    X = rt(M, N, seed=19)
    db = torch.ones(N, device=DEV)
    gh.colsum(X, db)
    torch.cuda.synchronize()
    torch.testing.assert_close(db, X.sum(0) + 1.0, rtol=1e-4, atol=1e-3)


@requires_gpu
def test_adamw_matches_golden(golden_dir):
    import numpy as np
    z = np.load(golden_dir / "adamw.npz")
    lr, b1, b2, eps, wd = [float(x) for x in z["hyper"]]
    p = torch.from_numpy(z["p0"]).to(DEV)
    m = torch.zeros_like(p)
    v = torch.zeros_like(p)
    grads = torch.from_numpy(z["grads"]).to(DEV)
    for step in range(1, grads.shape[0] + 1):
        gh.adamw(p, grads[step - 1].contiguous(), m, v, step, lr, b1, b2, eps, wd)
    torch.cuda.synchronize()
    torch.testing.assert_close(p.cpu(), torch.from_numpy(z["p_final"]),
                               rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(m.cpu(), torch.from_numpy(z["m_final"]),
                               rtol=1e-5, atol=1e-6)
    torch.testing.assert_close(v.cpu(), torch.from_numpy(z["v_final"]),
                               rtol=1e-5, atol=1e-7)
