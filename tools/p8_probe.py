# A/B probe for the 8-phase 256^2 NT GEMM (k_gemm_bf16_nt_8ph) vs the
# production dispatch: refcheck at small sizes, multi-seed race screen,
# then within-probe interleaved perf rounds (guide rule 24).
import pathlib
import sys

import torch

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
from tests.gpu_helpers import ptr, stream  # noqa: E402
from tests.test_gpu_bf16 import gemm_bf16  # noqa: E402
from oobleck_amd._ext import check, get_ext  # noqa: E402


def p8(A, B, C, M, N, K, alpha=1.0, beta=0.0, out_kind=0, splitk=1,
       bias=None, residual=None):
    check(get_ext().ob_gemm_bf16_nt_8ph(
        ptr(A), ptr(B), ptr(C), ptr(bias), ptr(residual), M, N, K,
        K, K, N, 0, 0, 0, 0, 0, 0, 1, 1, alpha, beta, out_kind, splitk,
        stream(), M), "p8")


def refcheck():
    for (M, N, K) in [(256, 256, 64), (256, 512, 128), (512, 256, 192),
                      (512, 512, 960)]:
        g = torch.Generator().manual_seed(M + K)
        A = (torch.randn(M, K, generator=g) * 0.5).cuda().bfloat16()
        B = (torch.randn(N, K, generator=g) * 0.5).cuda().bfloat16()
        C = torch.empty(M, N, device="cuda", dtype=torch.bfloat16)
        p8(A, B, C, M, N, K)
        ref = (A.float() @ B.float().T)
        err = (C.float() - ref).norm() / ref.norm()
        assert err < 2e-2, (M, N, K, err.item())
        # f32 out + bias + beta
        Cf = torch.ones(M, N, device="cuda", dtype=torch.float32)
        bias = torch.randn(N, device="cuda", dtype=torch.float32)
        p8(A, B, Cf, M, N, K, alpha=0.5, beta=2.0, out_kind=1, bias=bias)
        reff = 0.5 * ref + bias + 2.0
        err = (Cf - reff).norm() / reff.norm()
        assert err < 2e-2, ("f32", M, N, K, err.item())
        # atomic accum with splitk
        Ca = torch.zeros(M, N, device="cuda", dtype=torch.float32)
        sk = 2 if K >= 128 else 1
        p8(A, B, Ca, M, N, K, out_kind=2, splitk=sk)
        err = (Ca - ref).norm() / ref.norm()
        assert err < 2e-2, ("atomic", M, N, K, err.item())
    print("refcheck OK")


def race_screen():
    M, N, K = 512, 512, 4096
    for seed in range(8):
        g = torch.Generator().manual_seed(100 + seed)
        A = (torch.randn(M, K, generator=g) * 0.5).cuda().bfloat16()
        B = (torch.randn(N, K, generator=g) * 0.5).cuda().bfloat16()
        C = torch.empty(M, N, device="cuda", dtype=torch.bfloat16)
        p8(A, B, C, M, N, K)
        C2 = torch.empty_like(C)
        gemm_bf16(A, B, C2, transB=1, M=M, N=N, K=K, lda=K, ldb=K, ldc=N)
        if not torch.equal(C, C2):
            d = (C.float() - C2.float()).abs()
            rel = d.norm() / C2.float().norm()
            print(f"seed {seed}: mismatch vs prod kernel rel {rel:.2e} "
                  f"max {d.max():.3e} (bf16 rounding-order ok if tiny)")
            assert rel < 5e-3, rel
    print("race screen OK")


SHAPES = [("fc", 8192, 3072, 768), ("qkv", 8192, 2304, 768),
          ("dX_fc", 8192, 768, 3072), ("lmhead", 8192, 50432, 768),
          ("cube4k", 4096, 4096, 4096)]


def perf():
    ev = [torch.cuda.Event(True) for _ in range(2)]
    res = {}
    for name, M, N, K in SHAPES:
        g = torch.Generator().manual_seed(7)
        A = (torch.randn(M, K, generator=g)).cuda().bfloat16()
        B = (torch.randn(N, K, generator=g)).cuda().bfloat16()
        C = torch.empty(M, N, device="cuda", dtype=torch.bfloat16)
        variants = {
            "prod": lambda: gemm_bf16(A, B, C, transB=1, M=M, N=N, K=K,
                                      lda=K, ldb=K, ldc=N),
            "8ph": lambda: p8(A, B, C, M, N, K),
        }
        for fn in variants.values():
            for _ in range(3):
                fn()
        torch.cuda.synchronize()
        best = {k: float("inf") for k in variants}
        for _ in range(5):  # interleaved rounds
            for k, fn in variants.items():
                ev[0].record()
                for _ in range(10):
                    fn()
                ev[1].record()
                torch.cuda.synchronize()
                best[k] = min(best[k], ev[0].elapsed_time(ev[1]) / 10)
        tf = {k: 2.0 * M * N * K / v / 1e9 for k, v in best.items()}
        res[name] = {k: round(v, 1) for k, v in tf.items()}
        print(f"{name:8s} M{M} N{N} K{K}: " +
              "  ".join(f"{k} {best[k]:.3f} ms = {tf[k]:.0f} TF"
                        for k in variants))
    import json
    print(json.dumps(res))


if __name__ == "__main__":
    refcheck()
    race_screen()
    perf()
