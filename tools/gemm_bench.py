# Times every hot GEMM shape of the GPT-2-small step (HIP events, 20 reps)
# and prints achieved TFLOP/s per shape — the per-shape evidence behind the
# step-time breakdown in DESIGN.md.  Run on the GPU box:
#   python tools/gemm_bench.py
from __future__ import annotations

import json
import pathlib
import sys

import torch

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
from tests.gpu_helpers import gemm  # noqa: E402

B, S, H, nh, V = 8, 1024, 768, 12, 50257
hd = H // nh
M = B * S

SHAPES = [
    # name, kwargs-builder
    ("fwd_qkv NN", dict(M=M, N=3 * H, K=H)),
    ("fwd_fc NN", dict(M=M, N=4 * H, K=H)),
    ("fwd_mlpproj NN", dict(M=M, N=H, K=4 * H)),
    ("fwd_attnproj NN", dict(M=M, N=H, K=H)),
    ("fwd_lmhead NT", dict(M=M, N=V, K=H, transB=1)),
    ("bwd_dlnout NN K=V", dict(M=M, N=H, K=V)),
    ("bwd_dX_fc NT", dict(M=M, N=4 * H, K=H, transB=1)),
    ("bwd_dW_fc TN splitk", dict(M=H, N=4 * H, K=M, transA=1, atomic=1,
                                 splitk=4)),
    ("bwd_dW_lmhead TN", dict(M=V, N=H, K=M, transA=1, atomic=1)),
    ("attn_scores NT b96", dict(M=S, N=S, K=hd, transB=1, batch=B * nh)),
    ("attn_PV NN b96", dict(M=S, N=hd, K=S, batch=B * nh)),
    ("attn_dV TN b96", dict(M=S, N=hd, K=S, transA=1, batch=B * nh)),
    # diagnostics: the guide's fp32 reference point (122 TF untuned @4096^3)
    ("diag_4096cubed NN", dict(M=4096, N=4096, K=4096)),
    ("diag_sqK3072 NN", dict(M=8192, N=3072, K=3072)),
    ("diag_xl_fc NN", dict(M=2048, N=6400, K=1600)),
]


def run_shape(name, spec, reps=20):
    Mm, Nn, Kk = spec["M"], spec["N"], spec["K"]
    tA, tB = spec.get("transA", 0), spec.get("transB", 0)
    batch = spec.get("batch", 1)
    at, sk = spec.get("atomic", 0), spec.get("splitk", 1)
    A = torch.randn(batch, *( (Kk, Mm) if tA else (Mm, Kk) ), device="cuda")
    Bm = torch.randn(batch, *( (Nn, Kk) if tB else (Kk, Nn) ), device="cuda")
    C = torch.zeros(batch, Mm, Nn, device="cuda")
    kw = dict(transA=tA, transB=tB, M=Mm, N=Nn, K=Kk,
              lda=A.shape[2], ldb=Bm.shape[2], ldc=Nn,
              sA=(A.shape[1] * A.shape[2], 0), sB=(Bm.shape[1] * Bm.shape[2], 0),
              sC=(Mm * Nn, 0), n1=batch, n2=1, atomic=at, splitk=sk)
    for _ in range(3):
        gemm(A, Bm, C, **kw)
    torch.cuda.synchronize()
    st, en = torch.cuda.Event(True), torch.cuda.Event(True)
    st.record()
    for _ in range(reps):
        gemm(A, Bm, C, **kw)
    en.record()
    torch.cuda.synchronize()
    avg_ms = st.elapsed_time(en) / reps
    tf = 2.0 * batch * Mm * Nn * Kk / (avg_ms * 1e-3) / 1e12
    return avg_ms, tf


def main():
    out = {}
    for name, spec in SHAPES:
        avg_ms, tf = run_shape(name, spec)
        out[name] = {"avg_ms": round(avg_ms, 3), "tflops": round(tf, 1)}
        print(f"{name:24s} {avg_ms:8.3f} ms  {tf:7.1f} TF")
    print(json.dumps(out))


if __name__ == "__main__":
    main()
