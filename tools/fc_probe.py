# fc-family GEMM variant sweep at K=768: measure every
# hand-written NT kernel (OB_BF16_FORCE hook) and hipBLASLt on the step's
# plain-GEMM shapes, so the retake target is chosen from data.
import json
import os
import pathlib
import sys

import torch

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
from tests.test_gpu_bf16 import gemm_bf16  # noqa: E402

SHAPES = [
    ("fc_fwd", 8192, 3072, 768),
    ("qkv_fwd", 8192, 2304, 768),
    ("proj_fwd", 8192, 768, 3072),
    ("dX_fc", 8192, 768, 3072),
    ("dX_mlp", 8192, 3072, 768),
]
VARIANTS = ["lib", "glds", "n128", "n256", "8ph"]


def run_one(name, M, N, K, variant, reps=30):
    A = torch.randn(M, K, device="cuda").bfloat16()
    B = torch.randn(N, K, device="cuda").bfloat16()
    C = torch.empty(M, N, device="cuda", dtype=torch.bfloat16)
    if variant == "lib":
        os.environ.pop("OB_BF16_FORCE", None)
    else:
        os.environ["OB_BF16_FORCE"] = variant
    def call():
        gemm_bf16(A, B, C, transB=1, M=M, N=N, K=K, lda=K, ldb=K, ldc=N)
    for _ in range(3):
        call()
    torch.cuda.synchronize()
    st, en = torch.cuda.Event(True), torch.cuda.Event(True)
    best = float("inf")
    for _ in range(3):
        st.record()
        for _ in range(reps):
            call()
        en.record()
        torch.cuda.synchronize()
        best = min(best, st.elapsed_time(en) / reps)
    os.environ.pop("OB_BF16_FORCE", None)
    return best, 2.0 * M * N * K / best / 1e9


def main():
    out = {}
    for name, M, N, K in SHAPES:
        row = {}
        for v in VARIANTS:
            try:
                ms, tf = run_one(name, M, N, K, v)
                row[v] = round(tf, 1)
            except Exception as e:  # noqa: BLE001
                row[v] = f"err: {e}"[:60]
        out[name] = row
        print(name, row, flush=True)
    print(json.dumps(out))


if __name__ == "__main__":
    main()
