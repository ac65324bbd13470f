# PMC capture target: prod (nt256) vs 8ph at cube4k, 10 reps each.
import pathlib
import sys

import torch

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
from tests.test_gpu_bf16 import gemm_bf16  # noqa: E402
from tools.p8_probe import p8  # noqa: E402

M = N = K = 4096
g = torch.Generator().manual_seed(7)
A = (torch.randn(M, K, generator=g)).cuda().bfloat16()
B = (torch.randn(N, K, generator=g)).cuda().bfloat16()
C = torch.empty(M, N, device="cuda", dtype=torch.bfloat16)
for _ in range(2):
    gemm_bf16(A, B, C, transB=1, M=M, N=N, K=K, lda=K, ldb=K, ldc=N)
    p8(A, B, C, M, N, K)
torch.cuda.synchronize()
for _ in range(10):
    gemm_bf16(A, B, C, transB=1, M=M, N=N, K=K, lda=K, ldb=K, ldc=N)
for _ in range(10):
    p8(A, B, C, M, N, K)
torch.cuda.synchronize()
print("done")
