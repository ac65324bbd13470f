# bf16 GEMM probe for rocprofv3 PMC capture: N reps of the hot TB (fwd fc)
# shape only.
import pathlib
import sys

import torch

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
from tests.test_gpu_bf16 import gemm_bf16  # noqa: E402

M, N, K = 8192, 3072, 768
A = torch.randn(M, K, device="cuda").bfloat16()
B = torch.randn(N, K, device="cuda").bfloat16()
C = torch.empty(M, N, device="cuda", dtype=torch.bfloat16)
for _ in range(3):
    gemm_bf16(A, B, C, transB=1, M=M, N=N, K=K, lda=K, ldb=K, ldc=N)
torch.cuda.synchronize()
st, en = torch.cuda.Event(True), torch.cuda.Event(True)
st.record()
for _ in range(30):
    gemm_bf16(A, B, C, transB=1, M=M, N=N, K=K, lda=K, ldb=K, ldc=N)
en.record()
torch.cuda.synchronize()
ms = st.elapsed_time(en) / 30
print(f"TB fc: {ms:.3f} ms  {2.0*M*N*K/ms/1e9:.0f} TF")
