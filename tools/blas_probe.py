# ceiling probe: hipBLASLt (via torch.matmul) on the step's hot GEMM
# shapes, same interleaved-round discipline as p8_probe.
import torch

SHAPES = [("fc", 8192, 3072, 768), ("qkv", 8192, 2304, 768),
          ("dX_fc", 8192, 768, 3072), ("lmhead", 8192, 50432, 768),
          ("dW_fc", 768, 3072, 8192), ("cube4k", 4096, 4096, 4096)]
ev = [torch.cuda.Event(True) for _ in range(2)]
for name, M, N, K in SHAPES:
    A = torch.randn(M, K, device="cuda").bfloat16()
    B = torch.randn(N, K, device="cuda").bfloat16().T  # [K,N] view, NT-ish
    C = torch.empty(M, N, device="cuda", dtype=torch.bfloat16)
    for _ in range(3):
        torch.matmul(A, B, out=C)
    torch.cuda.synchronize()
    best = float("inf")
    for _ in range(5):
        ev[0].record()
        for _ in range(10):
            torch.matmul(A, B, out=C)
        ev[1].record()
        torch.cuda.synchronize()
        best = min(best, ev[0].elapsed_time(ev[1]) / 10)
    print(f"{name:8s} {best:.3f} ms  {2.0*M*N*K/best/1e9:.0f} TF")
