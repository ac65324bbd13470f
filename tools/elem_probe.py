# HBM-bound kernel probe at the step's shapes: measured GB/s vs the ~6.3
# TB/s achievable HBM roofline (MI355X_MICROARCH.md §HBM).  Identifies
# which non-GEMM families still have headroom.
import ctypes
import pathlib
import sys

import torch

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
from oobleck_amd._ext import check, get_ext  # noqa: E402

DEV = "cuda"
R, H = 8192, 768


def ptr(t):
    return ctypes.c_void_p(0 if t is None else t.data_ptr())


def stream():
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


_SEQ = ["ln_fwd", "ln_bwd", "gelu_fwd", "gelu_bwd", "colsum_3H", "ce_fwd",
        "ce_bwd", "flash_dsum", "transpose_b"]
_IDX = [0]


def timeit(fn, reps=50, name=""):
    label = _SEQ[_IDX[0]] if _IDX[0] < len(_SEQ) else str(_IDX[0])
    _IDX[0] += 1
    print(f"-> {label}", flush=True)
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    st, en = torch.cuda.Event(True), torch.cuda.Event(True)
    best = float("inf")
    for _ in range(3):
        st.record()
        for _ in range(reps):
            fn()
        en.record()
        torch.cuda.synchronize()
        best = min(best, st.elapsed_time(en) / reps)
    return best


def report(name, ms, bytes_moved):
    print(f"{name:18s} {ms*1e3:8.1f} us  {bytes_moved/ms/1e6:7.0f} GB/s",
          flush=True)


ext = get_ext()

# LayerNorm fwd bf16: read x (bf16) + write y (bf16) + stats
x = torch.randn(R, H, device=DEV).bfloat16()
w = torch.randn(H, device=DEV)
b = torch.randn(H, device=DEV)
y = torch.empty_like(x)
mean = torch.empty(R, device=DEV)
rstd = torch.empty(R, device=DEV)
ms = timeit(lambda: check(ext.ob_layernorm_fwd_bf16(
    ptr(x), ptr(w), ptr(b), ptr(y), ptr(mean), ptr(rstd), R, H,
    ctypes.c_float(1e-5), stream())))
report("ln_fwd", ms, R * H * 2 * 2 + R * 8)

# LayerNorm bwd bf16: read x,dy (bf16) + write dx (bf16) + dw/db atomics
dy = torch.randn(R, H, device=DEV).bfloat16()
dx = torch.empty_like(x)
dw = torch.zeros(H, device=DEV)
db = torch.zeros(H, device=DEV)
ms = timeit(lambda: check(ext.ob_layernorm_bwd_bf16(
    ptr(x), ptr(w), ptr(mean), ptr(rstd), ptr(dy), ptr(dx), ptr(dw),
    ptr(db), R, H, 0, stream())))
report("ln_bwd", ms, R * H * 2 * 3 + R * 8)

# gelu fwd/bwd at the 4H width
n4 = R * 4 * H
u = torch.randn(R, 4 * H, device=DEV).bfloat16()
g = torch.empty_like(u)
ms = timeit(lambda: check(ext.ob_gelu_fwd_bf16(ptr(u), ptr(g), n4, stream())))
report("gelu_fwd", ms, n4 * 2 * 2)
ms = timeit(lambda: check(ext.ob_gelu_bwd_bf16(ptr(u), ptr(g), ptr(g), n4,
                                               stream())))
report("gelu_bwd", ms, n4 * 2 * 3)

# colsum at [R, 3H] (qkv bias grad, biggest colsum)
X3 = torch.randn(R, 3 * H, device=DEV).bfloat16()
db3 = torch.zeros(3 * H, device=DEV)
ms = timeit(lambda: check(ext.ob_colsum_bf16(ptr(X3), ptr(db3), R, 3 * H,
                                             stream())))
report("colsum_3H", ms, R * 3 * H * 2)

# CE fwd/bwd at the padded-vocab logits
V, VP = 50257, 50432
logits = torch.randn(R, VP, device=DEV).bfloat16()
labs = torch.randint(0, V, (8, 1024), device=DEV)
lse = torch.empty(R, device=DEV)
loss = torch.zeros(1, device=DEV)
ms = timeit(lambda: check(ext.ob_ce_fwd_bf16(
    ptr(logits), ptr(labs), ptr(lse), ptr(loss), 8, 1024, V, VP, stream())),
    reps=10)
report("ce_fwd", ms, R * VP * 2)
ms = timeit(lambda: check(ext.ob_ce_bwd_bf16(
    ptr(logits), ptr(labs), ptr(lse), None, 8, 1024, V, VP, stream())),
    reps=10)
report("ce_bwd", ms, R * VP * 2 * 2)

# flash dsum at step shape
O = torch.randn(8, 1024, H, device=DEV).bfloat16()
dO = torch.randn_like(O)
D = torch.empty(8 * 12 * 1024, device=DEV)
ms = timeit(lambda: check(ext.ob_flash_dsum_bf16(
    ptr(O), ptr(dO), ptr(D), 8, 1024, H, 12, stream())))
report("flash_dsum", ms, O.numel() * 2 * 2)

# batched head transpose (flash staging) at step shape
QT = torch.empty(12 * 8, 64, 1024, device=DEV).bfloat16()
qkv = torch.randn(8, 1024, 3 * H, device=DEV).bfloat16()
ms = timeit(lambda: check(ext.ob_transpose_bf16_b(
    ptr(qkv), ptr(QT), 1024, 64, 1024 * 3 * H, 64, 3 * H, 8, 12,
    stream())))
report("transpose_b", ms, QT.numel() * 2 * 2)
