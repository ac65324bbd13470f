# Within-box A/B of the flash backward kernels at the step shape
# (B=8, nh=12, S=1024, head_dim 64): round-1 2-occupancy dkdv vs the
# paired-wave 3/4-occupancy dkdv (OB_FLASH_PAIR), plus dq and fwd.
import ctypes
import os
import pathlib
import sys

import torch

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
from oobleck_amd._ext import check, get_ext  # noqa: E402

B, nh, S, hd = 8, 12, 1024, 64
H = nh * hd
DEV = "cuda"


def ptr(t):
    return ctypes.c_void_p(0 if t is None else t.data_ptr())


def stream():
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


def timeit(fn, reps=30):
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


ext = get_ext()
g = torch.Generator().manual_seed(3)
qkv = (torch.randn(B, S, 3 * H, generator=g) * 0.3).to(DEV).bfloat16()
dO = (torch.randn(B, S, H, generator=g) * 0.3).to(DEV).bfloat16()
O = torch.empty(B, S, H, device=DEV, dtype=torch.bfloat16)
lse = torch.empty(B * nh * S, device=DEV)
D = torch.empty(B * nh * S, device=DEV)
dqkv = torch.zeros_like(qkv)
VT = torch.empty(B * nh, hd, S, device=DEV, dtype=torch.bfloat16)
QT = torch.empty_like(VT)
KT = torch.empty_like(VT)
dOT = torch.empty_like(VT)
scale = ctypes.c_float(1.0 / hd ** 0.5)

check(ext.ob_transpose_bf16_b(ptr(qkv) if False else
                              ctypes.c_void_p(qkv.data_ptr() + 2 * H * 2),
                              ptr(VT), S, hd, S * 3 * H, hd, 3 * H, B, nh,
                              stream()))
# causal attention fwd flops: 2 matmuls, ~half masked
fl_fwd = 2 * 2 * B * nh * S * S * hd / 2
for mode, name in [("4", "v4-4wave "), ("3", "v3-ladder"),
                   ("pf", "pipelined"), ("v1", "round1   ")]:
    os.environ["OB_FLASH_FWD"] = mode
    ms = timeit(lambda: check(ext.ob_flash_fwd_bf16(
        ptr(qkv), ptr(VT), ptr(O), ptr(lse), B, S, H, nh, scale, stream())))
    print(f"flash_fwd {name} {ms*1e3:8.1f} us  {fl_fwd/ms/1e9:6.0f} TF")
os.environ.pop("OB_FLASH_FWD", None)

check(ext.ob_transpose_bf16_b(ptr(qkv), ptr(QT), S, hd, S * 3 * H, hd,
                              3 * H, B, nh, stream()))
check(ext.ob_transpose_bf16_b(ctypes.c_void_p(qkv.data_ptr() + H * 2),
                              ptr(KT), S, hd, S * 3 * H, hd, 3 * H, B, nh,
                              stream()))
check(ext.ob_transpose_bf16_b(ptr(dO), ptr(dOT), S, hd, S * H, hd, H, B,
                              nh, stream()))
check(ext.ob_flash_dsum_bf16(ptr(O), ptr(dO), ptr(D), B, S, H, nh, stream()))

fl_bwd = 4 * 2 * B * nh * S * S * hd / 2 + 3 * 2 * B * nh * S * S * hd / 2
results = {}
for dkdv, dq, name in [("s", "", "split+v3"), ("3", "", "v3+v3   "),
                       ("p", "", "pair+v3 "), ("p", "1", "pair+pf "),
                       ("0", "1", "round1  ")]:
    os.environ["OB_FLASH_PAIR"] = dkdv
    if dq:
        os.environ["OB_FLASH_DQ"] = dq
    else:
        os.environ.pop("OB_FLASH_DQ", None)
    ms = timeit(lambda: check(ext.ob_flash_bwd_bf16(
        ptr(qkv), ptr(QT), ptr(KT), ptr(dOT), ptr(dO), ptr(lse), ptr(D),
        ptr(dqkv), B, S, H, nh, scale, stream())))
    results[name] = ms
    print(f"flash_bwd {name} {ms*1e3:8.1f} us  {fl_bwd/ms/1e9:6.0f} TF "
          f"(dkdv+dq together)")
print(f"speedup v3 vs round1: {results['round1  ']/results['v3+v3   ']:.3f}x")
