# ctypes binding of the C-ABI (include/oobleck_stage.h).
#
# The product compute path runs ONLY through this extension.  If the .so is
# missing on a machine with a GPU, get_ext() raises — there is no eager /
# PyTorch fallback for the hot path (tier contract: a silent fallback would
# void every parity and perf claim).
from __future__ import annotations

import ctypes
import pathlib

_LIB = None
SO_PATH = pathlib.Path(__file__).resolve().parent / "libob_stage.so"


class ObLayerDesc(ctypes.Structure):
    _fields_ = [
        ("kind", ctypes.c_int32),
        ("n_embd", ctypes.c_int32),
        ("n_head", ctypes.c_int32),
        ("n_positions", ctypes.c_int32),
        ("vocab_size", ctypes.c_int32),
        ("max_batch", ctypes.c_int32),
        ("seq_len", ctypes.c_int32),
        ("n_slots", ctypes.c_int32),
        ("dtype", ctypes.c_int32),
    ]


def _configure(lib: ctypes.CDLL) -> ctypes.CDLL:
    i64, i32, f32, vp = ctypes.c_int64, ctypes.c_int32, ctypes.c_float, ctypes.c_void_p
    lib.ob_last_error.restype = ctypes.c_char_p
    lib.ob_build_arch.restype = ctypes.c_char_p
    lib.ob_layer_param_count.restype = i64
    lib.ob_layer_param_count.argtypes = [ctypes.POINTER(ObLayerDesc)]
    lib.ob_layer_create.argtypes = [ctypes.POINTER(ObLayerDesc),
                                    ctypes.POINTER(vp)]
    lib.ob_layer_bind.argtypes = [vp, vp, vp]
    lib.ob_layer_set_batch.argtypes = [vp, i32]
    lib.ob_layer_forward.argtypes = [vp, i32, vp, vp, vp, vp]
    lib.ob_layer_backward.argtypes = [vp, i32, vp, vp, vp]
    lib.ob_layer_destroy.argtypes = [vp]
    lib.ob_layer_refresh_weights.argtypes = [vp, vp]
    lib.ob_adamw_step.argtypes = [vp, vp, vp, vp, i64, i32, f32, f32, f32,
                                  f32, f32, vp]
    lib.ob_gemm_f32.argtypes = [i32, i32, i64, i64, i64, f32, vp, i64, i64,
                                i64, vp, i64, i64, i64, f32, vp, i64, i64,
                                i64, i64, i64, vp, vp, i32, i32, vp]
    lib.ob_layernorm_fwd_f32.argtypes = [vp, vp, vp, vp, vp, vp, i64, i64,
                                         f32, vp]
    lib.ob_layernorm_bwd_f32.argtypes = [vp, vp, vp, vp, vp, vp, vp, vp, i64,
                                         i64, i32, vp]
    lib.ob_softmax_causal_fwd_f32.argtypes = [vp, i64, i64, f32, vp]
    lib.ob_softmax_causal_bwd_f32.argtypes = [vp, vp, i64, i64, vp]
    lib.ob_gelu_fwd_f32.argtypes = [vp, vp, i64, vp]
    lib.ob_gelu_bwd_f32.argtypes = [vp, vp, vp, i64, vp]
    lib.ob_colsum_f32.argtypes = [vp, vp, i64, i64, vp]
    lib.ob_gemm_bf16.argtypes = [i32, i32, i64, i64, i64, f32, vp, i64, i64,
                                 i64, vp, i64, i64, i64, f32, vp, i64, i64,
                                 i64, i64, i64, vp, vp, i32, i32, vp]
    lib.ob_f32_to_bf16.argtypes = [vp, vp, i64, vp]
    lib.ob_f32_to_bf16_t.argtypes = [vp, vp, i64, i64, vp]
    lib.ob_f32_to_bf16_t_ld.argtypes = [vp, vp, i64, i64, i64, vp]
    lib.ob_bf16_to_f32.argtypes = [vp, vp, i64, vp]
    # every remaining kernel entry point: ctypes without argtypes passes
    # Python ints as 32-bit c_int, leaving int64 params with undefined
    # upper halves (the elem_probe/ce_probe GPU faults) — declare them all
    lib.ob_gelu_fwd_bf16.argtypes = [vp, vp, i64, vp]
    lib.ob_gelu_bwd_bf16.argtypes = [vp, vp, vp, i64, vp]
    lib.ob_ce_fwd_bf16.argtypes = [vp, vp, vp, vp, i64, i64, i64, i64, vp]
    lib.ob_ce_bwd_bf16.argtypes = [vp, vp, vp, vp, i64, i64, i64, i64, vp]
    lib.ob_embed_fwd_bf16.argtypes = [vp, vp, vp, vp, i64, i64, i64, vp]
    lib.ob_embed_bwd_bf16.argtypes = [vp, vp, vp, vp, i64, i64, i64, vp]
    lib.ob_softmax_causal_fwd_bf16.argtypes = [vp, i64, i64, f32, vp]
    lib.ob_softmax_causal_bwd_bf16.argtypes = [vp, vp, i64, i64, vp]
    lib.ob_transpose_bf16.argtypes = [vp, vp, i64, i64, vp]
    lib.ob_gemm_lt_bias.argtypes = [i32, i32, i64, i64, i64, f32, vp, i64,
                                    vp, i64, f32, vp, i64, i32, vp, vp]
    lib.ob_gemm_lt_f32.argtypes = [i32, i32, i64, i64, i64, f32, vp, i64,
                                   vp, i64, f32, vp, i64, vp]
    lib.ob_flash_fwd_bf16.argtypes = [vp, vp, vp, vp, i64, i64, i64, i64,
                                      f32, vp]
    lib.ob_transpose_bf16_b.argtypes = [vp, vp, i64, i64, i64, i64, i64,
                                        i64, i64, vp]
    lib.ob_layernorm_fwd_bf16.argtypes = [vp, vp, vp, vp, vp, vp, i64, i64,
                                          f32, vp]
    lib.ob_layernorm_bwd_bf16.argtypes = [vp, vp, vp, vp, vp, vp, vp, vp,
                                          i64, i64, i32, vp]
    lib.ob_colsum_bf16.argtypes = [vp, vp, i64, i64, vp]
    lib.ob_gemm_lt.argtypes = [i32, i32, i64, i64, i64, f32, vp, i64, vp,
                               i64, f32, vp, i64, i32, vp]
    lib.ob_gemm_bf16_nt_8ph.argtypes = [vp, vp, vp, vp, vp, i64, i64, i64,
                                        i64, i64, i64, i64, i64, i64, i64,
                                        i64, i64, i64, i64, f32, f32, i32,
                                        i32, vp, i64]
    lib.ob_profile_enable.argtypes = [i32]
    lib.ob_profile_enable.restype = None
    lib.ob_profile_reset.argtypes = []
    lib.ob_profile_reset.restype = None
    lib.ob_profile_read.argtypes = [i32, ctypes.POINTER(ctypes.c_double),
                                    ctypes.POINTER(ctypes.c_longlong)]
    lib.ob_flash_dsum_bf16.argtypes = [vp, vp, vp, i64, i64, i64, i64, vp]
    lib.ob_flash_bwd_bf16.argtypes = [vp, vp, vp, vp, vp, vp, vp, vp, i64,
                                      i64, i64, i64, f32, vp]
    return lib


def get_ext(build_if_missing: bool = True) -> ctypes.CDLL:
    """Load (building in-tree if necessary) the HIP extension.  Raises
    RuntimeError — loudly, no fallback — when unavailable."""
    global _LIB
    if _LIB is not None:
        return _LIB
    if not SO_PATH.exists():
        if build_if_missing:
            from .build import build
            try:
                build()
            except Exception as e:  # noqa: BLE001
                raise RuntimeError(
                    f"oobleck_amd: HIP extension missing and build failed: {e}. "
                    "The hot path has no fallback — run __graft_entry__.build().") from e
        else:
            raise RuntimeError(
                "oobleck_amd: HIP extension libob_stage.so is missing. "
                "The hot path has no fallback — run __graft_entry__.build().")
    _LIB = _configure(ctypes.CDLL(str(SO_PATH)))
    return _LIB


def check(rc: int, what: str = "") -> None:
    if rc != 0:
        err = get_ext().ob_last_error().decode()
        raise RuntimeError(f"oobleck_amd extension error in {what}: {err}")
