# Canonical flat-parameter layout per fx-sharded layer — the product-side
# definition of the contract shared with include/oobleck_stage.h (the C
# extension computes identical offsets in ob_layer.hip::block_params) and
# restated by the test oracle (oracle/gpt2_oracle.py::layer_param_spec;
# tests/test_params.py asserts the two agree).
#
# The flat buffer is the drop-in equivalent of the reference's
# FlatParamHandle.flat_param (/root/reference/oobleck/execution/layer.py:
# 96-111): one contiguous fp32 tensor per layer, consumed whole by the
# optimizer (pipeline.py:117-119), DP all-reduce (layer.py:272-291) and the
# reconfiguration broadcast (engine.py:283-299).
from __future__ import annotations

import math

import torch

from .config import ModelConfig

KIND_EMBED = 0
KIND_BLOCK = 1
KIND_FINAL = 2


def layer_param_spec(cfg: ModelConfig, kind: int) -> list[tuple[str, tuple[int, ...]]]:
    H, V, P = cfg.n_embd, cfg.vocab_size, cfg.n_positions
    if kind == KIND_EMBED:
        return [("wte", (V, H)), ("wpe", (P, H))]
    if kind == KIND_BLOCK:
        return [
            ("ln1_w", (H,)), ("ln1_b", (H,)),
            ("w_qkv", (H, 3 * H)), ("b_qkv", (3 * H,)),
            ("w_attnproj", (H, H)), ("b_attnproj", (H,)),
            ("ln2_w", (H,)), ("ln2_b", (H,)),
            ("w_fc", (H, 4 * H)), ("b_fc", (4 * H,)),
            ("w_mlpproj", (4 * H, H)), ("b_mlpproj", (H,)),
        ]
    if kind == KIND_FINAL:
        return [("lnf_w", (H,)), ("lnf_b", (H,)), ("w_lm", (V, H))]
    raise ValueError(f"bad kind {kind}")


def layer_param_numel(cfg: ModelConfig, kind: int) -> int:
    return sum(math.prod(s) for _, s in layer_param_spec(cfg, kind))


def init_layer_params(cfg: ModelConfig, layer_id: int, seed: int,
                      style: str = "gpt2") -> torch.Tensor:
    """Random-init a layer's flat fp32 parameter buffer (CPU tensor).

    style="gpt2": HF-like init (normal 0.02 weights, zero biases, unit LN
    weights).  style="reference": uniform [0,1) for every tensor, as the
    reference's init_tensors does (layer.py:26-37, torch.rand).
    style="zeros": all-zero buffer with NO host-side random generation —
    for layers whose params arrive by reconfiguration broadcast
    (engine.py:283-299), where a CPU random init is pure wasted wall-clock
    on the <2 s recovery path."""
    kind = cfg.layer_kind(layer_id)
    if style == "zeros":
        return torch.zeros(layer_param_numel(cfg, kind))
    g = torch.Generator().manual_seed(seed * 1000 + layer_id)
    flats = []
    for name, shape in layer_param_spec(cfg, kind):
        if style == "reference":
            t = torch.rand(shape, generator=g)
        elif name.startswith(("ln1_w", "ln2_w", "lnf_w")):
            t = torch.ones(shape)
        elif name.endswith("_b") or name.startswith("b_"):
            t = torch.zeros(shape)
        else:
            t = torch.randn(shape, generator=g) * 0.02
        flats.append(t.reshape(-1))
    return torch.cat(flats).float()
