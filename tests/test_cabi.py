# CPU: the C-ABI shared library loads and exports every symbol
# include/oobleck_stage.h declares; host-only entry points (param counts)
# agree with the product/oracle layout.  No compute calls here (no GPU).
import ctypes
import re

import pytest

from tests.conftest import REPO_ROOT

HEADER = REPO_ROOT / "include" / "oobleck_stage.h"


def _ext():
    from oobleck_amd._ext import get_ext
    try:
        return get_ext()
    except RuntimeError as e:
        pytest.fail(f"extension must build on CPU via hipcc: {e}")


def header_symbols():
    text = HEADER.read_text()
    syms = re.findall(r"^\s*(?:int|int64_t|const char\*)\s+(ob_\w+)\s*\(",
                      text, re.MULTILINE)
    assert len(syms) >= 15
    return syms


def test_all_header_symbols_exported():
    lib = _ext()
    for sym in header_symbols():
        assert hasattr(lib, sym), f"missing export: {sym}"


def test_build_arch():
    assert _ext().ob_build_arch() == b"gfx950"


def test_param_counts_match_layout():
    from oobleck_amd._ext import ObLayerDesc
    from oobleck_amd.config import GPT2_SMALL, GPT2_XL
    from oobleck_amd.params import layer_param_numel
    lib = _ext()
    for cfg in (GPT2_SMALL, GPT2_XL):
        for lid in (0, 1, cfg.n_layers_total - 1):
            kind = cfg.layer_kind(lid)
            desc = ObLayerDesc(kind=kind, n_embd=cfg.n_embd, n_head=cfg.n_head,
                               n_positions=cfg.n_positions,
                               vocab_size=cfg.vocab_size, max_batch=8,
                               seq_len=1024, n_slots=2)
            assert lib.ob_layer_param_count(ctypes.byref(desc)) == \
                layer_param_numel(cfg, kind)


def test_product_layout_matches_oracle_layout():
    """The product param spec and the oracle restatement must agree — the
    flat-buffer contract both sides of the parity tests rely on."""
    from oracle.gpt2_oracle import OracleConfig
    from oracle.gpt2_oracle import layer_param_spec as oracle_spec
    from oobleck_amd.config import ModelConfig
    from oobleck_amd.params import layer_param_spec as product_spec
    mc = ModelConfig(n_embd=96, n_head=4, n_layer=3, n_positions=64,
                     vocab_size=211)
    oc = OracleConfig(n_embd=96, n_head=4, n_layer=3, n_positions=64,
                      vocab_size=211)
    for kind in (0, 1, 2):
        assert product_spec(mc, kind) == oracle_spec(oc, kind)
