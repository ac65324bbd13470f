# Pins the bench topology the driver's scaling run exercises (BENCH/SCALE):
# TOPOLOGY N -> (stages, replicas), the stage splits, and the rank grids.
import pytest

from oobleck_amd.config import GPT2_SMALL, GPT2_XL
from oobleck_amd.engine import even_stage_split, make_rank_grid


@pytest.mark.parametrize("n,stages,replicas",
                         [(1, 1, 1), (2, 2, 1), (4, 4, 1), (8, 4, 2)])
def test_bench_topology(n, stages, replicas):
    from bench import TOPOLOGY
    assert TOPOLOGY[n] == (stages, replicas)
    split = even_stage_split(GPT2_SMALL, stages)
    # contiguous cover of all 14 layers, each stage non-empty
    flat = [lid for s in split for lid in s]
    assert flat == list(range(GPT2_SMALL.n_layers_total))
    assert all(split)
    for pid in range(replicas):
        ranks = [pid * stages + s for s in range(stages)]
        grid = make_rank_grid(GPT2_SMALL.n_layers_total, split,
                              [[r] for r in ranks])
        # every layer maps to exactly its stage's rank
        for sid, lids in enumerate(split):
            for lid in lids:
                assert grid[lid] == [ranks[sid]]


def test_stage_split_balances_compute():
    """The lm_head-heavy final layer must not share a stage with a full
    block pile at 4 stages (the N=4/8 config)."""
    split = even_stage_split(GPT2_SMALL, 4)
    # matmul-weight proxy (mirrors even_stage_split's weights)
    H, V = GPT2_SMALL.n_embd, GPT2_SMALL.vocab_size
    L = GPT2_SMALL.n_layers_total

    def w(lid):
        if lid == 0:
            return 0.1
        if lid == L - 1:
            return V / (12.0 * H)
        return 1.0

    loads = [sum(w(l) for l in s) for s in split]
    assert max(loads) / (sum(loads) / len(loads)) < 1.6, (split, loads)


def test_xl_config_flash_eligible():
    """config[3]'s dims (gpt3.yaml = 1600x48) keep head_dim 64, so the
    fused flash path applies there unchanged."""
    assert GPT2_XL.n_embd // GPT2_XL.n_head == 64
    assert GPT2_XL.n_layer == 48
    split = even_stage_split(GPT2_XL, 8)
    flat = [lid for s in split for lid in s]
    assert flat == list(range(GPT2_XL.n_layers_total))
