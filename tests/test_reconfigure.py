# Golden tests for the reconfiguration rank arithmetic — the 9 no-FSDP +
# 13 FSDP scenarios the reference pins with exact expected rank lists
# (/root/reference/tests/execution/test_reconfiguration.py:151-448),
# restated against our compute_new_ranks_list.  Initial layouts: 4
# pipelines of 2/3/4/5 nodes, num_gpus_per_node ∈ {1,2,4}; min pipeline =
# 2 nodes.  Plus tests for the pyomo-free planning arithmetic.
import pytest

from oobleck_amd.instantiator import (TemplateSpec, distribute_batch,
                                      enumerate_instantiation_options,
                                      partition_microbatches)
from oobleck_amd.reconfigure import compute_new_ranks_list


def initial_pipelines(gpus_per_node: int) -> list[list[int]]:
    out, start = [], 0
    for nodes in (2, 3, 4, 5):
        n = nodes * gpus_per_node
        out.append(list(range(start, start + n)))
        start += n
    return out


NO_FSDP = [
    ([2], [[0, 1], [3, 4], [5, 6, 7, 8], [9, 10, 11, 12, 13]], "base1"),
    ([6, 8], [[0, 1], [5, 7], [2, 3, 4], [9, 10, 11, 12, 13]], "base2"),
    ([10, 11], [[0, 1], [2, 3, 4], [9, 12, 13], [5, 6, 7, 8]], "base3"),
    ([1], [[0, 13], [2, 3, 4], [5, 6, 7, 8], [9, 10, 11, 12]], "borrow1"),
    ([1, 3, 4], [[0, 13], [2, 12], [9, 10, 11], [5, 6, 7, 8]], "borrow2"),
    ([2, 4, 6, 7, 8], [[0, 1], [3, 13], [5, 12], [9, 10, 11]], "borrow3"),
    ([1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 11], [[0, 12, 13]], "merge1"),
    ([1, 2, 3, 5, 6, 7, 9, 11, 12, 13], [[0, 4], [8, 10]], "merge2"),
    ([1, 2, 3, 5, 6, 7, 9, 10, 11], [[0, 4], [8, 12, 13]], "merge3"),
]

FSDP = [
    (2, [6, 7], [list(range(0, 4)), [4, 5, 8, 9], list(range(10, 18)),
                 list(range(18, 28))], "fsdp2_base1"),
    (2, [10, 11, 18, 19], [list(range(0, 4)), list(range(4, 10)),
                           [12, 13, 14, 15, 16, 17],
                           [20, 21, 22, 23, 24, 25, 26, 27]], "fsdp2_base2"),
    (4, [8, 9, 10, 11], [list(range(0, 8)),
                         [12, 13, 14, 15, 16, 17, 18, 19],
                         list(range(20, 36)), list(range(36, 56))],
     "fsdp4_base1"),
    (4, [20, 21, 22, 23, 28, 29, 30, 31],
     [list(range(0, 8)), [24, 25, 26, 27, 32, 33, 34, 35],
      list(range(8, 20)), list(range(36, 56))], "fsdp4_base2"),
    (2, [2, 3], [[0, 1, 26, 27], list(range(4, 10)), list(range(10, 18)),
                 [18, 19, 20, 21, 22, 23, 24, 25]], "fsdp2_borrow1"),
    (2, [2, 3, 4, 5, 8, 9], [[0, 1, 26, 27], [6, 7, 24, 25],
                             [18, 19, 20, 21, 22, 23], list(range(10, 18))],
     "fsdp2_borrow2"),
    (2, [2, 3, 10, 11, 14, 15, 16, 17],
     [[0, 1, 26, 27], [12, 13, 24, 25], list(range(4, 10)),
      [18, 19, 20, 21, 22, 23]], "fsdp2_borrow3"),
    (4, [4, 5, 6, 7], [[0, 1, 2, 3, 52, 53, 54, 55], list(range(8, 20)),
                       list(range(20, 36)),
                       [36, 37, 38, 39, 40, 41, 42, 43, 44, 45, 46, 47,
                        48, 49, 50, 51]], "fsdp4_borrow1"),
    (4, [4, 5, 6, 7, 8, 9, 10, 11, 16, 17, 18, 19],
     [[0, 1, 2, 3, 52, 53, 54, 55], [12, 13, 14, 15, 48, 49, 50, 51],
      [36, 37, 38, 39, 40, 41, 42, 43, 44, 45, 46, 47],
      list(range(20, 36))], "fsdp4_borrow2"),
    (4, [4, 5, 6, 7, 8, 9, 10, 11, 12, 13, 14, 15, 36, 37, 38, 39],
     [[0, 1, 2, 3, 52, 53, 54, 55], [16, 17, 18, 19, 32, 33, 34, 35],
      [20, 21, 22, 23, 24, 25, 26, 27, 28, 29, 30, 31],
      [40, 41, 42, 43, 44, 45, 46, 47, 48, 49, 50, 51]], "fsdp4_borrow3"),
    (2, [2, 3, 8, 9, 14, 15, 16, 17, 22, 23, 24, 25, 26, 27],
     [[10, 11, 12, 13], [18, 19, 20, 21], [0, 1, 4, 5, 6, 7]],
     "fsdp2_merge1"),
    (2, [2, 3, 8, 9, 14, 15, 16, 17, 22, 23, 24, 25, 26, 27],
     [[10, 11, 12, 13], [18, 19, 20, 21], [0, 1, 4, 5, 6, 7]],
     "fsdp2_merge2"),
    (2, [2, 3, 6, 7, 8, 9, 10, 11, 14, 15, 16, 17, 22, 23, 24, 25, 26, 27],
     [[0, 1, 4, 5], [12, 13, 18, 19, 20, 21]], "fsdp2_merge3"),
]


@pytest.mark.parametrize("failed,expected,name", NO_FSDP,
                         ids=[c[2] for c in NO_FSDP])
def test_no_fsdp_reconfiguration(failed, expected, name):
    got = compute_new_ranks_list(initial_pipelines(1), failed,
                                 min_num_ranks=2)
    assert got == expected, name


@pytest.mark.parametrize("gpus,failed,expected,name", FSDP,
                         ids=[c[3] for c in FSDP])
def test_fsdp_reconfiguration(gpus, failed, expected, name):
    got = compute_new_ranks_list(initial_pipelines(gpus), failed,
                                 min_num_ranks=2 * gpus)
    assert got == expected, name


# ---------------------------------------------------------------------------
# planning arithmetic (pyomo-free replacements)
# ---------------------------------------------------------------------------

def test_enumerate_instantiation_options():
    t2 = TemplateSpec(2, 1, 1.0, 2)
    t3 = TemplateSpec(3, 1, 1.4, 3)
    options = enumerate_instantiation_options([t2, t3], 7)
    # node-count combinations summing to 7 from {2,3}: 2+2+3, 2+2+... ->
    # {2:2,3:1}, {3:1,2:2} same, {2:... 7 = 2*2+3 = 2+2+3 only, or 3+2+2...
    # and 7 = 3+... 3*1+2*2; also 7 = 2*0+3*.. no (3*2=6, 3*1=3).
    sets = {tuple(sorted((k.num_nodes, v) for k, v in o.items() if v))
            for o in options}
    assert sets == {((2, 2), (3, 1))}


def test_distribute_batch_equal_pipelines():
    t = TemplateSpec(2, 1, 1.0, 2)
    nb = distribute_batch(16, {t: 2})
    assert nb == {t: 8}  # 2 identical pipelines: 8 microbatches each


def test_distribute_batch_heterogeneous():
    # slower pipeline (higher T/s) gets fewer microbatches
    fast = TemplateSpec(4, 1, 1.0, 4)   # T/s = 0.25
    slow = TemplateSpec(2, 1, 1.0, 2)   # T/s = 0.5
    nb = distribute_batch(24, {fast: 1, slow: 1})
    assert nb is not None
    assert nb[fast] + nb[slow] == 24
    assert nb[fast] > nb[slow]
    # exact optimum of sum((w_i nb_i - avg)^2): nb_fast=16, nb_slow=8 gives
    # perfectly equal w*nb = 4.0
    assert nb == {fast: 16, slow: 8}


def test_partition_microbatches():
    a = TemplateSpec(2, 1, 1.0, 2)
    b = TemplateSpec(3, 1, 1.2, 3)
    parts = partition_microbatches({a: 2, b: 1}, {a: 4, b: 8})
    assert parts == [4, 4, 8]
    assert sum(parts) == 16
