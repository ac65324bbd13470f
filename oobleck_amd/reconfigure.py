# Reconfiguration rank arithmetic — restates the reference's
# ReconfigurationEngine.on_reconfigure / _find_biggest_pipeline /
# _merge_pipelines (/root/reference/oobleck/execution/engine.py:91-180,
# 311-360) exactly; behaviour is pinned by the reference's own 22 golden
# scenarios (tests/execution/test_reconfiguration.py there), ported to
# tests/test_reconfigure.py here.
#
# The data path this feeds (RCCL PG re-init + per-layer flat-param
# broadcast) is oobleck_amd.engine.copy_model_states.
from __future__ import annotations


def _find_biggest(ranks_list: list[list[int]], min_num_ranks: int) -> list[int] | None:
    """reference _find_biggest_pipeline (engine.py:335-349): the LAST
    pipeline among those with maximal rank count ('>=' comparison), and only
    if it can yield a rank (len > min)."""
    biggest: list[int] | None = None
    for ranks in ranks_list:
        if biggest is None or len(ranks) >= len(biggest):
            biggest = ranks
    if biggest is not None and len(biggest) > min_num_ranks:
        return biggest
    return None


def _merge_pipelines(ranks_list: list[list[int]], min_num_ranks: int) -> list[list[int]]:
    """reference _merge_pipelines (engine.py:311-333)."""
    ranks_to_merge: list[list[int]] = []
    results: list[list[int]] = []
    for ranks in ranks_list:
        (ranks_to_merge if len(ranks) < min_num_ranks else results).append(ranks)
    try:
        while ranks_to_merge:
            ranks = ranks_to_merge.pop(0)
            try:
                while len(ranks) < min_num_ranks:
                    ranks.extend(ranks_to_merge.pop(0))
            except IndexError:
                ranks.extend(results.pop(0))
            assert len(ranks) >= min_num_ranks
            results.append(ranks)
    except IndexError:
        raise RuntimeError("Ranks are insufficient")
    assert ranks_to_merge == []
    return results


def compute_new_ranks_list(pipelines_ranks: list[list[int]],
                           lost_ranks: list[int],
                           min_num_ranks: int) -> list[list[int]]:
    """The pure rank surgery of on_reconfigure (engine.py:104-157): drop
    lost ranks, let under-sized pipelines borrow from the biggest, merge
    when nothing can yield, then sort each pipeline's ranks and order
    pipelines by (size, first rank).

    `pipelines_ranks` is mutated the way the reference mutates
    pipeline._ranks (shared list objects matter: a pipeline appended to the
    result can still lose ranks to a later borrower)."""
    live = [[r for r in ranks if r not in lost_ranks] for ranks in pipelines_ranks]

    need_merge = False
    new_ranks_list: list[list[int]] = []
    for ranks in live:
        if len(ranks) == 0:
            continue
        if len(ranks) >= min_num_ranks:
            new_ranks_list.append(ranks)
            continue
        while len(ranks) < min_num_ranks:
            biggest = _find_biggest(live, min_num_ranks)
            if biggest is None:
                need_merge = True
                break
            while (len(biggest) > min_num_ranks and len(ranks) < min_num_ranks):
                ranks.append(biggest.pop())
        new_ranks_list.append(ranks)

    if need_merge:
        new_ranks_list = _merge_pipelines(new_ranks_list, min_num_ranks)

    for ranks in new_ranks_list:
        ranks.sort()
    new_ranks_list.sort(key=lambda ranks: (len(ranks), ranks[0]))
    return new_ranks_list
