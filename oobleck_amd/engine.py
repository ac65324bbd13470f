# DataParallelEngine + reconfiguration layer-copy — mirrors
# /root/reference/oobleck/execution/engine.py:363-412 (per-(layer,fsdp)
# all-reduce groups across pipelines) and :238-309 (_copy_model_states, the
# post-failure RCCL broadcast path).
from __future__ import annotations

from collections import defaultdict

import torch
import torch.distributed as dist

from .config import ModelConfig


def make_rank_grid(n_layers_total: int, stage_layer_ids: list[list[int]],
                   stage_ranks: list[list[int]]) -> dict[int, list[int]]:
    """layer index -> list of ranks (one per fsdp shard slot), the shape
    PipelineTemplate.get_rank_grid produces (pipeline_template.h:57-84)."""
    grid: dict[int, list[int]] = {}
    for lids, ranks in zip(stage_layer_ids, stage_ranks):
        for lid in lids:
            grid[lid] = list(ranks)
    assert sorted(grid) == list(range(n_layers_total))
    return grid


def even_stage_split(cfg: ModelConfig, n_stages: int) -> list[list[int]]:
    """Contiguous, parameter-balanced split of the model's layers into
    n_stages (stand-in for the C++ planner's stage assignment; the planner
    itself is out of scope this round — SURVEY.md §8 f2)."""
    import bisect

    from .params import KIND_BLOCK, KIND_FINAL
    L = cfg.n_layers_total
    assert 1 <= n_stages <= L
    # balance by compute (matmul FLOPs/token ∝ matmul params), not by
    # parameter count: the embedding layer is a gather (≈free) while the
    # final layer's lm_head GEMM costs ≈ V*H/(12*H^2) blocks of work.
    H, V = cfg.n_embd, cfg.vocab_size

    def w(lid):
        kind = cfg.layer_kind(lid)
        if kind == KIND_BLOCK:
            return 12 * H * H
        if kind == KIND_FINAL:
            return V * H
        return H  # embedding: negligible compute
    weights = [w(i) for i in range(L)]
    prefix = [0]
    for w in weights:
        prefix.append(prefix[-1] + w)
    bounds = [0]
    for i in range(1, n_stages):
        target = prefix[-1] * i / n_stages
        pos = bisect.bisect_left(prefix, target)
        if pos > 0 and abs(prefix[pos - 1] - target) < abs(prefix[pos] - target):
            pos -= 1
        pos = max(bounds[-1] + 1, min(pos, L - (n_stages - i)))
        bounds.append(pos)
    bounds.append(L)
    return [list(range(bounds[i], bounds[i + 1])) for i in range(n_stages)]


class DataParallelEngine:
    """Builds the per-(layer, fsdp-index) all-reduce process groups across
    pipelines and drives the per-layer grad all-reduce
    (reference engine.py:363-412)."""

    def __init__(self, pipelines, num_gpus_per_shard: int | None = None):
        ranks_grid: dict[int, dict[int, list[int]]] = defaultdict(dict)
        for pipeline in pipelines:
            for layer_index, ranks in pipeline.rank_grid.items():
                for fsdp_index, rank in enumerate(ranks):
                    ranks_grid[layer_index].setdefault(fsdp_index, []).append(rank)

        dp_process_groups: dict[int, dict[int, dist.ProcessGroup]] = defaultdict(dict)
        my_rank = dist.get_rank()
        self._my_fsdp_indices: dict[int, list[int]] = defaultdict(list)
        for layer_index, per_layer in ranks_grid.items():
            for fsdp_index, ranks in per_layer.items():
                dp_process_groups[layer_index][fsdp_index] = dist.new_group(
                    sorted(set(ranks)))
                if my_rank in ranks:
                    self._my_fsdp_indices[layer_index].append(fsdp_index)
        self._dp_process_groups = dp_process_groups

    def do_allreduce(self, my_pipeline) -> None:
        # reference engine.py:404-412
        for layer in my_pipeline.execution._layers:
            pgs = {
                fsdp_index: pg
                for fsdp_index, pg in self._dp_process_groups[layer.layer_id].items()
                if dist.get_rank(pg) >= 0
            }
            if pgs:
                layer.reduce_gradients(pgs)


def copy_model_states(old_rank_grids: list[dict[int, list[int]]],
                      new_rank_grids: list[dict[int, list[int]]],
                      my_layers_by_id: dict[int, "object"],
                      dp_engine: DataParallelEngine) -> None:
    """Post-reconfiguration layer copy (reference engine.py:238-309): for
    each layer whose rank set changed, broadcast the flat param from a
    surviving rank over the layer's DP group.  `my_layers_by_id` maps
    layer_id -> Layer for the layers this rank will own after reconfig
    (with flat_param already allocated)."""
    my_rank = dist.get_rank()
    works = []
    for layer_index in range(len(old_rank_grids[0])):
        old_ranks = [g[layer_index] for g in old_rank_grids]
        new_ranks = [g[layer_index] for g in new_rank_grids]
        if all(rank in old_ranks for rank in new_ranks):
            continue
        alive = [ranks for ranks in old_ranks if ranks in new_ranks]
        if not alive:
            raise RuntimeError(f"No alive ranks for layer {layer_index}.")
        ranks_to_send = alive[0]
        for ranks_recv in new_ranks:
            if my_rank in ranks_recv:
                fsdp_index = ranks_recv.index(my_rank)
                dp_group = dp_engine._dp_process_groups[layer_index][fsdp_index]
                param = my_layers_by_id[layer_index].flat_param
                works.append(dist.broadcast(
                    tensor=param, src=ranks_to_send[fsdp_index],
                    group=dp_group, async_op=True))
    # the reference relies on barrier + cuda.synchronize (engine.py:308-309);
    # waiting the async works is required for correctness on gloo and is
    # free on RCCL (stream-ordered).
    for w in works:
        w.wait()
    # the copy wrote master params behind the layers' backs: refresh any
    # derived state (bf16 shadows; FULL_SHARD resident-full re-gather)
    for layer in my_layers_by_id.values():
        refresh = getattr(layer, "refresh_weights", None)
        if refresh is not None:
            refresh()
    dist.barrier()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
