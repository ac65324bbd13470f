# End-to-end reconfiguration drill on CPU/gloo (the config[4] analog,
# BASELINE.json: "kill 1 rank mid-run, reinstantiate, measure recovery"):
#
#   phase 1: 5 ranks, two pipelines [[0,1] (2 stages), [2,3,4] (3 stages)],
#            per-rank marker params.
#   loss:    rank 1 dies.  compute_new_ranks_list borrows rank 4 into
#            pipeline 0 -> [[0,4], [2,3]] (the reference's borrow rule).
#   phase 2: survivors destroy the world, re-rendezvous at world 4 (rank
#            remap preserves identity), rebuild pipelines + DP groups,
#            copy_model_states broadcasts the layers whose rank sets
#            changed from surviving owners, then the new pipelines run a
#            full 1F1B step.
#
# Composes ONLY product pieces: compute_new_ranks_list, even_stage_split,
# make_rank_grid, OobleckPipeline, DataParallelEngine, copy_model_states.
from __future__ import annotations

import os
import pathlib
import sys
import time

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

REPO_ROOT = pathlib.Path(__file__).resolve().parent.parent

TINY = dict(n_embd=96, n_head=4, n_layer=3, n_positions=64, vocab_size=211)
B, S = 2, 32


def _drill(rank: int, world: int, tmp: str):
    if str(REPO_ROOT) not in sys.path:
        sys.path.insert(0, str(REPO_ROOT))
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    from oobleck_amd.config import ModelConfig, TrainingConfig
    from oobleck_amd.engine import (DataParallelEngine, copy_model_states,
                                    even_stage_split, make_rank_grid)
    from oobleck_amd.pipeline import OobleckPipeline
    from oobleck_amd.reconfigure import compute_new_ranks_list
    from oracle.gpt2_oracle import OracleConfig, init_layer_params
    from tests.oracle_layer import NoOpOptimizer, OracleLayer

    mc, oc = ModelConfig(**TINY), OracleConfig(**TINY)
    tc = TrainingConfig(microbatch_size=B, global_microbatch_size=2 * B,
                        seq_len=S)
    L = oc.n_layers_total
    base_flats = [init_layer_params(oc, oc.layer_kind(i), 55 + i)
                  for i in range(L)]

    def build_pipelines(ranks_lists, my_layers_prev):
        """Build all pipelines (co-calling new_group everywhere); return
        (pipelines, my_pipeline, my_layers dict)."""
        pipelines, my_pipeline = [], None
        my_layers: dict[int, OracleLayer] = {}
        me = dist.get_rank()
        for pid, ranks in enumerate(ranks_lists):
            stages = even_stage_split(mc, len(ranks))
            grid = make_rank_grid(L, stages, [[r] for r in ranks])

            class Loader:
                def __iter__(self_inner):
                    g = torch.Generator().manual_seed(500 + pid)
                    while True:
                        ids = torch.randint(0, mc.vocab_size, (B, S),
                                            generator=g)
                        yield {"input_ids": ids, "labels": ids.clone()}

            p = OobleckPipeline(pid, grid, mc, tc, Loader(), 2,
                                torch.device("cpu"))
            p.initialize_distributed_fsdp()
            p.initialize_distributed_pipeline()
            if p.my_pipeline:
                def factory(lid, pg, n_slots):
                    prev = my_layers_prev.get(lid)
                    flat = (prev.flat_param.detach().clone() if prev is not None
                            else torch.zeros_like(base_flats[lid]))
                    layer = OracleLayer(lid, oc, flat)
                    my_layers[lid] = layer
                    return layer
                p.initialize_execution(
                    layer_factory=factory,
                    optimizer_factory=lambda layers: (NoOpOptimizer(layers),
                                                      None))
                my_pipeline = p
            pipelines.append(p)
        return pipelines, my_pipeline, my_layers

    # ---- phase 1: world 5, pipelines [[0,1],[2,3,4]] -----------------------
    dist.init_process_group("gloo", init_method=f"file://{tmp}/rdzv1",
                            rank=rank, world_size=world)
    old_ranks_lists = [[0, 1], [2, 3, 4]]
    seed_layers = {lid: OracleLayer(lid, oc, base_flats[lid] + rank)
                   for lid in range(L)}  # +rank marker to verify the copy
    pipelines, my_pipeline, my_layers = build_pipelines(old_ranks_lists,
                                                        seed_layers)
    old_grids = [p.rank_grid for p in pipelines]
    dist.barrier()
    dist.destroy_process_group()

    # ---- the failure: rank 1 dies -----------------------------------------
    lost = [1]
    if rank in lost:
        return
    t0 = time.perf_counter()
    new_ranks = compute_new_ranks_list([list(r) for r in old_ranks_lists],
                                       lost, min_num_ranks=2)
    assert new_ranks == [[0, 4], [2, 3]]

    # survivors re-rendezvous with contiguous ranks (the reference re-inits
    # distributed from the updated rank_map, engine.py:526-598)
    survivors = sorted(r for rl in old_ranks_lists for r in rl
                       if r not in lost)
    remap = {old: new for new, old in enumerate(survivors)}
    new_rank = remap[rank]
    dist.init_process_group("gloo", init_method=f"file://{tmp}/rdzv2",
                            rank=new_rank, world_size=len(survivors))
    new_ranks_re = [[remap[r] for r in rl] for rl in new_ranks]
    old_grids_re = [{lid: [remap.get(r, -1) for r in rs]
                     for lid, rs in g.items()} for g in old_grids]

    # ---- rebuild + copy + resume ------------------------------------------
    pipelines2, my_pipeline2, my_layers2 = build_pipelines(new_ranks_re,
                                                           my_layers)
    dp2 = DataParallelEngine(pipelines2)
    new_grids = [p.rank_grid for p in pipelines2]
    copy_model_states(old_grids_re, new_grids, my_layers2, dp2)

    # verify: every layer this rank owns now equals the BROADCAST SOURCE's
    # marker (or its own, when its rank set did not change)
    for lid, layer in my_layers2.items():
        new_owner_lists = [g[lid] for g in new_grids]
        old_owner_lists = [g[lid] for g in old_grids_re]
        if all(r in old_owner_lists for r in new_owner_lists):
            # unchanged: kept whatever this rank had (its own marker) if it
            # owned the layer before, else it is not reachable here
            continue
        alive = [rs for rs in old_owner_lists if rs in new_owner_lists]
        src_old = [o for o, n in remap.items() if n == alive[0][0]][0]
        torch.testing.assert_close(layer.flat_param,
                                   base_flats[lid] + src_old)

    my_pipeline2.train()
    recovery_s = time.perf_counter() - t0
    assert my_pipeline2.execution is not None
    if my_pipeline2.is_last_stage():
        assert torch.isfinite(my_pipeline2.execution.total_loss).all()
    if new_rank == 0:
        print(f"[drill] gloo recovery (plan+rebuild+copy+1 step): "
              f"{recovery_s:.2f}s")
    dist.barrier()
    dist.destroy_process_group()


def test_reconfiguration_drill(tmp_path):
    mp.spawn(_drill, args=(5, str(tmp_path)), nprocs=5, join=True)
