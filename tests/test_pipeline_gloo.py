# Multi-process CPU tests (gloo, world_size 2) of the distributed host
# logic: 2-stage pipeline p2p + schedule, per-layer DP all-reduce, and the
# reconfiguration layer-copy broadcast — with OracleLayer standing in for
# the HIP layers (compute parity of the HIP path itself is covered by the
# @gpu tests).  Mirrors the reference's multi-process test pattern
# (tests/conftest.py:347-474 there) with a FileStore rendezvous.
from __future__ import annotations

import os
import pathlib
import sys

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

REPO_ROOT = pathlib.Path(__file__).resolve().parent.parent

TINY = dict(n_embd=96, n_head=4, n_layer=3, n_positions=64, vocab_size=211)
B, S, MB = 2, 32, 2  # microbatch size, seq, microbatches per pipeline


def _setup(rank: int, world: int, tmp: str):
    if str(REPO_ROOT) not in sys.path:
        sys.path.insert(0, str(REPO_ROOT))
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    dist.init_process_group(
        "gloo", init_method=f"file://{tmp}/rdzv", rank=rank, world_size=world)


def _configs():
    from oobleck_amd.config import ModelConfig, TrainingConfig
    from oracle.gpt2_oracle import OracleConfig
    mc = ModelConfig(**TINY)
    oc = OracleConfig(**TINY)
    tc = TrainingConfig(microbatch_size=B, global_microbatch_size=B * MB,
                        seq_len=S)
    return mc, oc, tc


def _flats(oc, seed=42):
    from oracle.gpt2_oracle import init_layer_params
    return [init_layer_params(oc, oc.layer_kind(i), seed * 1000 + i)
            for i in range(oc.n_layers_total)]


def _reference_grads(oc, flats, batches):
    """Single-process oracle: per-layer grads accumulated (summed) over the
    microbatches, matching the pipeline's accumulation semantics."""
    from oracle.gpt2_oracle import stage_forward_backward
    grads = [torch.zeros_like(f) for f in flats]
    losses = []
    for ids, labels in batches:
        loss, _, gs = stage_forward_backward(
            oc, flats, list(range(oc.n_layers_total)), ids, labels=labels)
        losses.append(loss)
        for g, gi in zip(grads, gs):
            g += gi
    return losses, grads


def _batches(cfg, n, seed):
    g = torch.Generator().manual_seed(seed)
    out = []
    for _ in range(n):
        ids = torch.randint(0, cfg.vocab_size, (B, S), generator=g)
        out.append((ids, ids.clone()))
    return out


# ---------------------------------------------------------------------------
# 2-stage pipeline parity
# ---------------------------------------------------------------------------

def _run_2stage(rank: int, world: int, tmp: str):
    _setup(rank, world, tmp)
    from oobleck_amd.engine import make_rank_grid
    from oobleck_amd.pipeline import OobleckPipeline
    from tests.oracle_layer import NoOpOptimizer, OracleLayer

    mc, oc, tc = _configs()
    flats = _flats(oc)
    grid = make_rank_grid(oc.n_layers_total, [[0, 1, 2], [3, 4]], [[0], [1]])

    class Loader:
        def __iter__(self):
            return iter({"input_ids": i, "labels": l}
                        for i, l in _batches(oc, MB, seed=7))

    pipe = OobleckPipeline(0, grid, mc, tc, Loader(), MB,
                           torch.device("cpu"))
    pipe.initialize_distributed_fsdp()
    pipe.initialize_distributed_pipeline()
    pipe.initialize_execution(
        layer_factory=lambda lid, pg, n_slots: OracleLayer(lid, oc, flats[lid]),
        optimizer_factory=lambda layers: (NoOpOptimizer(layers), None))
    pipe.train()

    losses_ref, grads_ref = _reference_grads(oc, flats, _batches(oc, MB, seed=7))
    if pipe.is_last_stage():
        total_ref = sum(l.item() for l in losses_ref)
        assert abs(pipe.execution.total_loss.item() - total_ref) < 1e-4 * abs(total_ref)
    for layer in pipe.execution._layers:
        torch.testing.assert_close(layer.flat_grad, grads_ref[layer.layer_id],
                                   rtol=1e-4, atol=1e-5)
    dist.barrier()
    dist.destroy_process_group()


# ---------------------------------------------------------------------------
# DP all-reduce across two 1-stage pipelines
# ---------------------------------------------------------------------------

def _run_dp(rank: int, world: int, tmp: str):
    _setup(rank, world, tmp)
    from oobleck_amd.engine import DataParallelEngine, make_rank_grid
    from oobleck_amd.pipeline import OobleckPipeline
    from tests.oracle_layer import NoOpOptimizer, OracleLayer

    mc, oc, tc = _configs()
    flats = _flats(oc)
    L = oc.n_layers_total
    all_layers = [list(range(L))]
    grids = [make_rank_grid(L, all_layers, [[0]]),
             make_rank_grid(L, all_layers, [[1]])]

    def loader_for(pid):
        class Loader:
            def __iter__(self):
                return iter({"input_ids": i, "labels": l}
                            for i, l in _batches(oc, 1, seed=100 + pid))
        return Loader()

    pipelines = []
    my_pipeline = None
    for pid, grid in enumerate(grids):
        p = OobleckPipeline(pid, grid, mc, tc, loader_for(pid), 1,
                            torch.device("cpu"))
        p.initialize_distributed_fsdp()
        p.initialize_distributed_pipeline()
        if p.my_pipeline:
            p.initialize_execution(
                layer_factory=lambda lid, pg, n_slots: OracleLayer(lid, oc, flats[lid]),
                optimizer_factory=lambda layers: (NoOpOptimizer(layers), None))
            my_pipeline = p
        pipelines.append(p)

    dp = DataParallelEngine(pipelines)
    my_pipeline.train()
    dp.do_allreduce(my_pipeline)

    # expected: sum of both pipelines' grads (reference all-reduces SUM)
    _, g0 = _reference_grads(oc, flats, _batches(oc, 1, seed=100))
    _, g1 = _reference_grads(oc, flats, _batches(oc, 1, seed=101))
    for layer in my_pipeline.execution._layers:
        torch.testing.assert_close(
            layer.flat_grad, g0[layer.layer_id] + g1[layer.layer_id],
            rtol=1e-4, atol=1e-5)
    dist.barrier()
    dist.destroy_process_group()


# ---------------------------------------------------------------------------
# reconfiguration layer-copy broadcast (engine.py:238-309 semantics)
# ---------------------------------------------------------------------------

def _run_copy(rank: int, world: int, tmp: str):
    _setup(rank, world, tmp)
    from oobleck_amd.engine import (DataParallelEngine, copy_model_states,
                                    make_rank_grid)
    from tests.oracle_layer import OracleLayer

    mc, oc, _tc = _configs()
    flats = _flats(oc)
    L = oc.n_layers_total
    all_layers = [list(range(L))]
    # new configuration: pipeline0 on rank0 (survivor), pipeline1 on rank1
    # (fresh worker whose layer states must be copied)
    new_grids = [make_rank_grid(L, all_layers, [[0]]),
                 make_rank_grid(L, all_layers, [[1]])]
    old_grids = [make_rank_grid(L, all_layers, [[0]])]

    class FakePipe:
        def __init__(self, grid):
            self.rank_grid = grid
    dp = DataParallelEngine([FakePipe(g) for g in new_grids])

    if rank == 0:
        my_layers = {lid: OracleLayer(lid, oc, flats[lid]) for lid in range(L)}
    else:
        my_layers = {lid: OracleLayer(lid, oc, torch.zeros_like(flats[lid]))
                     for lid in range(L)}

    copy_model_states(old_grids, new_grids, my_layers, dp)

    for lid in range(L):
        torch.testing.assert_close(my_layers[lid].flat_param, flats[lid])
    dist.barrier()
    dist.destroy_process_group()


# ---------------------------------------------------------------------------
# FULL_SHARD intra-stage sharding (§8 f1): one stage on 2 fsdp ranks — both
# ranks hold parameter shards, all-gather on unshard, per-microbatch
# reduce-scatter of grads.  Parity vs the single-process oracle.
# ---------------------------------------------------------------------------

def _run_fsdp(rank: int, world: int, tmp: str):
    _setup(rank, world, tmp)
    from oobleck_amd.engine import DataParallelEngine, make_rank_grid
    from oobleck_amd.pipeline import OobleckPipeline
    from tests.oracle_layer import NoOpOptimizer, OracleLayer

    mc, oc, tc = _configs()
    flats = _flats(oc)
    L = oc.n_layers_total
    grid = make_rank_grid(L, [list(range(L))], [[0, 1]])

    class Loader:
        def __iter__(self):
            return iter({"input_ids": i, "labels": l}
                        for i, l in _batches(oc, MB, seed=11))

    pipe = OobleckPipeline(0, grid, mc, tc, Loader(), MB, torch.device("cpu"))
    pipe.initialize_distributed_fsdp()
    pipe.initialize_distributed_pipeline()
    pipe.initialize_execution(
        layer_factory=lambda lid, pg, n_slots: OracleLayer(
            lid, oc, flats[lid], process_group=pg),
        optimizer_factory=lambda layers: (NoOpOptimizer(layers), None))
    dp = DataParallelEngine([pipe])
    pipe.train()
    dp.do_allreduce(pipe)

    losses_ref, grads_ref = _reference_grads(oc, flats, _batches(oc, MB, seed=11))
    total_ref = sum(l.item() for l in losses_ref)
    assert abs(pipe.execution.total_loss.item() - total_ref) < 1e-4 * abs(total_ref)
    for layer in pipe.execution._layers:
        sh = layer._sharded
        full_ref = torch.zeros(sh.padded)
        # reference-faithful semantics: every fsdp rank of a stage consumes
        # the SAME microbatches (the sampler keys on pipeline_index,
        # dataloader.py:43-100), so the post-backward reduce-scatter SUM
        # yields world_size x the single-rank gradient — the reference
        # never rescales (layer.py:199-223).
        full_ref[:sh.n_params] = world * grads_ref[layer.layer_id]
        expect = full_ref[rank * sh.shard_size:(rank + 1) * sh.shard_size]
        torch.testing.assert_close(layer.flat_grad, expect,
                                   rtol=1e-4, atol=1e-5)
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.parametrize("target", [_run_2stage, _run_dp, _run_copy, _run_fsdp],
                         ids=["2stage_parity", "dp_allreduce", "layer_copy",
                              "fsdp_full_shard"])
def test_multiprocess(target, tmp_path):
    mp.spawn(target, args=(2, str(tmp_path)), nprocs=2, join=True)


# ---------------------------------------------------------------------------
# the 8-GPU topology (4 stages x 2 DP replicas) on gloo world 8 — the exact
# shape bench.py builds for the driver's N=8 scaling run (TOPOLOGY[8])
# ---------------------------------------------------------------------------

def _run_4x2(rank: int, world: int, tmp: str):
    _setup(rank, world, tmp)
    from oobleck_amd.engine import (DataParallelEngine, even_stage_split,
                                    make_rank_grid)
    from oobleck_amd.pipeline import OobleckPipeline
    from tests.oracle_layer import NoOpOptimizer, OracleLayer

    mc, oc, tc0 = _configs()
    from oobleck_amd.config import TrainingConfig
    stages, replicas = 4, 2
    tc = TrainingConfig(microbatch_size=B,
                        global_microbatch_size=B * MB * replicas, seq_len=S)
    flats = _flats(oc)
    stage_layers = even_stage_split(mc, stages)

    def loader_for(pid):
        class Loader:
            def __iter__(self):
                return iter({"input_ids": i, "labels": l}
                            for i, l in _batches(oc, MB, seed=7 + pid))
        return Loader()

    pipelines, my_pipeline = [], None
    for pid in range(replicas):
        ranks = [pid * stages + s for s in range(stages)]
        grid = make_rank_grid(oc.n_layers_total, stage_layers,
                              [[r] for r in ranks])
        pipe = OobleckPipeline(pid, grid, mc, tc, loader_for(pid), MB,
                               torch.device("cpu"))
        pipe.initialize_distributed_fsdp()
        pipe.initialize_distributed_pipeline()
        pipelines.append(pipe)
    for p in pipelines:
        if p.my_pipeline:
            p.initialize_execution(
                layer_factory=lambda lid, pg, n_slots: OracleLayer(
                    lid, oc, flats[lid]),
                optimizer_factory=lambda layers: (NoOpOptimizer(layers),
                                                  None))
            my_pipeline = p
    dp = DataParallelEngine(pipelines)
    assert my_pipeline is not None
    my_pipeline.train()
    dp.do_allreduce(my_pipeline)

    # reference: grads summed over BOTH replicas' microbatches (DP SUM)
    losses0, g0 = _reference_grads(oc, flats, _batches(oc, MB, seed=7))
    losses1, g1 = _reference_grads(oc, flats, _batches(oc, MB, seed=8))
    for layer in my_pipeline.execution._layers:
        ref = g0[layer.layer_id] + g1[layer.layer_id]
        torch.testing.assert_close(layer.flat_grad, ref, rtol=1e-4,
                                   atol=1e-5)
    if my_pipeline.is_last_stage():
        ref_losses = losses0 if rank < stages else losses1
        total_ref = sum(l.item() for l in ref_losses)
        got = my_pipeline.execution.total_loss.item()
        assert abs(got - total_ref) < 1e-4 * abs(total_ref), (got, total_ref)
    dist.barrier()
    dist.destroy_process_group()


def test_multiprocess_4x2(tmp_path):
    mp.spawn(_run_4x2, args=(8, str(tmp_path)), nprocs=8, join=True)
