# @gpu pp>1 composition test: OobleckPipeline.train() driving REAL HIP
# Layer objects at pp2 — schedule + p2p + HIP compute + optimizer, the
# exact composition the multi-GPU bench depends on (reference
# pipeline.py:288-427 + :458-487), previously covered only piecewise
# (gloo tests used OracleLayer; GPU tests drove layers directly).
#
# Two processes SHARE cuda:0 (the round's lease is one GPU): rendezvous is
# gloo, so the p2p path stages activations/grads through host memory
# (PipelineCommunication._gloo_cuda) — same instruction order, same wire
# content as the RCCL path.
from __future__ import annotations


import os
import pathlib
import sys

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs MI355X")

REPO_ROOT = pathlib.Path(__file__).resolve().parent.parent

# head_dim 64 + S % 128 == 0 -> the bf16 leg runs the flash-attention path
DIMS = dict(n_embd=128, n_head=2, n_layer=2, n_positions=128, vocab_size=307)
B, S, MB = 2, 128, 4


def _setup(rank: int, world: int, tmp: str):
    if str(REPO_ROOT) not in sys.path:
        sys.path.insert(0, str(REPO_ROOT))
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    dist.init_process_group(
        "gloo", init_method=f"file://{tmp}/rdzv", rank=rank, world_size=world)
    torch.cuda.set_device(0)
    # each rank recomputes the CPU oracle reference: don't oversubscribe
    torch.set_num_threads(max(2, (os.cpu_count() or 8) // (2 * world)))


def _batches(vocab, n, seed):
    g = torch.Generator().manual_seed(seed)
    out = []
    for _ in range(n):
        ids = torch.randint(0, vocab, (B, S), generator=g)
        out.append((ids, ids.clone()))
    return out


def _run_pp2(rank: int, world: int, tmp: str, dtype: str):
    _setup(rank, world, tmp)
    from oracle.gpt2_oracle import OracleConfig, adamw_step, init_layer_params
    from oracle.gpt2_oracle import stage_forward_backward

    from oobleck_amd.config import ModelConfig, TrainingConfig
    from oobleck_amd.engine import DataParallelEngine, make_rank_grid
    from oobleck_amd.layer import Layer
    from oobleck_amd.optimizer import FusedAdamW, WarmupLR
    from oobleck_amd.pipeline import OobleckPipeline

    dev = torch.device("cuda", 0)
    mc = ModelConfig(**DIMS)
    oc = OracleConfig(**DIMS)
    tc = TrainingConfig(microbatch_size=B, global_microbatch_size=B * MB,
                        seq_len=S, lr=1e-3)
    L = oc.n_layers_total
    flats = [init_layer_params(oc, oc.layer_kind(i), 42 * 100 + i)
             for i in range(L)]
    # stage 0: embedding + block 1; stage 1: block 2 + final
    grid = make_rank_grid(L, [[0, 1], [2, 3]], [[0], [1]])

    class Loader:
        def __iter__(self):
            return iter({"input_ids": i, "labels": l}
                        for i, l in _batches(oc.vocab_size, MB, seed=7))

    pipe = OobleckPipeline(0, grid, mc, tc, Loader(), MB, dev)
    pipe.initialize_distributed_fsdp()
    pipe.initialize_distributed_pipeline()

    def layer_factory(lid, pg, n_slots):
        layer = Layer(lid, mc, B, S, n_slots, dev, dtype=dtype)
        layer.flat_param.copy_(flats[lid].to(dev))
        layer.refresh_weights()
        return layer

    def optimizer_factory(layers):
        opt = FusedAdamW(layers, lr=tc.lr)
        return opt, WarmupLR(opt, 0)

    pipe.initialize_execution(layer_factory, optimizer_factory)
    dp = DataParallelEngine([pipe])

    pipe.train()
    dp.do_allreduce(pipe)
    torch.cuda.synchronize()

    # single-process oracle reference: grads summed over the microbatches
    grads_ref = [torch.zeros_like(f) for f in flats]
    losses_ref = []
    for ids, labels in _batches(oc.vocab_size, MB, seed=7):
        loss, _, gs = stage_forward_backward(oc, flats, list(range(L)), ids,
                                             labels=labels)
        losses_ref.append(loss)
        for g, gi in zip(grads_ref, gs):
            g += gi

    if pipe.is_last_stage():
        total_ref = sum(l.item() for l in losses_ref)
        got = pipe.execution.total_loss.item()
        tol = 1e-4 if dtype == "f32" else 2e-2
        assert abs(got - total_ref) < tol * abs(total_ref), (got, total_ref)
    for layer in pipe.execution._layers:
        got = layer.flat_grad.cpu()
        ref = grads_ref[layer.layer_id]
        if dtype == "f32":
            torch.testing.assert_close(got, ref, rtol=1e-3, atol=1e-3)
        else:
            rel = (got - ref).norm() / ref.norm().clamp_min(1e-12)
            assert rel < 8e-2, (layer.layer_id, rel.item())

    # optimizer step through the C-ABI fused AdamW; parity vs the oracle's
    # adamw on the same accumulated grads
    pipe.execution.optimizer_step()
    torch.cuda.synchronize()
    for layer in pipe.execution._layers:
        lid = layer.layer_id
        p = flats[lid].clone()
        m = torch.zeros_like(p)
        v = torch.zeros_like(p)
        adamw_step(p, layer.flat_grad.cpu(), m, v, step=1, lr=tc.lr,
                   beta1=tc.adam_beta1, beta2=tc.adam_beta2,
                   eps=tc.adam_eps, weight_decay=tc.weight_decay)
        torch.testing.assert_close(layer.flat_param.cpu(), p, rtol=1e-5,
                                   atol=1e-6)

    dist.barrier()
    dist.destroy_process_group()


@requires_gpu
@pytest.mark.parametrize("dtype", ["f32", "bf16"])
def test_pp2_hip_end_to_end(dtype, tmp_path):
    """pp2 with HIP compute: two ranks on one GPU, full 1F1B schedule."""
    mp.spawn(_run_pp2, args=(2, str(tmp_path), dtype), nprocs=2, join=True)


def _run_pp1_overlap(rank: int, world: int, tmp: str, dtype: str = "bf16"):
    """pp1 with the fwd/bwd dual-stream overlap (default on): the full
    OobleckPipeline.train() against the oracle — the exact path bench.py
    times at N=1 (PipelineExecution._overlap engages automatically:
    cuda + single stage + no FSDP)."""
    _setup(rank, world, tmp)
    from oracle.gpt2_oracle import OracleConfig, init_layer_params
    from oracle.gpt2_oracle import stage_forward_backward

    from oobleck_amd.config import ModelConfig, TrainingConfig
    from oobleck_amd.engine import DataParallelEngine, make_rank_grid
    from oobleck_amd.layer import Layer
    from oobleck_amd.optimizer import FusedAdamW, WarmupLR
    from oobleck_amd.pipeline import OobleckPipeline

    dev = torch.device("cuda", 0)
    mc = ModelConfig(**DIMS)
    oc = OracleConfig(**DIMS)
    tc = TrainingConfig(microbatch_size=B, global_microbatch_size=B * MB,
                        seq_len=S, lr=1e-3)
    L = oc.n_layers_total
    flats = [init_layer_params(oc, oc.layer_kind(i), 42 * 100 + i)
             for i in range(L)]
    grid = make_rank_grid(L, [list(range(L))], [[0]])

    class Loader:
        def __iter__(self):
            return iter({"input_ids": i, "labels": l}
                        for i, l in _batches(oc.vocab_size, MB, seed=31))

    pipe = OobleckPipeline(0, grid, mc, tc, Loader(), MB, dev)
    pipe.initialize_distributed_fsdp()
    pipe.initialize_distributed_pipeline()

    def layer_factory(lid, pg, n_slots):
        layer = Layer(lid, mc, B, S, n_slots, dev, dtype=dtype)
        layer.flat_param.copy_(flats[lid].to(dev))
        layer.refresh_weights()
        return layer

    def optimizer_factory(layers):
        opt = FusedAdamW(layers, lr=tc.lr)
        return opt, WarmupLR(opt, 0)

    pipe.initialize_execution(layer_factory, optimizer_factory)
    assert pipe.execution._overlap, "overlap should engage at pp1 on cuda"
    dp = DataParallelEngine([pipe])
    pipe.train()
    dp.do_allreduce(pipe)
    torch.cuda.synchronize()

    grads_ref = [torch.zeros_like(f) for f in flats]
    losses_ref = []
    for ids, labels in _batches(oc.vocab_size, MB, seed=31):
        loss, _, gs = stage_forward_backward(oc, flats, list(range(L)), ids,
                                             labels=labels)
        losses_ref.append(loss)
        for g, gi in zip(grads_ref, gs):
            g += gi
    total_ref = sum(l.item() for l in losses_ref)
    got = pipe.execution.total_loss.item()
    ltol = 1e-4 if dtype == "f32" else 2e-2
    assert abs(got - total_ref) < ltol * abs(total_ref), (got, total_ref)
    for layer in pipe.execution._layers:
        gotg = layer.flat_grad.cpu()
        ref = grads_ref[layer.layer_id]
        if dtype == "f32":
            torch.testing.assert_close(gotg, ref, rtol=1e-3, atol=1e-3)
        else:
            rel = (gotg - ref).norm() / ref.norm().clamp_min(1e-12)
            assert rel < 8e-2, (layer.layer_id, rel.item())
    dist.barrier()
    dist.destroy_process_group()


@requires_gpu
@pytest.mark.parametrize("dtype", ["bf16", "f32"])
def test_pp1_overlap_end_to_end(dtype, tmp_path):
    mp.spawn(_run_pp1_overlap, args=(1, str(tmp_path), dtype), nprocs=1,
             join=True)


def _run_pp2dp2(rank: int, world: int, tmp: str):
    """2 stages x 2 DP replicas with HIP compute, four ranks sharing one
    GPU — the single-GPU analog of BASELINE config[2]'s topology
    (stages x replicas), composing p2p + per-layer cross-pipeline DP
    all-reduce + HIP layers + fused AdamW in one run."""
    _setup(rank, world, tmp)
    from oracle.gpt2_oracle import OracleConfig, init_layer_params
    from oracle.gpt2_oracle import stage_forward_backward

    from oobleck_amd.config import ModelConfig, TrainingConfig
    from oobleck_amd.engine import DataParallelEngine, make_rank_grid
    from oobleck_amd.layer import Layer
    from oobleck_amd.optimizer import FusedAdamW, WarmupLR
    from oobleck_amd.pipeline import OobleckPipeline

    dev = torch.device("cuda", 0)
    mc = ModelConfig(**DIMS)
    oc = OracleConfig(**DIMS)
    tc = TrainingConfig(microbatch_size=B,
                        global_microbatch_size=B * MB * 2, seq_len=S)
    L = oc.n_layers_total
    flats = [init_layer_params(oc, oc.layer_kind(i), 42 * 100 + i)
             for i in range(L)]

    def loader_for(pid):
        class Loader:
            def __iter__(self):
                return iter({"input_ids": i, "labels": l}
                            for i, l in _batches(oc.vocab_size, MB,
                                                 seed=7 + pid))
        return Loader()

    pipelines, my_pipeline = [], None
    for pid in range(2):
        ranks = [pid * 2, pid * 2 + 1]
        grid = make_rank_grid(L, [[0, 1], [2, 3]], [[ranks[0]], [ranks[1]]])
        p = OobleckPipeline(pid, grid, mc, tc, loader_for(pid), MB, dev)
        p.initialize_distributed_fsdp()
        p.initialize_distributed_pipeline()
        pipelines.append(p)
    for p in pipelines:
        if p.my_pipeline:
            def layer_factory(lid, pg, n_slots):
                layer = Layer(lid, mc, B, S, n_slots, dev, dtype="bf16")
                layer.flat_param.copy_(flats[lid].to(dev))
                layer.refresh_weights()
                return layer

            def optimizer_factory(layers):
                opt = FusedAdamW(layers, lr=tc.lr)
                return opt, WarmupLR(opt, 0)
            p.initialize_execution(layer_factory, optimizer_factory)
            my_pipeline = p
    dp = DataParallelEngine(pipelines)
    my_pipeline.train()
    dp.do_allreduce(my_pipeline)
    torch.cuda.synchronize()

    # reference: per-layer grads summed over BOTH replicas' microbatches
    grads_ref = [torch.zeros_like(f) for f in flats]
    for pid in range(2):
        for ids, labels in _batches(oc.vocab_size, MB, seed=7 + pid):
            _, _, gs = stage_forward_backward(oc, flats, list(range(L)),
                                              ids, labels=labels)
            for g, gi in zip(grads_ref, gs):
                g += gi
    for layer in my_pipeline.execution._layers:
        got = layer.flat_grad.cpu()
        ref = grads_ref[layer.layer_id]
        rel = (got - ref).norm() / ref.norm().clamp_min(1e-12)
        assert rel < 8e-2, (layer.layer_id, rel.item())
    dist.barrier()
    dist.destroy_process_group()


@requires_gpu
def test_pp2dp2_hip_end_to_end(tmp_path):
    """config[2]'s stages-x-replicas composition on one GPU (4 ranks)."""
    mp.spawn(_run_pp2dp2, args=(4, str(tmp_path)), nprocs=4, join=True)
