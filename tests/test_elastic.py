# End-to-end elastic kill-an-agent drill THROUGH THE CONTROL PLANE (CPU):
# the config[4] flow with the real master/agent wire protocol instead of
# test-harness-injected failures (VERDICT round-1 missing #5).
#
#   master (asyncio TCP, oobleck_amd/elastic.py — the reference protocol
#   restated; its own modules need deepspeed/asyncssh/simple_parsing,
#   absent here)
#     └ 3 agents (processes) x 1 worker each: register -> receive job
#       args -> spawn worker with an mp.Pipe -> forward the rank-0
#       rendezvous port up to the master, which broadcasts it to every
#       agent -> workers gloo-init via TCPStore, build the [[0,1],[2]]
#       pipelines (OracleLayer), run a 1F1B step.
#   the test SIGKILLs agent 1  ->  master sees the disconnection and
#   broadcasts Response.RECONFIGURATION + the lost identity
#   (master.py:192-204)  ->  surviving agents signal their workers down
#   the pipes (agent.py:214-222 / engine.py:59-90 listener semantics)
#   ->  workers destroy the world, run the rank surgery
#   (compute_new_ranks_list: [[0,1],[2]] - rank 1 -> merge [[0,2]]),
#   re-rendezvous at world 2 through the SAME port-forwarding flow,
#   rebuild pipelines, copy_model_states, and run a post-recovery step.
from __future__ import annotations

import json
import os
import pathlib
import signal
import sys
import time

import pytest
import torch
import torch.distributed as dist

REPO_ROOT = pathlib.Path(__file__).resolve().parent.parent

TINY = dict(n_embd=96, n_head=4, n_layer=3, n_positions=64, vocab_size=211)
B, S = 2, 32
WPA = 1  # workers per agent
NAGENTS = 3


def _free_port() -> int:
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _pipe_wait(pipe, tag, timeout=60.0):
    """Wait for a tagged message, tolerating unrelated broadcasts."""
    t0 = time.monotonic()
    while time.monotonic() - t0 < timeout:
        if pipe.poll(0.2):
            msg = pipe.recv()
            if msg[0] == tag:
                return msg[1]
    raise TimeoutError(f"no '{tag}' message within {timeout}s")


def _rendezvous(pipe, world: int, my_new_rank: int):
    """The reference's initialize_distributed flow (engine.py:526-598):
    rank 0 owns the TCPStore, its port travels worker -> agent -> master
    -> every agent -> every worker."""
    if my_new_rank == 0:
        port = _free_port()
        # wait_for_workers=False: the port must travel through the control
        # plane BEFORE clients can join — a waiting constructor deadlocks
        store = dist.TCPStore("127.0.0.1", port, world, True,
                              timeout=__import__("datetime")
                              .timedelta(seconds=60),
                              wait_for_workers=False)
        pipe.send(("port_out", port))
        # consume our own port broadcast so it doesn't confuse later waits
        _pipe_wait(pipe, "port")
    else:
        port = _pipe_wait(pipe, "port")
        store = dist.TCPStore("127.0.0.1", port, world, False, timeout=
                              __import__("datetime").timedelta(seconds=60))
    dist.init_process_group("gloo", store=store, rank=my_new_rank,
                            world_size=world)
    return store


def _build(ranks_lists, prev_layers, mc, oc, tc, mb):
    from oobleck_amd.engine import (DataParallelEngine, even_stage_split,
                                    make_rank_grid)
    from oobleck_amd.pipeline import OobleckPipeline
    from tests.oracle_layer import NoOpOptimizer, OracleLayer
    from oracle.gpt2_oracle import init_layer_params
    L = oc.n_layers_total
    pipelines, my_pipeline = [], None
    my_layers = {}
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

        p = OobleckPipeline(pid, grid, mc, tc, Loader(), mb,
                            torch.device("cpu"))
        p.initialize_distributed_fsdp()
        p.initialize_distributed_pipeline()
        if p.my_pipeline:
            def factory(lid, pg, n_slots):
                prev = prev_layers.get(lid)
                flat = (prev.flat_param.detach().clone()
                        if prev is not None else
                        init_layer_params(oc, oc.layer_kind(lid), 55 + lid))
                layer = OracleLayer(lid, oc, flat)
                my_layers[lid] = layer
                return layer
            p.initialize_execution(
                layer_factory=factory,
                optimizer_factory=lambda layers: (NoOpOptimizer(layers),
                                                  None))
            my_pipeline = p
        pipelines.append(p)
    dp = DataParallelEngine(pipelines)
    return pipelines, my_pipeline, my_layers, dp


def drill_worker(local_rank: int, agent_id: int, pipe, args) -> None:
    """The reference worker_main's steps (elastic/worker.py:13-34) over
    our engine pieces, with the engine's reconfiguration-listener
    semantics (engine.py:59-90) inlined as a pipe poll."""
    if str(REPO_ROOT) not in sys.path:
        sys.path.insert(0, str(REPO_ROOT))
    torch.set_num_threads(2)
    from oobleck_amd.config import ModelConfig, TrainingConfig
    from oobleck_amd.engine import copy_model_states
    from oobleck_amd.reconfigure import compute_new_ranks_list
    from oracle.gpt2_oracle import OracleConfig

    try:
        dist_info = _pipe_wait(pipe, "dist")
        mc, oc = ModelConfig(**args.model_dims), OracleConfig(**args.model_dims)
        tc = TrainingConfig(microbatch_size=B,
                            global_microbatch_size=B * args.microbatches,
                            seq_len=S)
        my_rank = local_rank
        _rendezvous(pipe, dist_info.world_size, my_rank)
        pipelines, my_pipeline, my_layers, dp = _build(
            args.ranks_lists, {}, mc, oc, tc, args.microbatches)
        my_pipeline.train()
        dp.do_allreduce(my_pipeline)
        dist.barrier()
        loss1 = (my_pipeline.execution.total_loss.item()
                 if my_pipeline.is_last_stage() else None)
        pathlib.Path(args.result_dir, f"phase1_rank{my_rank}").write_text("ok")

        # reconfiguration listener (engine.py:59-90): wait for the lost-
        # agent notification forwarded by our agent
        lost_agent = _pipe_wait(pipe, "lost", timeout=120)
        lost_ranks = [lost_agent * args.workers_per_agent + i
                      for i in range(args.workers_per_agent)]
        dist.destroy_process_group()
        new_ranks = compute_new_ranks_list(
            [list(r) for r in args.ranks_lists], lost_ranks,
            min_num_ranks=args.min_num_ranks)
        survivors = sorted(r for rl in args.ranks_lists for r in rl
                           if r not in lost_ranks)
        remap = {old: new for new, old in enumerate(survivors)}
        new_ranks_re = [[remap[r] for r in rl] for rl in new_ranks]
        _rendezvous(pipe, len(survivors), remap[my_rank])

        from oobleck_amd.engine import even_stage_split, make_rank_grid
        old_grids_re = []
        for rl in args.ranks_lists:
            stages = even_stage_split(mc, len(rl))
            g = make_rank_grid(oc.n_layers_total, stages,
                               [[r] for r in rl])
            old_grids_re.append({lid: [remap.get(r, -1) for r in rs]
                                 for lid, rs in g.items()})
        pipelines2, my_pipeline2, my_layers2, dp2 = _build(
            new_ranks_re, my_layers, mc, oc, tc, args.microbatches)
        new_grids = [p.rank_grid for p in pipelines2]
        copy_model_states(old_grids_re, new_grids, my_layers2, dp2)
        my_pipeline2.train()
        dp2.do_allreduce(my_pipeline2)
        dist.barrier()
        loss2 = (my_pipeline2.execution.total_loss.item()
                 if my_pipeline2.is_last_stage() else None)
        out = {"rank": my_rank, "new_rank": remap[my_rank],
               "new_ranks": new_ranks, "loss1": loss1, "loss2": loss2,
               "finite": bool(loss2 is None or
                              torch.isfinite(torch.tensor(loss2)))}
        pathlib.Path(args.result_dir, f"done_rank{my_rank}.json").write_text(
            json.dumps(out))
        dist.destroy_process_group()
    except EOFError:
        # our agent died (we are on the lost node): exit quietly
        pass


def _agent_entry(agent_id: int, master_port: int):
    if str(REPO_ROOT) not in sys.path:
        sys.path.insert(0, str(REPO_ROOT))
    from oobleck_amd.elastic import agent_main
    agent_main(agent_id, master_port, drill_worker)


def test_elastic_kill_agent_drill(tmp_path):
    import multiprocessing
    from oobleck_amd.elastic import ElasticMaster, JobArgs

    args = JobArgs(num_agents=NAGENTS, workers_per_agent=WPA,
                   model_dims=TINY, ranks_lists=[[0, 1], [2]],
                   min_num_ranks=2, microbatches=2,
                   result_dir=str(tmp_path))
    master = ElasticMaster(args)
    port = master.start_in_thread()
    assert port

    ctx = multiprocessing.get_context("spawn")
    agents = [ctx.Process(target=_agent_entry, args=(i, port))
              for i in range(NAGENTS)]
    for a in agents:
        a.start()

    # phase 1: all three workers complete a step through the control plane
    deadline = time.monotonic() + 180
    while time.monotonic() < deadline:
        if all((tmp_path / f"phase1_rank{r}").exists() for r in range(3)):
            break
        time.sleep(0.5)
    else:
        pytest.fail("phase 1 did not complete")

    # the failure: agent 1 (rank 1) dies hard
    os.kill(agents[1].pid, signal.SIGKILL)

    # recovery: survivors 0 and 2 merge into [[0, 2]] and finish a step
    deadline = time.monotonic() + 180
    while time.monotonic() < deadline:
        if all((tmp_path / f"done_rank{r}.json").exists() for r in (0, 2)):
            break
        time.sleep(0.5)
    else:
        pytest.fail("recovery did not complete")

    for r in (0, 2):
        out = json.loads((tmp_path / f"done_rank{r}.json").read_text())
        assert out["new_ranks"] == [[0, 2]]
        assert out["finite"]
    for a in agents:
        if a.is_alive():
            a.terminate()
        a.join(timeout=30)
