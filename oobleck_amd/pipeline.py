# OobleckPipeline / PipelineExecution / PipelineCommunication — the host
# orchestration of the hot path, mirroring the reference's public surface
# (/root/reference/oobleck/execution/pipeline.py) over the C-ABI layers.
#
# Differences by design (MI355X-first), documented in DESIGN.md:
#   * compute is the HIP extension, not torch autograd: backward_pass drives
#     layers in reverse with explicit activation-grad buffers instead of
#     autograd.backward(outputs, grads) (layer.py:250-260 semantics kept).
#   * the per-tensor int64 meta handshake of pipeline.py:293-323 (1+4n
#     separate sends) is collapsed into ONE packed int64 descriptor per pair
#     — same information, one message.
from __future__ import annotations

import torch
import torch.distributed as dist

from .config import ModelConfig, TrainingConfig
from .params import KIND_EMBED, KIND_FINAL
from .schedule import (BackwardPass, ForwardPass, LoadMicroBatch,
                       OobleckPipelineSchedule, RecvActivation, RecvGrad,
                       SendActivation, SendGrad)

# wire dtype table — part of the reference's p2p protocol
# (/root/reference/oobleck/execution/utils.py:4-18); order preserved.
ID_TO_DTYPE = [
    torch.float32, torch.float64, torch.complex64, torch.complex128,
    torch.float16, torch.bfloat16, torch.uint8, torch.int8, torch.int16,
    torch.int32, torch.int64, torch.bool,
]
DTYPE_TO_ID = {d: i for i, d in enumerate(ID_TO_DTYPE)}

_META_MAX_TENSORS = 4
_META_WORDS = 1 + _META_MAX_TENSORS * 7  # ntensors + per tensor: dtype, req_grad, ndims, 4 dims


class SyntheticDataLoader:
    """Synthetic replacement for the reference's HF-datasets loader
    (dataset.py:150-208 / dataloader.py) — no network in this environment.
    Yields {input_ids, labels} with labels = input_ids (dataset.py:201)."""

    def __init__(self, cfg: ModelConfig, batch: int, seq: int, seed: int = 42):
        self.cfg = cfg
        self.batch = batch
        self.seq = seq
        self.seed = seed
        self._gen = torch.Generator().manual_seed(seed)

    def __iter__(self):
        while True:
            ids = torch.randint(0, self.cfg.vocab_size, (self.batch, self.seq),
                                generator=self._gen)
            yield {"input_ids": ids, "labels": ids.clone()}


class PipelineExecution:
    """Per-instruction handlers (reference pipeline.py:87-244)."""

    def __init__(self, pipeline: "OobleckPipeline", layers, dataloader,
                 training_cfg: TrainingConfig, optimizer=None, lr_scheduler=None):
        self._pipeline_ref = pipeline
        self._layers = layers
        self._dataloader = dataloader
        self._data_iterator = iter(dataloader)
        self._tc = training_cfg
        self._loss: torch.Tensor | None = None
        self.total_loss: torch.Tensor | None = None
        self._optimizer = optimizer
        self._lr_scheduler = lr_scheduler
        import torch as _torch
        self._act_dtype = getattr(layers[0], "act_dtype", _torch.float32) \
            if layers else _torch.float32
        # lazily-allocated working buffers (indices 0/1: forward-pass
        # inter-layer temps, 2/3: backward-pass temps — disjoint so the
        # dual-stream overlap below can run a forward beside a backward)
        self._tmp: list[torch.Tensor | None] = [None, None, None, None]
        self._inbufs: dict[int, tuple] = {}  # per-slot device id/label bufs
        self._out_bufs: dict[int, torch.Tensor] = {}
        self._loss_bufs: dict[int, torch.Tensor] = {}
        self._din_bufs: dict[int, torch.Tensor] = {}
        # single-stage (pp1) GPU overlap: forward of microbatch m+1 runs
        # on its own HIP stream beside backward of microbatch m (the 1F1B
        # schedule at stages=1 alternates F/B).  Slot reuse and the
        # F(m)->B(m) edge are event-fenced; the optimizer/all-reduce epoch
        # re-joins both streams (train()).  Disabled under FSDP (its
        # collectives must keep a single well-ordered stream).
        # DEFAULT ON since round 2: the round-1 "hangs at 16 microbatches"
        # was hipBLASLt's shared handle under cross-stream enqueue depth
        # (fixed by per-stream Lt contexts, ob_blaslt.hip) plus pageable
        # H2D staging and per-microbatch device allocations on a side
        # stream (fixed by pinned + per-slot buffers here).  Measured:
        # 246.5 ms/step vs 267.8 without (+8.6%).  OB_PP1_OVERLAP=0
        # disables.
        import os
        self._overlap = (
            os.environ.get("OB_PP1_OVERLAP", "1") == "1"
            and pipeline.device.type == "cuda"
            and pipeline.is_first_stage() and pipeline.is_last_stage()
            and not any(getattr(l, "_sharded", None) is not None
                        for l in layers))
        if self._overlap:
            self._s_fwd = torch.cuda.Stream()
            self._s_bwd = torch.cuda.Stream()
            self._ev_fwd: dict[int, torch.cuda.Event] = {}
            self._ev_bwd: dict[int, torch.cuda.Event] = {}
            # FRESH event per record (64-deep ring): re-recording a
            # still-waited event serializes against the FUTURE record on
            # ROCm, which compounded with pipeline depth
            self._evt_ring = [torch.cuda.Event() for _ in range(64)]
            self._evt_i = 0

    def _fresh_event(self) -> "torch.cuda.Event":
        e = self._evt_ring[self._evt_i & 63]
        self._evt_i += 1
        return e

    @property
    def pipeline(self) -> "OobleckPipeline":
        return self._pipeline_ref

    # -- helpers ------------------------------------------------------------
    def _hidden_shape(self, batch: int):
        cfg = self.pipeline.model_cfg
        return (batch, self._tc.seq_len, cfg.n_embd)

    def _get_tmp(self, i: int, batch: int) -> torch.Tensor:
        shape = self._hidden_shape(batch)
        t = self._tmp[i]
        if t is None or t.shape[0] < batch:
            t = torch.empty(shape, dtype=self._act_dtype,
                            device=self.pipeline.device)
            self._tmp[i] = t
        return t[:batch]

    # -- instruction handlers ------------------------------------------------
    def load_microbatch(self, buffer_id: int) -> None:
        # reference pipeline.py:158-167: emitted on first AND last stage,
        # populates inputs only on the first stage (the last stage receives
        # labels through the activation tuple).
        assert self.pipeline.is_first_stage() or self.pipeline.is_last_stage()
        if self.pipeline.is_first_stage():
            batch = next(self._data_iterator)
            dev = self.pipeline.device
            import contextlib
            ctx = (torch.cuda.stream(self._s_fwd) if self._overlap
                   else contextlib.nullcontext())
            ids_cpu, labels_cpu = batch["input_ids"], batch["labels"]
            if self._overlap and dev.type == "cuda":
                # (a) pin the host side: a pageable H2D on a non-default
                # stream is staged SYNCHRONOUSLY against the stream's
                # backlog; (b) allocate the DEVICE buffers once per slot
                # and copy_ into them — fresh per-microbatch allocations
                # on a side stream push the caching allocator into
                # hipMalloc (device-wide sync) once the queues run deep:
                # the round-1 "overlap hangs at 16 microbatches" grew
                # ~quadratically with depth from exactly this.
                ids_cpu = ids_cpu.pin_memory()
                labels_cpu = labels_cpu.pin_memory()
                bufs = self._inbufs.get(buffer_id)
                if bufs is None:
                    bufs = (torch.empty(ids_cpu.shape, dtype=ids_cpu.dtype,
                                        device=dev),
                            torch.empty(labels_cpu.shape,
                                        dtype=labels_cpu.dtype, device=dev))
                    self._inbufs[buffer_id] = bufs
                with ctx:
                    bufs[0].copy_(ids_cpu, non_blocking=True)
                    bufs[1].copy_(labels_cpu, non_blocking=True)
                self.pipeline.pipe_buffers["inputs"][buffer_id] = bufs
                return
            with ctx:
                ids = ids_cpu.to(dev, non_blocking=True)
                labels = labels_cpu.to(dev, non_blocking=True)
            self.pipeline.pipe_buffers["inputs"][buffer_id] = (ids, labels)

    def forward_pass(self, buffer_id: int) -> None:
        if self._overlap:
            with torch.cuda.stream(self._s_fwd):
                ev = self._ev_bwd.get(buffer_id)
                if ev is not None:  # slot reuse: wait its previous backward
                    self._s_fwd.wait_event(ev)
                self._forward_impl(buffer_id)
                fev = self._fresh_event()
                fev.record(self._s_fwd)
                self._ev_fwd[buffer_id] = fev
        else:
            self._forward_impl(buffer_id)

    def _forward_impl(self, buffer_id: int) -> None:
        x, labels = self.pipeline.pipe_buffers["inputs"][buffer_id]
        batch = x.shape[0]
        n = len(self._layers)
        for i, layer in enumerate(self._layers):
            layer.set_batch(batch)
            if layer.kind == KIND_FINAL:
                out = self._loss_bufs.get(buffer_id)
                if out is None:
                    out = torch.zeros(1, dtype=torch.float32,
                                      device=self.pipeline.device)
                    self._loss_bufs[buffer_id] = out
                layer.forward_slot(buffer_id, x, out, labels)
            else:
                if i == n - 1:
                    out = self._out_bufs.get(buffer_id)
                    if out is None or out.shape[0] < batch:
                        out = torch.empty(self._hidden_shape(batch),
                                          dtype=self._act_dtype,
                                          device=self.pipeline.device)
                        self._out_bufs[buffer_id] = out
                    out = out[:batch]
                else:
                    out = self._get_tmp(i % 2, batch)
                layer.forward_slot(buffer_id, x, out)
            x = out

        if self.pipeline.is_last_stage():
            self._loss = self._loss_bufs[buffer_id]
            if self.total_loss is None:
                self.total_loss = torch.zeros_like(self._loss)
            self.total_loss += self._loss.detach()
        else:
            self.pipeline.pipe_buffers["outputs"][buffer_id] = (x, labels)

    def backward_pass(self, buffer_id: int) -> None:
        if self._overlap:
            with torch.cuda.stream(self._s_bwd):
                self._s_bwd.wait_event(self._ev_fwd[buffer_id])
                self._backward_impl(buffer_id)
                bev = self._fresh_event()
                bev.record(self._s_bwd)
                self._ev_bwd[buffer_id] = bev
        else:
            self._backward_impl(buffer_id)

    def _backward_impl(self, buffer_id: int) -> None:
        if self.pipeline.is_last_stage():
            dout = None  # final layer seeds dloss = 1.0 (layer.py:250-253)
        else:
            (dout,) = self.pipeline.communication.grad_recv_buf
        batch = dout.shape[0] if dout is not None else self._layers[0]._batch
        first_stage = self.pipeline.is_first_stage()
        for i in range(len(self._layers) - 1, -1, -1):
            layer = self._layers[i]
            if i == 0:
                if first_stage:
                    din = None  # embedding: int input, no activation grad
                else:
                    din = self._din_bufs.get(buffer_id)
                    if din is None or din.shape[0] < batch:
                        din = torch.empty(self._hidden_shape(batch),
                                          dtype=self._act_dtype,
                                          device=self.pipeline.device)
                        self._din_bufs[buffer_id] = din
                    din = din[:batch]
            else:
                din = self._get_tmp(2 + i % 2, batch)
            layer.backward_slot(buffer_id, dout, din)
            dout = din
        # free forward output (reference pipeline.py:235-239)
        self.pipeline.pipe_buffers["outputs"][buffer_id] = None
        self._loss = None

    def optimizer_step(self, lr_kwargs=None) -> None:
        self._optimizer.step()
        if self._lr_scheduler is not None:
            self._lr_scheduler.step()


class PipelineCommunication:
    """Activation/grad p2p between adjacent stages over the per-shard
    process group (reference pipeline.py:247-427), RCCL on ROCm.  The meta
    handshake is one packed int64 descriptor (vs 1+4n separate tensors)."""

    def __init__(self, pipeline: "OobleckPipeline", process_group,
                 prev_rank: int | None, next_rank: int | None):
        self._pipeline_ref = pipeline
        self._process_group = process_group
        self.prev_rank = prev_rank
        self.next_rank = next_rank
        self.sent_activation_meta = False
        self.activation_recv_buf: tuple[torch.Tensor, ...] | None = None
        self.grad_recv_buf: tuple[torch.Tensor, ...] | None = None

    @property
    def pipeline(self) -> "OobleckPipeline":
        return self._pipeline_ref

    def _gloo_cuda(self, tensor: torch.Tensor) -> bool:
        # gloo has no CUDA send/recv: stage through host memory.  Used by
        # the single-GPU pp>1 composition tests (two ranks sharing one
        # device); the production path is RCCL (backend "nccl").
        return tensor.is_cuda and \
            dist.get_backend(self._process_group) == "gloo"

    def _send(self, tensor: torch.Tensor, dest: int) -> None:
        if self._gloo_cuda(tensor):
            tensor = tensor.cpu()
        dist.send(tensor, dest, group=self._process_group)

    def _recv(self, tensor: torch.Tensor, src: int) -> None:
        if self._gloo_cuda(tensor):
            host = torch.empty(tensor.shape, dtype=tensor.dtype, device="cpu")
            dist.recv(host, src, group=self._process_group)
            with torch.no_grad():
                tensor.copy_(host)
        else:
            dist.recv(tensor, src, group=self._process_group)

    def _pack_meta(self, buffers: tuple[torch.Tensor, ...]) -> torch.Tensor:
        assert len(buffers) <= _META_MAX_TENSORS
        meta = torch.zeros(_META_WORDS, dtype=torch.int64)
        meta[0] = len(buffers)
        for i, t in enumerate(buffers):
            o = 1 + i * 7
            meta[o] = DTYPE_TO_ID[t.dtype]
            meta[o + 1] = 1 if t.requires_grad or t.is_floating_point() else 0
            meta[o + 2] = t.dim()
            for d in range(t.dim()):
                meta[o + 3 + d] = t.shape[d]
        return meta.to(self.pipeline.device)

    def send_activations(self, buffer_id: int) -> None:
        outputs = self.pipeline.pipe_buffers["outputs"][buffer_id]
        if not self.sent_activation_meta:
            self._send(self._pack_meta(outputs), self.next_rank)
            self.sent_activation_meta = True
        for t in outputs:
            self._send(t.contiguous(), self.next_rank)

    def recv_activations(self, buffer_id: int) -> None:
        if self.activation_recv_buf is None:
            meta = torch.zeros(_META_WORDS, dtype=torch.int64,
                               device=self.pipeline.device)
            self._recv(meta, self.prev_rank)
            meta = meta.cpu()
            bufs = []
            for i in range(int(meta[0])):
                o = 1 + i * 7
                dtype = ID_TO_DTYPE[int(meta[o])]
                ndims = int(meta[o + 2])
                shape = [int(meta[o + 3 + d]) for d in range(ndims)]
                t = torch.zeros(shape, dtype=dtype, device=self.pipeline.device)
                t.requires_grad = bool(meta[o + 1]) and t.is_floating_point()
                bufs.append(t)
            self.activation_recv_buf = tuple(bufs)
        recvd = []
        for buf in self.activation_recv_buf:
            self._recv(buf, self.prev_rank)
            c = buf.clone().detach()
            c.requires_grad = buf.requires_grad
            recvd.append(c)
        self.pipeline.pipe_buffers["inputs"][buffer_id] = tuple(recvd)

    def send_gradients(self, buffer_id: int) -> None:
        # grad of the stage's input activation, produced by backward_pass
        din = self.pipeline.execution._din_bufs[buffer_id]
        self._send(din.contiguous(), self.prev_rank)
        self.pipeline.pipe_buffers["inputs"][buffer_id] = None

    def recv_gradients(self, buffer_id: int) -> None:
        outputs = self.pipeline.pipe_buffers["outputs"][buffer_id]
        if self.grad_recv_buf is None:
            self.grad_recv_buf = tuple(
                torch.zeros_like(t) for t in outputs
                if t.is_floating_point())
        for buf in self.grad_recv_buf:
            self._recv(buf, self.next_rank)


class OobleckPipeline:
    """Public surface kept from the reference: train(),
    initialize_distributed_fsdp/pipeline/execution, rank_grid, my_pipeline,
    is_first_stage/is_last_stage."""

    def __init__(self, pipeline_id: int, rank_grid: dict[int, list[int]],
                 model_cfg: ModelConfig, training_cfg: TrainingConfig,
                 dataloader, num_microbatches: int, device,
                 stage_of_layer: dict[int, int] | None = None):
        self._pipeline_id = pipeline_id
        self.rank_grid = rank_grid
        self.model_cfg = model_cfg
        self.training_cfg = training_cfg
        self._dataloader = dataloader
        self.num_microbatches = num_microbatches
        self.device = device
        assert dist.is_initialized()
        ranks = sorted({r for rs in rank_grid.values() for r in rs})
        self._ranks = ranks
        self.my_pipeline = dist.get_rank() in ranks
        self.execution: PipelineExecution | None = None
        self.communication: PipelineCommunication | None = None

    # -- process-group setup (all ranks co-call new_group) -------------------
    def initialize_distributed_fsdp(self) -> None:
        # per-layer groups (reference pipeline.py:565-580)
        self._per_layer_pgs = {}
        for layer_id, ranks in self.rank_grid.items():
            self._per_layer_pgs[layer_id] = dist.new_group(sorted(set(ranks)))

    def initialize_distributed_pipeline(self) -> None:
        # per-shard pipeline columns (reference pipeline.py:582-617)
        self._per_sharded_pp_pgs = {}
        my_rank = dist.get_rank()
        for shard_id in range(len(self.rank_grid[min(self.rank_grid)])):
            col = [ranks[shard_id] for ranks in self.rank_grid.values()]
            unique = list(dict.fromkeys(col))  # de-dup, order preserved
            pg = dist.new_group(sorted(set(col)))
            self._per_sharded_pp_pgs[shard_id] = pg
            if my_rank in unique:
                idx = unique.index(my_rank)
                self.communication = PipelineCommunication(
                    pipeline=self, process_group=pg,
                    prev_rank=unique[idx - 1] if idx > 0 else None,
                    next_rank=unique[idx + 1] if idx < len(unique) - 1 else None)

    def initialize_execution(self, layer_factory, optimizer_factory) -> None:
        """Build this rank's layers and the schedule.
        layer_factory(layer_id, process_group, n_slots) -> Layer-like;
        optimizer_factory(layers) -> (optimizer, lr_scheduler)."""
        assert self.my_pipeline
        my_rank = dist.get_rank()
        my_layers = [lid for lid, ranks in self.rank_grid.items()
                     if my_rank in ranks]
        my_layers.sort()

        # stage index: contiguous runs of identical rank lists form stages
        stages = []
        for lid in sorted(self.rank_grid):
            rs = tuple(self.rank_grid[lid])
            if not stages or stages[-1][0] != rs:
                stages.append((rs, [lid]))
            else:
                stages[-1][1].append(lid)
        self.num_stages = len(stages)
        self.my_stage_index = next(i for i, (rs, lids) in enumerate(stages)
                                   if my_layers[0] in lids)

        import os as _os
        min_buf = 2
        if (self.num_stages == 1 and self.device.type == "cuda"
                and _os.environ.get("OB_PP1_OVERLAP", "1") == "1"):
            # deeper slot rotation would let the overlap's forward stream
            # run further ahead of the backward stream (slot reuse is the
            # F(m+k) -> B(m) fence); measured flat at this workload
            # (230.8 / 232.8 / 232.1 ms at 2/3/4 slots), so default 2
            min_buf = int(_os.environ.get("OB_PP1_SLOTS", "2"))
        self.train_schedule = OobleckPipelineSchedule(
            micro_batches=self.num_microbatches, stages=self.num_stages,
            stage_id=self.my_stage_index, min_pipe_buffers=min_buf)
        n_slots = self.train_schedule.num_pipe_buffers()
        self.pipe_buffers = {
            "inputs": [None] * n_slots,
            "labels": [None] * n_slots,
            "outputs": [None] * n_slots,
        }
        layers = [layer_factory(lid, self._per_layer_pgs[lid], n_slots)
                  for lid in my_layers]
        optimizer, lr_sched = optimizer_factory(layers)
        self.execution = PipelineExecution(
            pipeline=self, layers=layers, dataloader=self._dataloader,
            training_cfg=self.training_cfg, optimizer=optimizer,
            lr_scheduler=lr_sched)

    # -- the training dispatch loop (reference pipeline.py:458-487) ----------
    def train(self) -> None:
        instruction_map = {
            LoadMicroBatch: self.execution.load_microbatch,
            ForwardPass: self.execution.forward_pass,
            BackwardPass: self.execution.backward_pass,
            SendActivation: self.communication.send_activations,
            RecvActivation: self.communication.recv_activations,
            SendGrad: self.communication.send_gradients,
            RecvGrad: self.communication.recv_gradients,
        }
        import os
        import sys
        trace = os.environ.get("OB_TRACE_SCHED", "0") == "1"
        if getattr(self.execution, "_overlap", False):
            # the optimizer / DP all-reduce / zero_grads of the previous
            # step ran on the default stream AFTER the epoch join; the
            # overlap streams must not start this step's compute before
            # those writes land (param update, grad zero, shadow refresh)
            cur = torch.cuda.current_stream()
            self.execution._s_fwd.wait_stream(cur)
            self.execution._s_bwd.wait_stream(cur)
        for step_cmds in self.train_schedule:
            for cmd in step_cmds:
                handler = instruction_map.get(type(cmd))
                if handler is None:
                    raise RuntimeError(f"unknown instruction {cmd!r}")
                if trace:
                    import time as _t
                    print(f"[sched {_t.monotonic():.3f}] "
                          f"{type(cmd).__name__} buf={cmd.buffer_id}",
                          file=sys.stderr, flush=True)
                handler(**cmd.kwargs)
                if trace:
                    print(f"[sched {_t.monotonic():.3f}] done "
                          f"{type(cmd).__name__}", file=sys.stderr,
                          flush=True)
        if getattr(self.execution, "_overlap", False):
            cur = torch.cuda.current_stream()
            cur.wait_stream(self.execution._s_fwd)
            cur.wait_stream(self.execution._s_bwd)
        for name, bufs in self.pipe_buffers.items():
            self.pipe_buffers[name] = [None] * len(bufs)

    def reset_iterator(self) -> None:
        self.execution._data_iterator = iter(self.execution._dataloader)

    def is_first_stage(self) -> bool:
        return self.communication.prev_rank is None

    def is_last_stage(self) -> bool:
        return self.communication.next_rank is None
