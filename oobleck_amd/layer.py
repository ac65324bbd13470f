# Layer — host wrapper over one C-ABI layer object (the drop-in equivalent
# of the reference's Layer, /root/reference/oobleck/execution/layer.py).
#
# Surface kept duck-type compatible with the reference's consumers
# (SURVEY.md §8b): layer_id, _param_handle.flat_param (contiguous fp32),
# reduce_gradients(process_groups), remove_tensors().  Compute goes through
# the HIP extension only — no torch fallback.
from __future__ import annotations

import ctypes

import torch
import torch.distributed

from ._ext import ObLayerDesc, check, get_ext
from .config import ModelConfig
from .params import init_layer_params


class _FlatParamHandle:
    """Minimal stand-in for torch FSDP's FlatParamHandle: exposes
    .flat_param with .grad, which is all the reference's optimizer /
    reconfiguration consumers touch (pipeline.py:117-119, engine.py:283-299)."""

    def __init__(self, flat_param: torch.Tensor, flat_grad: torch.Tensor):
        self.flat_param = flat_param
        self.flat_param.grad = flat_grad


def _stream_ptr() -> ctypes.c_void_p:
    return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)


def _ptr(t: torch.Tensor | None) -> ctypes.c_void_p:
    if t is None:
        return ctypes.c_void_p(0)
    assert t.is_contiguous()
    return ctypes.c_void_p(t.data_ptr())


class Layer:
    """One fx-shard (embedding / block / ln_f+lm_head+loss).  NO_SHARD when
    the stage's per-layer process group has 1 rank; FULL_SHARD (per-rank
    parameter shards, all-gather on unshard, per-microbatch reduce-scatter
    of grads — reference layer.py:96-111 chooses by group size) otherwise."""

    def __init__(self, layer_id: int, cfg: ModelConfig, max_batch: int,
                 seq_len: int, n_slots: int, device: torch.device,
                 process_group: torch.distributed.ProcessGroup | None = None,
                 init_style: str = "gpt2", seed: int = 42,
                 dtype: str = "f32"):
        self.layer_id = layer_id
        self.cfg = cfg
        self.kind = cfg.layer_kind(layer_id)
        self.device = device
        assert dtype in ("f32", "bf16")
        self.dtype = dtype
        self.act_dtype = torch.float32 if dtype == "f32" else torch.bfloat16
        ext = get_ext()
        self._desc = ObLayerDesc(
            kind=self.kind, n_embd=cfg.n_embd, n_head=cfg.n_head,
            n_positions=cfg.n_positions, vocab_size=cfg.vocab_size,
            max_batch=max_batch, seq_len=seq_len, n_slots=n_slots,
            dtype=0 if dtype == "f32" else 1)
        n = ext.ob_layer_param_count(ctypes.byref(self._desc))
        assert n > 0
        init = init_layer_params(cfg, layer_id, seed, init_style).to(device)
        group_size = 1
        if process_group is not None and torch.distributed.is_initialized():
            group_size = torch.distributed.get_world_size(process_group)
        handle = ctypes.c_void_p()
        check(ext.ob_layer_create(ctypes.byref(self._desc),
                                  ctypes.byref(handle)), "layer_create")
        self._h = handle
        self._sharded = None
        if group_size > 1:
            from .sharding import ShardedFlatParam
            self._sharded = ShardedFlatParam(n, process_group, device,
                                             init_full=init)
            del init
            check(ext.ob_layer_bind(self._h, _ptr(self._sharded.full),
                                    _ptr(self._sharded.full_grad)),
                  "layer_bind")
            self._param_handle = _FlatParamHandle(self._sharded.shard,
                                                  self._sharded.shard.grad)
        else:
            flat = init
            grad = torch.zeros(n, dtype=torch.float32, device=device)
            check(ext.ob_layer_bind(self._h, _ptr(flat), _ptr(grad)),
                  "layer_bind")
            self._param_handle = _FlatParamHandle(flat, grad)
        self._batch = max_batch
        self._shadows_stale = False
        self.refresh_weights()

    # -- compute ------------------------------------------------------------
    @property
    def flat_param(self) -> torch.Tensor:
        return self._param_handle.flat_param

    @property
    def flat_grad(self) -> torch.Tensor:
        return self._param_handle.flat_param.grad

    def set_batch(self, batch: int) -> None:
        if batch != self._batch:
            check(get_ext().ob_layer_set_batch(self._h, batch), "set_batch")
            self._batch = batch

    def _unshard(self) -> None:
        """All-gather the shards into the bound full buffer, then (bf16)
        re-cast the extension's bf16 shadows if the optimizer touched the
        sharded masters since the last unshard — the shadows are cast from
        the FULL buffer, which only becomes current here."""
        self._sharded.unshard()
        if self._shadows_stale:
            check(get_ext().ob_layer_refresh_weights(self._h, _stream_ptr()),
                  f"refresh_weights layer {self.layer_id}")
            self._shadows_stale = False

    def forward_slot(self, slot: int, x: torch.Tensor, out: torch.Tensor,
                     labels: torch.Tensor | None = None) -> None:
        if self._sharded is not None:
            self._unshard()  # pre_forward_hook (layer.py:147-153)
        check(get_ext().ob_layer_forward(self._h, slot, _ptr(x), _ptr(out),
                                         _ptr(labels), _stream_ptr()),
              f"forward layer {self.layer_id}")

    def backward_slot(self, slot: int, dout: torch.Tensor | None,
                      din: torch.Tensor | None) -> None:
        if self._sharded is not None:
            self._unshard()  # pre_backward_hook (layer.py:160-166)
            # the post stream's previous reduce-scatter zeroes full_grad;
            # this backward accumulates into it — order after it
            self._sharded.wait_post()
        check(get_ext().ob_layer_backward(self._h, slot, _ptr(dout), _ptr(din),
                                          _stream_ptr()),
              f"backward layer {self.layer_id}")
        if self._sharded is not None:
            # post_backward_hook: reduce-scatter + accumulate on the post
            # stream (layer.py:167-225, post_backward_stream :183)
            self._sharded.reduce_scatter_grad()

    def zero_grads(self) -> None:
        if self._sharded is not None:
            self._sharded.wait_post()
        self.flat_grad.zero_()
        if self._sharded is not None:
            self._sharded.full_grad.zero_()

    def refresh_weights(self) -> None:
        """bf16 mode: re-cast the extension's bf16 weight shadows from the
        fp32 master params (after init and after every optimizer step).
        FULL_SHARD: the optimizer updates the SHARDED masters, but the
        shadows are cast from the bound full buffer, which is only
        re-gathered at the next unshard — so defer the cast to _unshard()
        instead of silently training on one-step-stale weights."""
        if self._sharded is not None:
            # shard changed: the resident full buffer must be re-gathered
            # at the next unshard (any dtype)
            self._sharded.mark_dirty()
            if self.dtype == "bf16":
                self._shadows_stale = True
            return
        if self.dtype == "bf16":
            check(get_ext().ob_layer_refresh_weights(self._h, _stream_ptr()),
                  f"refresh_weights layer {self.layer_id}")

    # -- distributed surface (reference layer.py:272-291) --------------------
    def _shard_param(self, tensor: torch.Tensor, number: int) -> list[torch.Tensor]:
        chunks = list(torch.flatten(tensor).chunk(number))
        if len(chunks) < number:
            chunks += [torch.zeros_like(chunks[0])] * (number - len(chunks))
        pad = chunks[0].numel() - chunks[-1].numel()
        if pad > 0:
            chunks[-1] = torch.nn.functional.pad(chunks[-1], [0, pad])
        return chunks

    def reduce_gradients(self, process_groups: dict[int, torch.distributed.ProcessGroup]) -> None:
        """Per-layer data-parallel all-reduce of the flat grad, chunked
        across fsdp sub-groups when the layer spans several shard indices
        (reference layer.py:283-291; SUM, not averaged — matching the
        reference, which never divides by the DP degree)."""
        assert all(torch.distributed.get_rank(pg) >= 0
                   for pg in process_groups.values())
        if self._sharded is not None:
            self._sharded.wait_post()  # sharded grad fully accumulated
        if len(process_groups) > 1:
            grads = self._shard_param(self.flat_grad, len(process_groups))
        else:
            grads = [self.flat_grad]
        for grad, (_idx, pg) in zip(grads, process_groups.items()):
            torch.distributed.all_reduce(tensor=grad, group=pg)
        if len(process_groups) > 1:
            # Deviation from the reference (layer.py:283-291): when the flat
            # grad is not divisible by the group count, F.pad COPIES the tail
            # chunk, so the reference's reduced tail values never land back
            # in the grad (silently un-reduced).  Write the reduced tail
            # back; every other chunk is a view and needs nothing.
            flat = self.flat_grad
            chunk = grads[0].numel()
            tail_off = ((flat.numel() - 1) // chunk) * chunk
            tail_len = flat.numel() - tail_off
            if tail_len < chunk:
                flat[tail_off:].copy_(grads[tail_off // chunk][:tail_len])

    def remove_tensors(self) -> None:
        # reference layer.py:66-69 (reconfiguration discards a layer's state).
        # Also unbind the extension from the raw device pointers so a stale
        # forward/backward/adamw on the discarded layer fails loudly
        # ("params not bound") instead of reading freed storage.
        check(get_ext().ob_layer_bind(self._h, ctypes.c_void_p(0),
                                      ctypes.c_void_p(0)), "unbind")
        self._param_handle.flat_param.grad = None
        self._param_handle.flat_param.data = torch.tensor([], device=self.device)

    def __del__(self):
        try:
            if getattr(self, "_h", None):
                get_ext().ob_layer_destroy(self._h)
        except Exception:  # noqa: BLE001  (interpreter teardown)
            pass
