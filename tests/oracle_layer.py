# Test-only CPU Layer implementing the same slot interface as
# oobleck_amd.layer.Layer, backed by the oracle restatement (allowed here:
# tests may use oracle/ as the checker).  Used to exercise the schedule /
# p2p / DP-allreduce host logic on CPU (gloo) where the HIP extension
# cannot run.  NEVER imported by product code.
from __future__ import annotations

import torch
import torch.distributed

from oracle.gpt2_oracle import (KIND_FINAL, OracleConfig, layer_forward)


class OracleLayer:
    def __init__(self, layer_id: int, cfg: OracleConfig, flat: torch.Tensor,
                 n_slots: int = 4, process_group=None):
        self.layer_id = layer_id
        self.cfg = cfg
        self.kind = cfg.layer_kind(layer_id)
        self._n = flat.numel()
        self._sharded = None
        group_size = 1
        if process_group is not None and torch.distributed.is_initialized():
            group_size = torch.distributed.get_world_size(process_group)
        if group_size > 1:
            from oobleck_amd.sharding import ShardedFlatParam
            self._sharded = ShardedFlatParam(self._n, process_group,
                                             torch.device("cpu"),
                                             init_full=flat)
            self.flat_param = self._sharded.shard
        else:
            self.flat_param = flat.clone()
            self.flat_param.grad = torch.zeros_like(flat)
        self._batch = 0
        self._saved: dict[int, tuple] = {}

    @property
    def flat_grad(self) -> torch.Tensor:
        return self.flat_param.grad

    def set_batch(self, batch: int) -> None:
        self._batch = batch

    def zero_grads(self) -> None:
        self.flat_grad.zero_()
        if self._sharded is not None:
            self._sharded.full_grad.zero_()

    def _full_params(self) -> torch.Tensor:
        if self._sharded is not None:
            self._sharded.unshard()
            return self._sharded.full[:self._n]
        return self.flat_param

    def forward_slot(self, slot: int, x: torch.Tensor, out: torch.Tensor,
                     labels: torch.Tensor | None = None) -> None:
        flat = self._full_params().detach().clone().requires_grad_(True)
        xi = x.detach().clone()
        if xi.is_floating_point():
            xi.requires_grad_(True)
        y = layer_forward(self.cfg, self.kind, flat, xi,
                          labels if self.kind == KIND_FINAL else None)
        self._saved[slot] = (flat, xi, y)
        out.copy_(y.detach().reshape(out.shape))

    def backward_slot(self, slot: int, dout: torch.Tensor | None,
                      din: torch.Tensor | None) -> None:
        flat, xi, y = self._saved.pop(slot)
        if dout is None:
            y.backward()
        else:
            torch.autograd.backward(y, dout.reshape(y.shape))
        if self._sharded is not None:
            self._sharded.full_grad[:self._n] += flat.grad
            self._sharded.reduce_scatter_grad()
        else:
            self.flat_param.grad += flat.grad
        if din is not None and xi.is_floating_point():
            din.copy_(xi.grad)

    # same chunking semantics as the product Layer / reference layer.py:272-291
    def _shard_param(self, tensor, number):
        chunks = list(torch.flatten(tensor).chunk(number))
        if len(chunks) < number:
            chunks += [torch.zeros_like(chunks[0])] * (number - len(chunks))
        pad = chunks[0].numel() - chunks[-1].numel()
        if pad > 0:
            chunks[-1] = torch.nn.functional.pad(chunks[-1], [0, pad])
        return chunks

    def reduce_gradients(self, process_groups) -> None:
        assert all(torch.distributed.get_rank(pg) >= 0
                   for pg in process_groups.values())
        grads = (self._shard_param(self.flat_grad, len(process_groups))
                 if len(process_groups) > 1 else [self.flat_grad])
        for grad, (_i, pg) in zip(grads, process_groups.items()):
            torch.distributed.all_reduce(tensor=grad, group=pg)

    def remove_tensors(self) -> None:
        self.flat_param.grad = None
        self.flat_param.data = torch.tensor([])


class NoOpOptimizer:
    def __init__(self, layers):
        self.layers = layers

    def step(self) -> None:
        pass

    def zero_grad(self) -> None:
        for l in self.layers:
            l.zero_grads()
