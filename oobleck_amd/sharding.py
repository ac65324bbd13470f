# FULL_SHARD intra-stage parameter sharding (§8 f1) — the MI355X-native
# equivalent of the reference's FlatParamHandle FULL_SHARD path
# (/root/reference/oobleck/execution/layer.py:96-142, 167-225):
#
#   * the authoritative state is each rank's contiguous SHARD of the flat
#     parameter (world-size-padded, rank-ordered chunks — FSDP layout);
#   * forward/backward UNSHARD by all-gathering the shards into the full
#     flat buffer the compute kernel is bound to (layer.py:117-131);
#   * after each microbatch's backward, the full grad is reduce-scattered
#     and ACCUMULATED into _saved_grad_shard, and the full grad cleared
#     (layer.py:167-225 post_backward_hook semantics);
#   * the optimizer and DP all-reduce then operate on the shard
#     (reduce_gradients' prepare_gradient_for_optim, layer.py:272-291).
#
# Used by both the HIP Layer (compute = C-ABI kernels) and the test-only
# OracleLayer (compute = oracle), so the collective choreography is covered
# by CPU gloo tests.  Round-1 note: collectives run on the compute stream
# (the reference overlaps on pre/post side streams — a later optimization).
from __future__ import annotations

import torch
import torch.distributed as dist


class ShardedFlatParam:
    def __init__(self, n_params: int, process_group, device,
                 init_full: torch.Tensor | None = None):
        self.pg = process_group
        self.ws = dist.get_world_size(process_group)
        self.rank = dist.get_rank(process_group)
        assert self.rank >= 0
        self.n_params = n_params
        self.shard_size = (n_params + self.ws - 1) // self.ws
        self.padded = self.shard_size * self.ws
        # full buffer: what the compute kernel is bound to (first n_params
        # entries are the parameters; tail is padding)
        self.full = torch.zeros(self.padded, dtype=torch.float32, device=device)
        self.full_grad = torch.zeros(self.padded, dtype=torch.float32,
                                     device=device)
        if init_full is not None:
            self.full[:n_params].copy_(init_full)
        # authoritative shard + accumulated sharded gradient
        self.shard = self.full[self.rank * self.shard_size:
                               (self.rank + 1) * self.shard_size].clone()
        self.shard.grad = torch.zeros_like(self.shard)
        self._scatter_tmp = torch.empty_like(self.shard)

    @property
    def saved_grad_shard(self) -> torch.Tensor:
        return self.shard.grad

    def unshard(self) -> None:
        # all_gather_into_tensor: rank-ordered concat == the full flat param
        dist.all_gather_into_tensor(self.full, self.shard, group=self.pg)

    def reduce_scatter_grad(self) -> None:
        """Per-microbatch: reduce-scatter the full grad, accumulate into the
        sharded grad, clear the full grad for the next microbatch."""
        dist.reduce_scatter_tensor(self._scatter_tmp, self.full_grad,
                                   group=self.pg)
        self.shard.grad += self._scatter_tmp
        self.full_grad.zero_()
