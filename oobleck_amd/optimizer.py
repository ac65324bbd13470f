# Fused AdamW over per-layer flat parameter buffers + WarmupLR.
#
# Replaces the reference's torch.optim.AdamW(fused=True) over
# _param_handle.flat_param (/root/reference/oobleck/execution/pipeline.py:
# 117-127) and deepspeed's WarmupLR (pipeline.py:125-127).  One HIP kernel
# launch per layer; 7 words/param of HBM traffic (read p,g,m,v; write p,m,v).
from __future__ import annotations

import torch

from ._ext import check, get_ext


class WarmupLR:
    """deepspeed.runtime.lr_schedules.WarmupLR semantics: lr rises
    log-or-linearly from warmup_min_lr to the optimizer lr over
    warmup_num_steps, then stays flat.  We implement the default linear
    ramp from 0 (deepspeed default warmup_min_lr=0)."""

    def __init__(self, optimizer: "FusedAdamW", warmup_num_steps: int):
        self.optimizer = optimizer
        self.warmup_num_steps = max(0, warmup_num_steps)
        self.base_lr = optimizer.lr
        self._step = 0
        self.step()  # deepspeed schedulers initialize lr at construction

    def step(self) -> None:
        self._step += 1
        if self.warmup_num_steps > 0 and self._step <= self.warmup_num_steps:
            self.optimizer.lr = self.base_lr * self._step / self.warmup_num_steps
        else:
            self.optimizer.lr = self.base_lr


class FusedAdamW:
    def __init__(self, layers, lr: float, betas=(0.9, 0.999), eps: float = 1e-8,
                 weight_decay: float = 0.0):
        self.layers = list(layers)
        self.lr = lr
        self.betas = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.step_count = 0
        self._m = [torch.zeros_like(l.flat_param) for l in self.layers]
        self._v = [torch.zeros_like(l.flat_param) for l in self.layers]

    def step(self) -> None:
        import ctypes
        self.step_count += 1
        ext = get_ext()
        stream = ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)
        for l, m, v in zip(self.layers, self._m, self._v):
            p, g = l.flat_param, l.flat_grad
            check(ext.ob_adamw_step(
                ctypes.c_void_p(p.data_ptr()), ctypes.c_void_p(g.data_ptr()),
                ctypes.c_void_p(m.data_ptr()), ctypes.c_void_p(v.data_ptr()),
                p.numel(), self.step_count, self.lr, self.betas[0],
                self.betas[1], self.eps, self.weight_decay, stream),
                f"adamw layer {l.layer_id}")
            refresh = getattr(l, "refresh_weights", None)
            if refresh is not None:
                refresh()  # bf16 shadows follow the fp32 master update

    def zero_grad(self) -> None:
        for l in self.layers:
            l.zero_grads()
