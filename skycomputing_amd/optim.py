"""Optimizers.

Replaces the reference's DistributedOptimizer-over-RPC
(reference: experiment/launch.py:152-156, runner.py:139): in the SPMD world
each rank steps its OWN stage's parameters locally after its backward —
there is no cross-rank optimizer traffic at all.

FusedSGD keeps optional fp32 master weights for bf16 parameters and uses
the multi-tensor HIP SGD kernel on GPU (single launch over all chunks);
on CPU it applies the same math eagerly.
"""

from __future__ import annotations

import torch

from . import ops
from .ops import hiplib


class FusedSGD:
    def __init__(
        self,
        params,
        lr: float = 1e-3,
        momentum: float = 0.0,
        weight_decay: float = 0.0,
        master_weights: bool | None = None,
    ):
        self.params = [p for p in params if p.requires_grad]
        self.lr = lr
        self.momentum = momentum
        self.weight_decay = weight_decay
        if master_weights is None:
            master_weights = any(p.dtype == torch.bfloat16 for p in self.params)
        self.masters = (
            [p.detach().clone().float() for p in self.params] if master_weights else None
        )
        # momentum buffers are ALWAYS fp32: the HIP multi-tensor path reads
        # them as float* (ops/hip/sgd.hip) regardless of the param dtype,
        # and fp32 accumulation is the numerically right choice for bf16
        # params even without master weights.
        self.momentum_bufs = (
            [torch.zeros(p.shape, dtype=torch.float32, device=p.device)
             for p in self.params]
            if momentum != 0.0
            else None
        )
        self._gpu_plan = None  # lazy multi-tensor launch plan

    @torch.no_grad()
    def sync_masters(self):
        """Refresh fp32 masters (and reset momentum) from the CURRENT param
        values. Must be called after any out-of-band weight mutation
        (checkpoint restore, ParameterServer scatter) or the next step()
        would write ``stale_master - lr*grad`` back over the restored
        weights."""
        if self.masters is not None:
            for m, p in zip(self.masters, self.params):
                m.copy_(p.detach().float())
        if self.momentum_bufs is not None:
            for b in self.momentum_bufs:
                b.zero_()

    def zero_grad(self, set_to_none: bool = True):
        for p in self.params:
            if p.grad is not None:
                if set_to_none:
                    p.grad = None
                else:
                    p.grad.detach_().zero_()

    @torch.no_grad()
    def step(self):
        grads = [p.grad for p in self.params]
        use_hip = (
            hiplib.available()
            and self.params
            and self.params[0].is_cuda
            and all(g is not None for g in grads)
        )
        if use_hip:
            self._hip_step(grads)
        else:
            ops.sgd_step(
                self.params, grads, self.lr, self.momentum, self.weight_decay,
                self.momentum_bufs, self.masters,
            )

    def _hip_step(self, grads):
        from .ops.multi_tensor import multi_tensor_sgd

        multi_tensor_sgd(
            self.params, grads, self.masters, self.momentum_bufs,
            self.lr, self.momentum, self.weight_decay,
        )

    def state_dict(self) -> dict:
        return {
            "lr": self.lr, "momentum": self.momentum, "weight_decay": self.weight_decay,
            "masters": self.masters, "momentum_bufs": self.momentum_bufs,
        }

    def load_state_dict(self, sd: dict):
        self.lr = sd["lr"]
        self.momentum = sd["momentum"]
        self.weight_decay = sd["weight_decay"]
        if sd.get("masters") is not None and self.masters is not None:
            for m, s in zip(self.masters, sd["masters"]):
                m.copy_(s.to(m.device))
        if sd.get("momentum_bufs") is not None and self.momentum_bufs is not None:
            for m, s in zip(self.momentum_bufs, sd["momentum_bufs"]):
                m.copy_(s.to(m.device))
