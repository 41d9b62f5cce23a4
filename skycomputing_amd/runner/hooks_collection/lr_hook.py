"""Learning-rate schedule hook (warmup + decay).

Sets ``runner.optimizer.lr`` before every iteration; FusedSGD reads the
attribute at each eager step. NOTE: the captured-step executors
(GraphedTrainStep / GraphedPipelineStep) freeze the lr INSIDE the graph at
capture time — schedules only act on the eager Runner path, which is where
hooks run. The reference has no scheduler at all (constant lr=1e-3).
"""

from __future__ import annotations

import math

from ...registry import HOOKS
from ..hooks import Hook


@HOOKS.register_module
class LRScheduleHook(Hook):
    def __init__(self, base_lr: float, warmup_iters: int = 0,
                 total_iters: int | None = None, decay: str = "constant",
                 min_lr: float = 0.0):
        assert decay in ("constant", "linear", "cosine")
        self.base_lr = base_lr
        self.warmup_iters = max(0, warmup_iters)
        self.total_iters = total_iters
        self.decay = decay
        self.min_lr = min_lr

    def lr_at(self, it: int, total: int | None) -> float:
        if self.warmup_iters and it < self.warmup_iters:
            return self.base_lr * (it + 1) / self.warmup_iters
        if self.decay == "constant" or not total or total <= self.warmup_iters:
            return self.base_lr
        t = min(1.0, (it - self.warmup_iters) / max(1, total - self.warmup_iters))
        if self.decay == "linear":
            f = 1.0 - t
        else:  # cosine
            f = 0.5 * (1.0 + math.cos(math.pi * t))
        return self.min_lr + (self.base_lr - self.min_lr) * f

    def before_train_iter(self, runner):
        total = self.total_iters or runner.max_iter
        runner.optimizer.lr = self.lr_at(runner.iter, total)
