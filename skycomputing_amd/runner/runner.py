"""Training loop.

Capability parity with the reference Runner
(reference: scaelum/runner/runner.py:15-156): epoch/iter loop, hook
dispatch, per-phase timing, max_iters guard — redesigned SPMD: every rank
runs the loop; the PipelineEngine moves activations/grads over RCCL and the
local FusedSGD steps this rank's own parameters (no dist_autograd /
DistributedOptimizer). The reference's broken
max_epochs/_max_epoch property pair (runner.py:39,83-85) is not reproduced.
"""

from __future__ import annotations

import time

from ..logger import Logger
from ..timer import DistributedTimer
from .hooks import Hook


class Runner:
    def __init__(
        self,
        engine,
        optimizer,
        comm,
        max_epoch: int = 1,
        max_iter: int | None = None,
        num_microbatches: int = 1,
        schedule: str = "gpipe",
        logger: Logger | None = None,
        log_interval: int = 1,
        grad_clip_norm: float | None = None,
    ):
        self.engine = engine
        self.optimizer = optimizer
        self.comm = comm
        self._max_epoch = max_epoch
        self._max_iter = max_iter
        self.num_microbatches = num_microbatches
        self.schedule = schedule
        self.logger = logger or Logger(rank=comm.rank)
        self.log_interval = log_interval
        self.grad_clip_norm = grad_clip_norm
        self.timer = DistributedTimer()
        self.hooks: list[Hook] = []
        self.epoch = 0
        self.iter = 0
        self.last_loss: float | None = None
        self.last_val_acc: float | None = None
        self.should_stop = False
        self.iter_times: list[float] = []

    # reference-compatible property surface (names fixed, see module docstring)
    @property
    def max_epoch(self) -> int:
        return self._max_epoch

    max_epochs = max_epoch  # alias; both names valid

    @property
    def max_iter(self):
        return self._max_iter

    max_iters = max_iter

    def register_hook(self, hook: Hook):
        assert isinstance(hook, Hook)
        self.hooks.append(hook)

    def call_hook(self, fn_name: str):
        for h in self.hooks:
            getattr(h, fn_name)(self)

    def train_step(self, data, labels) -> float | None:
        """One full forward+backward+step across the pipeline."""
        t0 = time.perf_counter()
        self.optimizer.zero_grad(set_to_none=True)
        loss = self.engine.run_iteration(
            inputs=data,
            labels=labels,
            num_microbatches=self.num_microbatches,
            schedule=self.schedule,
        )
        if self.grad_clip_norm is not None:
            # per-rank clip over this stage's grads (each rank owns its
            # slice; the reference has no clipping at all)
            import torch

            torch.nn.utils.clip_grad_norm_(
                self.engine.parameters(), self.grad_clip_norm
            )
        self.optimizer.step()
        self.iter_times.append(time.perf_counter() - t0)
        self.last_loss = loss
        return loss

    def train(self, data_loader):
        self.engine.train(True)
        self.call_hook("before_run")
        done = False
        # start from self.epoch so a resumed checkpoint (CheckpointHook
        # resume_counters) continues its schedule
        for epoch in range(self.epoch, self._max_epoch):
            self.epoch = epoch
            self.call_hook("before_train_epoch")
            for data, labels in data_loader:
                if self._max_iter is not None and self.iter >= self._max_iter:
                    done = True
                    break
                self.call_hook("before_train_iter")
                loss = self.train_step(data, labels)
                self.call_hook("after_train_iter")
                if (self.iter + 1) % self.log_interval == 0:
                    self.logger.info(
                        f"epoch {epoch} iter {self.iter} "
                        f"loss {loss if loss is not None else float('nan'):.4f} "
                        f"time {self.iter_times[-1]*1e3:.1f} ms"
                    )
                self.iter += 1
                if self.should_stop:
                    done = True
                    break
            self.call_hook("after_train_epoch")
            if done or self.should_stop:
                break
        self.call_hook("after_run")

    def val(self, data_loader, max_batches: int | None = None) -> float | None:
        """Forward-only evaluation; returns classification accuracy
        (broadcast to all ranks). The reference exposed val hooks but no
        loop (scaelum/runner/hooks.py:5-58); this completes the lifecycle."""
        self.engine.eval()
        self.call_hook("before_val_epoch")
        correct = total = 0
        for bi, (data, labels) in enumerate(data_loader):
            if max_batches is not None and bi >= max_batches:
                break
            self.call_hook("before_val_iter")
            logits = self.engine.evaluate_batch(data, labels)
            if logits is not None:
                pred = logits.float().argmax(-1).cpu()
                correct += int((pred == labels).sum())
                total += labels.numel()
            self.call_hook("after_val_iter")
        acc = (correct / total) if total else None
        src = self.engine.plan.stage_ranks[-1] if self.engine.plan.num_stages else 0
        acc = self.comm.broadcast_object(acc, src=src)
        self.last_val_acc = acc
        self.call_hook("after_val_epoch")
        self.engine.train(True)
        if acc is not None:
            self.logger.info(f"val accuracy {acc:.4f}")
        return acc

    def stop(self):
        self.should_stop = True
