"""Checkpoint save/restore hook.

Capability parity with the reference CheckpointHook
(reference: scaelum/runner/hooks_collection/checkpoint_hook.py:13-74):
``before_run`` restores from a checkpoint file (the reference's restore
path crashed on a nonexistent method, rpc_module.py:64,93 — fixed here);
``after_train_epoch`` every ``save_interval`` epochs gathers all stages'
weights into the ParameterServer and writes ``epoch_{n}.pth``. Collectives
run over the gloo control plane; every rank participates.
"""

from __future__ import annotations

import os

from ...dynamics.parameter_server import ParameterServer
from ...registry import HOOKS
from ..hooks import Hook


@HOOKS.register_module
class CheckpointHook(Hook):
    def __init__(
        self,
        save_path: str,
        save_interval: int = 1,
        load_from: str | None = None,
        num_layers: int | None = None,
        resume_counters: bool = False,
        save_best: bool = False,
    ):
        """``resume_counters``: also restore runner.epoch/iter from the
        checkpoint meta (continue the schedule where it stopped) instead
        of restarting the counters with restored weights.
        ``save_best``: additionally write ``best.pth`` whenever a val
        epoch improves ``runner.last_val_acc``."""
        self.save_path = save_path
        self.save_interval = save_interval
        self.load_from = load_from
        self.num_layers = num_layers
        self.resume_counters = resume_counters
        self.save_best = save_best
        self._best_acc: float | None = None
        self._ps: ParameterServer | None = None

    def _server(self, runner) -> ParameterServer:
        if self._ps is None:
            n = self.num_layers or len(runner.engine.layer_cfgs)
            self._ps = ParameterServer(n)
        return self._ps

    def before_run(self, runner):
        if not self.load_from:
            return
        ps = self._server(runner)
        meta = None
        if runner.comm.rank == 0:
            meta = ps.load_weights_from_file(self.load_from)
        ps.scatter_to_engine(runner.engine, runner.comm)
        # the optimizer cloned its fp32 masters at construction time, i.e.
        # BEFORE this restore — refresh them (and reset momentum) or the
        # first step() writes stale_master - lr*grad over the restored
        # weights
        if hasattr(runner.optimizer, "sync_masters"):
            runner.optimizer.sync_masters()
        if self.resume_counters:
            meta = runner.comm.broadcast_object(meta, src=0) or {}
            runner.epoch = int(meta.get("epoch", 0))
            runner.iter = int(meta.get("iter", 0))
        runner.logger.info(f"restored checkpoint from {self.load_from}")

    def after_train_epoch(self, runner):
        if not self.every_n_epochs(runner, self.save_interval):
            return
        ps = self._server(runner)
        ps.gather_from_engine(runner.engine, runner.comm)
        if runner.comm.rank == 0:
            os.makedirs(self.save_path, exist_ok=True)
            path = os.path.join(self.save_path, f"epoch_{runner.epoch + 1}.pth")
            ps.save_weights_to_file(path, meta={"epoch": runner.epoch + 1, "iter": runner.iter})
            runner.logger.info(f"saved checkpoint {path}")

    def after_val_epoch(self, runner):
        if not self.save_best:
            return
        acc = getattr(runner, "last_val_acc", None)
        if acc is None or (self._best_acc is not None and acc <= self._best_acc):
            return
        self._best_acc = acc
        ps = self._server(runner)
        ps.gather_from_engine(runner.engine, runner.comm)
        if runner.comm.rank == 0:
            os.makedirs(self.save_path, exist_ok=True)
            path = os.path.join(self.save_path, "best.pth")
            ps.save_weights_to_file(
                path, meta={"epoch": runner.epoch, "iter": runner.iter,
                            "val_acc": acc})
            runner.logger.info(f"saved best checkpoint {path} (acc {acc:.4f})")
