"""Per-iteration metrics stream.

Writes one JSON line per training iteration (rank 0 of the loss broadcast
sees the same loss everywhere, so every rank's file is equivalent for loss;
iteration wall time is per-rank). Machine-readable counterpart to the text
Logger — the reference had no metrics stream at all.
"""

from __future__ import annotations

import json
import time

from ...registry import HOOKS
from ..hooks import Hook


@HOOKS.register_module
class MetricsHook(Hook):
    def __init__(self, path: str, rank_zero_only: bool = True,
                 flush_interval: int = 1):
        self.path = path
        self.rank_zero_only = rank_zero_only
        self.flush_interval = max(1, flush_interval)
        self._fh = None

    def _active(self, runner) -> bool:
        if not self.rank_zero_only:
            return True
        comm = getattr(runner, "comm", None)
        return comm is None or comm.rank == 0

    def before_run(self, runner):
        if self._active(runner):
            self._fh = open(self.path, "a")

    def after_train_iter(self, runner):
        if self._fh is None:
            return
        rec = {
            "t": time.time(),
            "epoch": runner.epoch,
            "iter": runner.iter,
            "loss": runner.last_loss,
            "iter_time_s": runner.iter_times[-1] if runner.iter_times else None,
        }
        self._fh.write(json.dumps(rec) + "\n")
        if (runner.iter + 1) % self.flush_interval == 0:
            self._fh.flush()

    def after_run(self, runner):
        if self._fh is not None:
            self._fh.flush()
            self._fh.close()
            self._fh = None
