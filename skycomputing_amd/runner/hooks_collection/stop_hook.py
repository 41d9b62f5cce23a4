"""Cooperative cancellation hook.

Capability parity with the reference StopHook
(reference: scaelum/runner/hooks_collection/stop_hook.py:12-38): polls a
``stop_flag.txt`` file after each iteration; any agent can flip it with
``StopHook.stop(root)``. The decision is taken on rank 0 and broadcast so
all ranks leave the loop together (a lone rank stopping would deadlock the
pipeline). The reference's broken ``runner.max_iters/max_epochs`` attribute
reads (stop_hook.py:23-24) are not reproduced.
"""

from __future__ import annotations

import os

from ...registry import HOOKS
from ..hooks import Hook


@HOOKS.register_module
class StopHook(Hook):
    FLAG_NAME = "stop_flag.txt"

    def __init__(self, root: str = ".", poll_interval: int = 1):
        self.root = root
        self.poll_interval = poll_interval

    def _flag_path(self) -> str:
        return os.path.join(self.root, self.FLAG_NAME)

    def before_run(self, runner):
        if runner.comm.rank == 0 and os.path.isfile(self._flag_path()):
            os.remove(self._flag_path())

    def after_train_iter(self, runner):
        if not self.every_n_iters(runner, self.poll_interval):
            return
        flagged = os.path.isfile(self._flag_path()) if runner.comm.rank == 0 else None
        flagged = runner.comm.broadcast_object(flagged, src=0)
        if flagged:
            runner.logger.info("stop flag detected; stopping")
            runner.stop()

    def after_run(self, runner):
        if runner.comm.rank == 0 and os.path.isfile(self._flag_path()):
            os.remove(self._flag_path())

    @staticmethod
    def stop(root: str = "."):
        with open(os.path.join(root, StopHook.FLAG_NAME), "w") as f:
            f.write("stop\n")
