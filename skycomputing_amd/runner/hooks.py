"""Hook lifecycle base.

Capability parity with the reference's mmcv-style Hook
(reference: scaelum/runner/hooks.py:5-58)."""

from __future__ import annotations


class Hook:
    def before_run(self, runner):
        pass

    def after_run(self, runner):
        pass

    def before_train_epoch(self, runner):
        pass

    def after_train_epoch(self, runner):
        pass

    def before_train_iter(self, runner):
        pass

    def after_train_iter(self, runner):
        pass

    def before_val_epoch(self, runner):
        pass

    def after_val_epoch(self, runner):
        pass

    def before_val_iter(self, runner):
        pass

    def after_val_iter(self, runner):
        pass

    @staticmethod
    def every_n_epochs(runner, n: int) -> bool:
        return n > 0 and (runner.epoch + 1) % n == 0

    @staticmethod
    def every_n_iters(runner, n: int) -> bool:
        return n > 0 and (runner.iter + 1) % n == 0
