"""Peak device-memory logging per epoch.

Logs (and optionally resets) the HIP caching-allocator peak per rank —
the number that matters when sizing stages against 288 GB of HBM3E.
No-op on CPU.
"""

from __future__ import annotations

import torch

from ...registry import HOOKS
from ..hooks import Hook


@HOOKS.register_module
class MemoryHook(Hook):
    def __init__(self, reset_each_epoch: bool = True):
        self.reset_each_epoch = reset_each_epoch

    def before_train_epoch(self, runner):
        if torch.cuda.is_available() and self.reset_each_epoch:
            torch.cuda.reset_peak_memory_stats()

    def after_train_epoch(self, runner):
        if not torch.cuda.is_available():
            return
        peak = torch.cuda.max_memory_allocated() / (1 << 30)
        resv = torch.cuda.max_memory_reserved() / (1 << 30)
        free, total = torch.cuda.mem_get_info()
        runner.logger.info(
            f"epoch {runner.epoch}: peak allocated {peak:.2f} GiB, "
            f"reserved {resv:.2f} GiB, device free {free / (1 << 30):.1f}/"
            f"{total / (1 << 30):.1f} GiB"
        )
