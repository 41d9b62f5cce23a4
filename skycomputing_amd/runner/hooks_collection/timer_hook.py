"""Per-phase timing hook.

Replaces the reference's shared-file DistributedTimerHelperHook
(reference: scaelum/runner/hooks_collection/distributed_timer_helper_hook.py:10-16)
with in-memory host timestamps + stage HIP-event forward times; summarized
per epoch.
"""

from __future__ import annotations

from ...registry import HOOKS
from ..hooks import Hook


@HOOKS.register_module
class TimerHook(Hook):
    def __init__(self, trace_path: str | None = None):
        """``trace_path``: optional chrome://tracing JSON written at run
        end (per rank: ``{trace_path}.rank{r}.json``) from the recorded
        iteration intervals."""
        self.trace_path = trace_path

    def before_run(self, runner):
        runner.timer.clean()

    def before_train_iter(self, runner):
        runner.timer.add_timestamp("iter")

    def after_train_iter(self, runner):
        runner.timer.add_timestamp("iter")

    def after_train_epoch(self, runner):
        if runner.iter_times:
            n = len(runner.iter_times)
            mean = sum(runner.iter_times) / n
            runner.logger.info(
                f"epoch {runner.epoch}: {n} iters, mean {mean*1e3:.1f} ms/iter"
            )
        stage = getattr(runner.engine, "stage", None)
        if stage is not None and stage.forward_time:
            runner.logger.info(
                f"stage {runner.engine.stage_idx}: total fwd "
                f"{stage.total_forward_time()*1e3:.1f} ms over {len(stage.forward_time)} calls"
            )

    def after_run(self, runner):
        if self.trace_path:
            rank = getattr(runner.comm, "rank", 0) if getattr(runner, "comm", None) else 0
            runner.timer.export_chrome_trace(
                f"{self.trace_path}.rank{rank}.json", rank=rank
            )
        runner.timer.clean()
