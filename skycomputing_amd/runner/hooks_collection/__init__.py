from .checkpoint_hook import CheckpointHook
from .metrics_hook import MetricsHook
from .stop_hook import StopHook
from .timer_hook import TimerHook

__all__ = ["CheckpointHook", "MetricsHook", "StopHook", "TimerHook"]
