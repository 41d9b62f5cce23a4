from .checkpoint_hook import CheckpointHook
from .lr_hook import LRScheduleHook
from .memory_hook import MemoryHook
from .metrics_hook import MetricsHook
from .stop_hook import StopHook
from .timer_hook import TimerHook

__all__ = ["CheckpointHook", "LRScheduleHook", "MemoryHook", "MetricsHook",
           "StopHook", "TimerHook"]
