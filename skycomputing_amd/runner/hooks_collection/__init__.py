from .checkpoint_hook import CheckpointHook
from .stop_hook import StopHook
from .timer_hook import TimerHook

__all__ = ["CheckpointHook", "StopHook", "TimerHook"]
