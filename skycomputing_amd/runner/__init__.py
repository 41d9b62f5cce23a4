from .hooks import Hook
from .hooks_collection import CheckpointHook, MetricsHook, StopHook, TimerHook
from .runner import Runner

__all__ = ["Runner", "Hook", "CheckpointHook", "MetricsHook", "StopHook", "TimerHook"]
