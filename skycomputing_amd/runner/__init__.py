from .hooks import Hook
from .hooks_collection import (
    CheckpointHook, LRScheduleHook, MemoryHook, MetricsHook, StopHook,
    TimerHook,
)
from .runner import Runner

__all__ = ["Runner", "Hook", "CheckpointHook", "LRScheduleHook", "MemoryHook", "MetricsHook", "StopHook", "TimerHook"]
