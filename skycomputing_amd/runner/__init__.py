from .hooks import Hook
from .hooks_collection import CheckpointHook, StopHook, TimerHook
from .runner import Runner

__all__ = ["Runner", "Hook", "CheckpointHook", "StopHook", "TimerHook"]
