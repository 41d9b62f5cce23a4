"""Timers.

Replaces the reference's shared-filesystem timestamp file (reference:
scaelum/timer/timer.py:10-29) with device-event timing: HIP events
(``torch.cuda.Event``) bracket GPU work without host sync; on CPU a
perf_counter fallback is used. Aggregation across ranks goes over the
control-plane process group (gloo) instead of a shared file.
"""

from __future__ import annotations

import time
from collections import defaultdict

import torch


class DeviceTimer:
    """Wall-clock a region of device work with HIP events.

    ``start()``/``stop()`` enqueue events on the current stream; ``elapsed()``
    synchronizes lazily only when the number is read.
    """

    def __init__(self, use_cuda: bool | None = None):
        self._use_cuda = torch.cuda.is_available() if use_cuda is None else use_cuda
        self._pairs: list = []
        self._cpu_start = None

    def start(self):
        if self._use_cuda:
            e = torch.cuda.Event(enable_timing=True)
            e.record()
            self._pairs.append([e, None])
        else:
            self._cpu_start = time.perf_counter()

    def stop(self):
        if self._use_cuda:
            e = torch.cuda.Event(enable_timing=True)
            e.record()
            assert self._pairs and self._pairs[-1][1] is None, "stop() without start()"
            self._pairs[-1][1] = e
        else:
            assert self._cpu_start is not None, "stop() without start()"
            self._pairs.append(time.perf_counter() - self._cpu_start)
            self._cpu_start = None

    def elapsed(self) -> float:
        """Total seconds across all recorded start/stop pairs."""
        if self._use_cuda:
            if not self._pairs:
                return 0.0
            self._pairs[-1][1].synchronize()
            return sum(s.elapsed_time(e) for s, e in self._pairs) / 1e3
        return float(sum(self._pairs))

    def last(self) -> float:
        if not self._pairs:
            return 0.0
        if self._use_cuda:
            s, e = self._pairs[-1]
            e.synchronize()
            return s.elapsed_time(e) / 1e3
        return float(self._pairs[-1])

    def reset(self):
        self._pairs.clear()
        self._cpu_start = None


class DistributedTimer:
    """Named per-rank timers; intervals kept in memory, not a shared file.

    The reference's ``get_prev_interval`` read the delta of the last two
    timestamps in a shared file (reference: scaelum/timer/timer.py:23-29);
    here each named timer records (host) timestamps locally and the same
    query is served from memory.
    """

    def __init__(self):
        self._stamps: dict[str, list[float]] = defaultdict(list)
        self._device_timers: dict[str, DeviceTimer] = {}

    def add_timestamp(self, name: str = "default"):
        self._stamps[name].append(time.perf_counter())

    def get_prev_interval(self, name: str = "default") -> float:
        ts = self._stamps[name]
        if len(ts) < 2:
            return 0.0
        return ts[-1] - ts[-2]

    def device_timer(self, name: str) -> DeviceTimer:
        if name not in self._device_timers:
            self._device_timers[name] = DeviceTimer()
        return self._device_timers[name]

    def clean(self, name: str | None = None):
        if name is None:
            self._stamps.clear()
            self._device_timers.clear()
        else:
            self._stamps.pop(name, None)
            self._device_timers.pop(name, None)

    # back-compat with the reference's file-wipe API surface
    clean_prev_file = clean

    def export_chrome_trace(self, path: str, rank: int = 0):
        """Write recorded timestamp pairs as a chrome://tracing / Perfetto
        JSON: each named timer's consecutive (odd, even) timestamp pairs
        become complete ('X') duration events on a per-rank track. Open
        the file at chrome://tracing or ui.perfetto.dev."""
        import json

        events = []
        for name, ts in self._stamps.items():
            for i in range(0, len(ts) - 1, 2):
                events.append({
                    "name": name,
                    "ph": "X",
                    "ts": ts[i] * 1e6,
                    "dur": (ts[i + 1] - ts[i]) * 1e6,
                    "pid": 0,
                    "tid": rank,
                })
        with open(path, "w") as f:
            json.dump({"traceEvents": events,
                       "displayTimeUnit": "ms"}, f)
