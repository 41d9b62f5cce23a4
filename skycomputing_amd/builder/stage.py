"""StageModule — the per-rank pipeline stage executor.

MI355X-native replacement for the reference's ModuleWrapper
(reference: scaelum/builder/module_wrapper.py:22-299). Differences by design:

  * activations stay DEVICE-RESIDENT end to end — there is no output_to_cpu
    CPU staging (the reference round-tripped every hop through host memory,
    module_wrapper.py:172-175); inter-stage movement is RCCL P2P over xGMI,
    handled by the pipeline engine, not the stage;
  * forward timing uses HIP events (torch.cuda.Event) instead of host
    wall-clock + cuda.synchronize per call;
  * memory probing uses hipMemGetInfo via torch.cuda.mem_get_info instead of
    shelling out to nvidia-smi (module_wrapper.py:210-219);
  * heterogeneity simulation enqueues a GPU busy-spin (torch.cuda._sleep)
    on the stream, so a "slow" stage stays slow without serializing the
    host (module_wrapper.py:124-126 slept on the host).
"""

from __future__ import annotations

import time

import torch
import torch.nn as nn

from ..timer import DeviceTimer
from .sequential import SequentialWrapper

_CYCLES_PER_SEC: float | None = None


def _gpu_cycles_per_sec() -> float:
    """Calibrate torch.cuda._sleep cycles/second once."""
    global _CYCLES_PER_SEC
    if _CYCLES_PER_SEC is None:
        probe = 20_000_000
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        torch.cuda._sleep(probe)
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        _CYCLES_PER_SEC = probe / max(dt, 1e-6)
    return _CYCLES_PER_SEC


def device_sleep(seconds: float):
    """Enqueue a busy-spin of ~`seconds` on the current GPU stream (no host
    sync). On CPU, plain sleep."""
    if seconds <= 0:
        return
    if torch.cuda.is_available():
        torch.cuda._sleep(int(seconds * _gpu_cycles_per_sec()))
    else:
        time.sleep(seconds)


class _BackwardSlowdown(torch.autograd.Function):
    """Identity whose backward enqueues the simulated slowdown for the
    stage's backward segment (reference: module_wrapper.py:240-283)."""

    @staticmethod
    def forward(ctx, x, stage):
        ctx.stage = stage
        return x

    @staticmethod
    def backward(ctx, g):
        st = ctx.stage
        if st.slowdown > 0:
            # backward work ~ 2x forward for GEMM-dominated stages
            device_sleep(2.0 * st.last_forward_time * st.slowdown)
        return g, None


class StageModule(nn.Module):
    def __init__(
        self,
        module: SequentialWrapper,
        device: torch.device | str | None = None,
        dtype: torch.dtype | None = None,
        slowdown: float = 0.0,
        mem_limit: int | None = None,
        record_forward_time: bool = True,
    ):
        super().__init__()
        if not isinstance(module, SequentialWrapper):
            raise TypeError("StageModule wraps a SequentialWrapper")
        self.module = module
        self.slowdown = float(slowdown)
        self.mem_limit = mem_limit
        self.record_forward_time = record_forward_time
        self.forward_time: list[float] = []
        self.last_forward_time = 0.0
        self._timer = DeviceTimer(use_cuda=torch.cuda.is_available())
        if device is not None:
            self.device = torch.device(device)
        else:
            self.device = torch.device("cuda", torch.cuda.current_device()) if torch.cuda.is_available() else torch.device("cpu")
        self.compute_dtype = dtype
        self.to(self.device)
        if dtype is not None:
            self.module.to(dtype)

    def _prep(self, args):
        out = []
        for a in args:
            if torch.is_tensor(a):
                a = a.to(self.device, non_blocking=True)
                if (
                    self.compute_dtype is not None
                    and a.is_floating_point()
                    and a.dtype != self.compute_dtype
                ):
                    a = a.to(self.compute_dtype)
            out.append(a)
        return out

    def forward(self, *args):
        args = self._prep(args)
        if self.record_forward_time:
            self._timer.start()
        out = self.module(*args)
        if self.record_forward_time:
            self._timer.stop()
        if self.slowdown > 0 or self.record_forward_time:
            # reading the timer syncs on the stop event only
            t = self._timer.last() if self.record_forward_time else 0.0
            self.last_forward_time = t
            if self.record_forward_time:
                self.forward_time.append(t)
            if self.slowdown > 0:
                device_sleep(t * self.slowdown)
        if self.slowdown > 0:
            if isinstance(out, tuple):
                out = tuple(
                    _BackwardSlowdown.apply(o, self) if torch.is_tensor(o) and o.requires_grad else o
                    for o in out
                )
            elif torch.is_tensor(out):
                out = _BackwardSlowdown.apply(out, self)
        return out

    def reset_timing(self):
        self.forward_time.clear()
        self._timer.reset()

    def total_forward_time(self) -> float:
        return float(sum(self.forward_time))

    def detect_mem(self) -> int:
        """Free memory budget in bytes (reference: module_wrapper.py:187-224)."""
        if self.mem_limit is not None:
            return int(self.mem_limit)
        if self.device.type == "cuda":
            free, _total = torch.cuda.mem_get_info(self.device)
            return int(free) - 500 * 1024 * 1024  # keep the reference's 500MB reserve
        try:
            import psutil

            return int(psutil.virtual_memory().available)
        except ImportError:
            return 8 << 30

    def get_layer_state_dicts(self) -> list[dict]:
        """Per-layer CPU state dicts in stage order (checkpoint contract)."""
        # floating tensors are normalized to fp32 (partition/dtype-portable
        # contract); integer buffers (e.g. BatchNorm num_batches_tracked)
        # keep their dtype — casting them through float loses exactness
        return [
            {
                k: (v.detach().to("cpu", torch.float32)
                    if v.is_floating_point() else v.detach().cpu())
                for k, v in layer.state_dict().items()
            }
            for layer in self.module
        ]

    def load_layer_state_dicts(self, dicts: list[dict]):
        if len(dicts) != len(self.module):
            raise ValueError(f"expected {len(self.module)} layer state dicts, got {len(dicts)}")
        for layer, sd in zip(self.module, dicts):
            target_dtype = next(layer.parameters()).dtype if any(True for _ in layer.parameters()) else None
            cast = {
                k: (v.to(target_dtype) if target_dtype is not None and v.is_floating_point() else v)
                for k, v in sd.items()
            }
            layer.load_state_dict(cast)
        self.to(self.device)
