"""Per-layer cost estimation.

Replaces the reference's pthflops jit-tracing + hook-based memory estimate
(reference: scaelum/dynamics/estimator.py:13-152) with ANALYTIC flops from
the layer zoo (each registered layer implements ``layer_flops``) and an
analytic memory model sized for the MI355X training configuration
(bf16 params + bf16 grads + fp32 master copy + bf16 activations & grads).
Layers without an analytic hook fall back to a module-walk estimate
(Linear/Conv2d GEMM flops).
"""

from __future__ import annotations

import torch
import torch.nn as nn


class Estimator:
    # bytes per parameter: bf16 param (2) + bf16 grad (2) + fp32 master (4)
    PARAM_BYTES = 8
    # bytes per activation element: bf16 act (2) + bf16 grad (2)
    ACT_BYTES = 4

    @staticmethod
    def _fallback_flops(module: nn.Module, batch: int, seq: int) -> float:
        flops = 0.0
        for m in module.modules():
            if isinstance(m, nn.Linear):
                flops += 2.0 * batch * seq * m.in_features * m.out_features
            elif isinstance(m, nn.Conv2d):
                # rough: assumes spatial size `seq` x `seq`
                k = m.kernel_size[0] * m.kernel_size[1]
                flops += 2.0 * batch * seq * seq * m.in_channels * m.out_channels * k
        return flops

    @classmethod
    def layer_flops(cls, layer: nn.Module, batch: int, seq: int) -> float:
        if hasattr(layer, "layer_flops"):
            return float(layer.layer_flops(batch, seq))
        return cls._fallback_flops(layer, batch, seq)

    @classmethod
    def layer_mem_bytes(cls, layer: nn.Module, batch: int, seq: int) -> float:
        params = sum(p.numel() for p in layer.parameters())
        if hasattr(layer, "activation_numel"):
            acts = layer.activation_numel(batch, seq)
        else:
            acts = batch * seq * 1024  # conservative default
        return params * cls.PARAM_BYTES + acts * cls.ACT_BYTES

    @staticmethod
    def benchmark_speed(
        module: nn.Module,
        data_generator,
        iterations: int = 10,
        warmup: int = 2,
        backward: bool = True,
        device: torch.device | None = None,
    ) -> float:
        """Timed fwd(+bwd) iterations of a probe module; returns seconds for
        the timed iterations (replaces estimator.py:15-34, which read the
        wrapper's timing side channel)."""
        import time

        use_cuda = device is not None and device.type == "cuda"
        data = data_generator.generate()

        def to_dev(x):
            if torch.is_tensor(x):
                return x.to(device) if device is not None else x
            return type(x)(to_dev(t) for t in x)

        data = to_dev(data)

        def run_once():
            if isinstance(data, (tuple, list)):
                out = module(*data)
            else:
                out = module(data)
            if backward:
                outs = out if isinstance(out, (tuple, list)) else (out,)
                grads_out = [o for o in outs if torch.is_tensor(o) and o.requires_grad]
                if grads_out:
                    torch.autograd.backward(
                        grads_out, [torch.ones_like(o) for o in grads_out]
                    )
                    module.zero_grad(set_to_none=True)

        for _ in range(warmup):
            run_once()
        if use_cuda:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iterations):
            run_once()
        if use_cuda:
            torch.cuda.synchronize()
        return time.perf_counter() - t0
