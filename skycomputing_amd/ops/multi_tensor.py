"""Multi-tensor SGD launch planning.

Packs per-parameter {param, grad, master, momentum} pointers into a device
descriptor array (one entry per <=1M-element slab) so the whole stage
updates in ONE kernel launch (ops/hip/sgd.hip). The plan is rebuilt only
when a grad tensor's identity changes (autograd reuses .grad buffers, so in
steady state this is built once).
"""

from __future__ import annotations

import os

import torch

from . import hiplib
from .hiplib import check

_SLAB = int(os.environ.get("SKY_SGD_SLAB", 1 << 18))  # elements per descriptor (256k default: 1M-elem slabs win standalone (5.5 vs 4.7 TB/s) but LOSE in-app (+0.6 ms/step) - tools/sgd_bench.py + c7 profile)

_PLAN_CACHE: dict[int, tuple] = {}


def _build_plan(params, grads, masters, moms):
    rows = []
    for i, (p, g) in enumerate(zip(params, grads)):
        n = p.numel()
        pa, ga = p.data_ptr(), g.data_ptr()
        ma = masters[i].data_ptr() if masters is not None else 0
        mo = moms[i].data_ptr() if moms is not None else 0
        esz_p = p.element_size()
        esz_g = g.element_size()
        off = 0
        while off < n:
            cnt = min(_SLAB, n - off)
            rows.append([
                pa + off * esz_p, ga + off * esz_g,
                (ma + off * 4) if ma else 0, (mo + off * 4) if mo else 0,
                cnt, 0,
            ])
            off += cnt
    # pinned staging so the H2D copy is legal inside hipGraph capture
    cpu = torch.tensor(rows, dtype=torch.int64)
    try:
        cpu = cpu.pin_memory()
    except RuntimeError:
        pass
    return cpu.to(params[0].device, non_blocking=True), len(rows)


def multi_tensor_sgd(params, grads, masters, moms, lr, momentum, wd):
    key = id(params)
    sig = tuple(g.data_ptr() for g in grads)
    cached = _PLAN_CACHE.get(key)
    if cached is None or cached[0] != sig:
        desc, n = _build_plan(params, grads, masters, moms)
        _PLAN_CACHE[key] = (sig, desc, n)
    else:
        _, desc, n = cached
    lib = hiplib.require()
    dt = 1 if params[0].dtype == torch.bfloat16 else 0
    flags = (1 if masters is not None else 0) | (2 if moms is not None else 0)
    stream = torch.cuda.current_stream().cuda_stream
    check(
        lib.sky_sgd_step(stream, desc.data_ptr(), n, lr, momentum, wd, dt, flags),
        "sky_sgd_step",
    )
