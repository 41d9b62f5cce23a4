"""Builder — layer-config lists -> runnable modules.

Capability parity with the reference builder
(reference: scaelum/builder/builder.py:12-49): generic construction from the
registries plus stage assembly (SequentialWrapper + StageModule).
"""

from __future__ import annotations

import copy

import torch

from ..registry import DATA_GENERATOR, DATASET, HOOKS, LAYER, Registry
from .sequential import SequentialWrapper
from .stage import StageModule, device_sleep

__all__ = [
    "build_from_registry", "build_layer", "build_hook", "build_dataset",
    "build_data_generator", "build_module_from_cfg", "build_dataloader_from_cfg",
    "SequentialWrapper", "StageModule", "device_sleep",
]


def build_from_registry(cfg: dict, registry: Registry):
    cfg = copy.deepcopy(dict(cfg))
    key = "layer_type" if "layer_type" in cfg else "type"
    if key not in cfg:
        raise KeyError(f"config {cfg} has no 'layer_type'/'type' key")
    cls = registry.get_module(cfg.pop(key))
    return cls(**cfg)


def build_layer(cfg: dict):
    return build_from_registry(cfg, LAYER)


def build_hook(cfg: dict):
    return build_from_registry(cfg, HOOKS)


def build_dataset(cfg: dict):
    return build_from_registry(cfg, DATASET)


def build_data_generator(cfg: dict):
    return build_from_registry(cfg, DATA_GENERATOR)


def build_module_from_cfg(
    layer_cfgs: list[dict],
    device=None,
    dtype=None,
    slowdown: float = 0.0,
    mem_limit: int | None = None,
    record_forward_time: bool = True,
    **stage_kwargs,
) -> StageModule:
    """Layer-config list -> SequentialWrapper -> StageModule
    (reference: scaelum/builder/builder.py:29-41)."""
    layers = [build_layer(c) for c in layer_cfgs]
    seq = SequentialWrapper(*layers)
    return StageModule(
        seq,
        device=device,
        dtype=dtype,
        slowdown=slowdown,
        mem_limit=mem_limit,
        record_forward_time=record_forward_time,
        **stage_kwargs,
    )


def build_dataloader_from_cfg(batch_size: int, dataset_cfg: dict, **loader_kwargs):
    """(reference: scaelum/builder/builder.py:44-49)"""
    dataset = build_dataset(dataset_cfg)
    defaults = dict(shuffle=False, drop_last=True, num_workers=0)
    defaults.update(loader_kwargs)
    # Every rank iterates its own DataLoader and the pipeline pairs the
    # first stage's inputs with the last stage's labels, so iteration MUST
    # be rank-identical. With shuffle=True that only holds if the RNG is
    # shared: pin a fixed-seed generator unless the caller supplied one.
    if defaults.get("shuffle") and "generator" not in defaults:
        g = torch.Generator()
        g.manual_seed(0x5C1C)
        defaults["generator"] = g
    return torch.utils.data.DataLoader(dataset, batch_size=batch_size, **defaults)
