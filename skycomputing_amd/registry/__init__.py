"""String->class registries for the layer-config model IR.

Capability parity with the reference registry (reference:
scaelum/registry/registry.py:8-30): decorator registration, name lookup,
fallback to ``torch.nn`` attributes, and the four singletons LAYER / DATASET /
HOOKS / DATA_GENERATOR consumed by the builder.
"""

from __future__ import annotations

import torch.nn as nn


class Registry:
    """A name -> class map with decorator-style registration.

    Lookup falls back to ``torch.nn`` so configs may name stock torch modules
    (``dict(layer_type='Linear', ...)``) without explicit registration.
    """

    def __init__(self, name: str):
        self._name = name
        self._registry: dict[str, type] = {}

    @property
    def name(self) -> str:
        return self._name

    def register_module(self, cls=None, *, name: str | None = None, force: bool = False):
        def _register(c):
            key = name or c.__name__
            if not force and key in self._registry and self._registry[key] is not c:
                raise KeyError(f"{key!r} already registered in registry {self._name!r}")
            self._registry[key] = c
            return c

        if cls is None:
            return _register
        return _register(cls)

    def get_module(self, name: str) -> type:
        if name in self._registry:
            return self._registry[name]
        if hasattr(nn, name):
            return getattr(nn, name)
        raise KeyError(
            f"{name!r} is not registered in registry {self._name!r} and is not a torch.nn attribute"
        )

    def has_module(self, name: str) -> bool:
        return name in self._registry or hasattr(nn, name)

    def keys(self):
        return self._registry.keys()

    def __contains__(self, name: str) -> bool:
        return self.has_module(name)


LAYER = Registry("layer")
DATASET = Registry("dataset")
HOOKS = Registry("hooks")
DATA_GENERATOR = Registry("data_generator")
