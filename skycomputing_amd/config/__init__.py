"""Python-file config system.

Capability parity with the reference config loader (reference:
scaelum/config/config.py:10-78): a user ``.py`` file is executed and its
module-level variables harvested into an attribute-dict ``Config``; a
``base = 'other_config.py'`` key pulls in one level of inheritance, with the
child's keys overriding the base's.
"""

from __future__ import annotations

import importlib.util
import os
import types


class Config(dict):
    """dict with attribute access, recursively wrapping nested dicts."""

    def __init__(self, data: dict | None = None):
        super().__init__()
        if data:
            for k, v in data.items():
                self[k] = self._wrap(v)

    @classmethod
    def _wrap(cls, v):
        if isinstance(v, dict) and not isinstance(v, Config):
            return cls(v)
        if isinstance(v, (list, tuple)):
            return type(v)(cls._wrap(x) for x in v)
        return v

    def __setattr__(self, k, v):
        self[k] = self._wrap(v)

    def __getattr__(self, k):
        try:
            return self[k]
        except KeyError as e:
            raise AttributeError(k) from e

    def to_dict(self) -> dict:
        def _unwrap(v):
            if isinstance(v, Config):
                return {k: _unwrap(x) for k, x in v.items()}
            if isinstance(v, (list, tuple)):
                return type(v)(_unwrap(x) for x in v)
            return v

        return {k: _unwrap(v) for k, v in self.items()}


def _py2dict(path: str) -> dict:
    path = os.path.abspath(os.path.expanduser(path))
    if not os.path.isfile(path):
        raise FileNotFoundError(path)
    if not path.endswith(".py"):
        raise ValueError(f"config file must be a .py file, got {path}")
    spec = importlib.util.spec_from_file_location("_sky_config", path)
    module = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(module)
    cfg = {
        k: v
        for k, v in vars(module).items()
        if not k.startswith("__")
        and not isinstance(v, types.ModuleType)
        and not isinstance(v, (types.FunctionType, type))
    }
    return cfg, path


def load_config(path: str) -> Config:
    cfg, abspath = _py2dict(path)
    base = cfg.pop("base", None)
    if base is not None:
        base_path = os.path.join(os.path.dirname(abspath), base)
        base_cfg, _ = _py2dict(base_path)
        base_cfg.pop("base", None)
        base_cfg.update(cfg)
        cfg = base_cfg
    return Config(cfg)
