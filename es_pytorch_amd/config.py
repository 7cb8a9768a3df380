"""JSON -> attribute-access config.

Replaces the reference's json->munch loader (``src/utils/utils.py:48-53``)
without the third-party ``munch`` dependency. Configs are plain JSON files;
sections are accessed as ``cfg.general.policies_per_gen`` and may be mutated
at runtime by decay schedules (as the reference does in ``obj.py:81-83``).
"""
from __future__ import annotations

import json
from typing import Any, Dict


class AttrDict(dict):
    """dict with attribute access, recursively applied."""

    def __init__(self, d: Dict[str, Any] | None = None, **kw):
        super().__init__()
        d = dict(d or {}, **kw)
        for k, v in d.items():
            self[k] = self._wrap(v)

    @classmethod
    def _wrap(cls, v):
        if isinstance(v, dict) and not isinstance(v, AttrDict):
            return cls(v)
        if isinstance(v, (list, tuple)):
            return type(v)(cls._wrap(x) for x in v)
        return v

    def __getattr__(self, k):
        try:
            return self[k]
        except KeyError as e:
            raise AttributeError(k) from e

    def __setattr__(self, k, v):
        self[k] = self._wrap(v)

    def __delattr__(self, k):
        try:
            del self[k]
        except KeyError as e:
            raise AttributeError(k) from e

    def to_dict(self) -> dict:
        out = {}
        for k, v in self.items():
            if isinstance(v, AttrDict):
                v = v.to_dict()
            elif isinstance(v, (list, tuple)):
                v = type(v)(x.to_dict() if isinstance(x, AttrDict) else x for x in v)
            out[k] = v
        return out


def load_config(path: str) -> AttrDict:
    """Load a JSON config file into an AttrDict (reference ``utils.py:48-53``)."""
    with open(path) as f:
        return AttrDict(json.load(f))


def parse_args(argv=None) -> str:
    """Single positional config-file argument (reference ``utils.py:42-45``)."""
    import argparse

    p = argparse.ArgumentParser(description="es_pytorch_amd")
    p.add_argument("config", type=str, help="JSON config file")
    return p.parse_args(argv).config


def merge_override(base: dict, override: dict, _path: str = "") -> dict:
    """Recursive dict merge used by batch sweeps (reference ``batch_run.py:13-26``).

    Errors on keys in ``override`` that do not exist in ``base``.
    """
    for k, v in override.items():
        if k not in base:
            raise KeyError(f"unknown config key: {_path}{k}")
        if isinstance(v, dict) and isinstance(base[k], dict):
            merge_override(base[k], v, f"{_path}{k}.")
        else:
            base[k] = v
    return base
