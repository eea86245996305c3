"""Model registry: every factory registers by name so configs/CLIs can build
any model in the zoo with build_model(name, **kwargs)."""
from __future__ import annotations

from typing import Callable

_REGISTRY: dict[str, Callable] = {}


def register_model(fn: Callable = None, name: str | None = None):
    def deco(f):
        _REGISTRY[name or f.__name__] = f
        return f

    return deco(fn) if fn is not None else deco


def build_model(name: str, **kwargs):
    if name not in _REGISTRY:
        raise KeyError(f"unknown model '{name}'. available: {sorted(_REGISTRY)}")
    return _REGISTRY[name](**kwargs)


def list_models(prefix: str = "") -> list:
    return sorted(k for k in _REGISTRY if k.startswith(prefix))
