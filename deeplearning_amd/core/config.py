"""One config system for all subprojects: nested dot-access nodes + YAML files
(with _BASE_ inheritance) + CLI dotted-key overrides.

Replaces the reference's four config tiers (raw argparse,
argparse+YAML, yacs CfgNode w/ BASE inheritance (swin config.py), YOLOX Exp
classes) with one implementation.
"""
from __future__ import annotations

import copy
import json
from pathlib import Path
from typing import Any

import yaml

_BASE_KEY = "_BASE_"


class CfgNode(dict):
    """dict with attribute access, freeze support and YAML/CLI merging."""

    def __init__(self, init: dict | None = None):
        super().__init__()
        self.__dict__["_frozen"] = False
        if init:
            for k, v in init.items():
                self[k] = CfgNode(v) if isinstance(v, dict) else v

    # -- attribute protocol --------------------------------------------------
    def __getattr__(self, name: str) -> Any:
        try:
            return self[name]
        except KeyError as e:
            raise AttributeError(name) from e

    def __setattr__(self, name: str, value: Any) -> None:
        if self.__dict__.get("_frozen"):
            raise AttributeError(f"config is frozen; cannot set {name}")
        self[name] = CfgNode(value) if isinstance(value, dict) and not isinstance(value, CfgNode) else value

    def __setitem__(self, key, value):
        if self.__dict__.get("_frozen"):
            raise AttributeError(f"config is frozen; cannot set {key}")
        super().__setitem__(key, CfgNode(value) if isinstance(value, dict) and not isinstance(value, CfgNode) else value)

    # -- lifecycle -----------------------------------------------------------
    def freeze(self, frozen: bool = True) -> "CfgNode":
        self.__dict__["_frozen"] = frozen
        for v in self.values():
            if isinstance(v, CfgNode):
                v.freeze(frozen)
        return self

    def clone(self) -> "CfgNode":
        return CfgNode(copy.deepcopy(self.to_dict()))

    def to_dict(self) -> dict:
        return {k: (v.to_dict() if isinstance(v, CfgNode) else v) for k, v in self.items()}

    # -- merging -------------------------------------------------------------
    def merge_from_dict(self, other: dict) -> "CfgNode":
        for k, v in other.items():
            if isinstance(v, dict) and isinstance(self.get(k), CfgNode):
                self[k].merge_from_dict(v)
            else:
                self[k] = v
        return self

    def merge_from_file(self, path: str | Path) -> "CfgNode":
        path = Path(path)
        with open(path) as f:
            data = yaml.safe_load(f) or {}
        base = data.pop(_BASE_KEY, None)
        if base:
            for b in base if isinstance(base, list) else [base]:
                bp = Path(b)
                self.merge_from_file(bp if bp.is_absolute() else path.parent / bp)
        return self.merge_from_dict(data)

    def merge_from_list(self, opts: list) -> "CfgNode":
        """['train.lr', '0.1', 'model.name', 'resnet50'] style KV overrides."""
        assert len(opts) % 2 == 0, f"override list must be key,value pairs, got {opts}"
        for key, val in zip(opts[0::2], opts[1::2]):
            node = self
            parts = key.split(".")
            for p in parts[:-1]:
                if p not in node:
                    node[p] = CfgNode()
                node = node[p]
            node[parts[-1]] = _parse_value(val, node.get(parts[-1]))
        return self

    def dump(self, path: str | Path) -> None:
        with open(path, "w") as f:
            json.dump(self.to_dict(), f, indent=2, default=str)

    def dump_yaml(self, path: str | Path) -> None:
        with open(path, "w") as f:
            yaml.safe_dump(self.to_dict(), f, sort_keys=False)


def _parse_value(val: Any, old: Any) -> Any:
    if not isinstance(val, str):
        return val
    if old is not None and not isinstance(old, str):
        try:
            return type(old)(yaml.safe_load(val))
        except Exception:
            pass
    try:
        return yaml.safe_load(val)
    except Exception:
        return val


def load_config(defaults: dict, yaml_file: str | None = None, opts: list | None = None,
                freeze: bool = True) -> CfgNode:
    """defaults <- yaml (with _BASE_ chain) <- CLI opts, then freeze."""
    cfg = CfgNode(defaults)
    if yaml_file:
        cfg.merge_from_file(yaml_file)
    if opts:
        cfg.merge_from_list(list(opts))
    return cfg.freeze() if freeze else cfg
