"""Config loading (reference parity: src/common/config Configurable —
layered TOML file + env overrides GREPTIMEDB_AMD__<KEY>)."""

from __future__ import annotations

import os


def load_config(path: str | None) -> dict:
    cfg: dict = {}
    if path:
        import tomli
        with open(path, "rb") as f:
            cfg = _flatten(tomli.load(f))
    prefix = "GREPTIMEDB_AMD__"
    for k, v in os.environ.items():
        if k.startswith(prefix):
            cfg[k[len(prefix):].lower()] = v
    return cfg


def _flatten(d: dict, prefix: str = "") -> dict:
    out = {}
    for k, v in d.items():
        key = f"{prefix}{k}" if not prefix else f"{prefix}.{k}"
        if isinstance(v, dict):
            out.update(_flatten(v, key))
        else:
            out[key] = v
    return out
