"""YAML + dotted-override config loading.

MI355X-native replacement for the reference's Hydra pipeline
(reference: galvatron/core/arguments.py:124-154 `load_with_hydra`):
plain YAML -> deep-merge -> dotted `a.b.c=value` overrides -> pydantic
validation.  No hydra/omegaconf dependency.
"""
from __future__ import annotations

import ast
import copy
from pathlib import Path
from typing import Any, Dict, List, Optional, Sequence, Union

import yaml

from .schema import GalvatronConfig
from .model_configs import resolve_model_config


def _deep_merge(base: Dict[str, Any], extra: Dict[str, Any]) -> Dict[str, Any]:
    out = copy.deepcopy(base)
    for k, v in (extra or {}).items():
        if isinstance(v, dict) and isinstance(out.get(k), dict):
            out[k] = _deep_merge(out[k], v)
        else:
            out[k] = copy.deepcopy(v)
    return out


def _parse_value(text: str) -> Any:
    """Parse an override value: python literal if possible, else string."""
    t = text.strip()
    low = t.lower()
    if low in ("true", "false"):
        return low == "true"
    if low in ("null", "none", "~"):
        return None
    try:
        return ast.literal_eval(t)
    except (ValueError, SyntaxError):
        return t


def apply_overrides(cfg: Dict[str, Any], overrides: Sequence[str]) -> Dict[str, Any]:
    """Apply `a.b.c=value` dotted overrides to a nested dict."""
    out = copy.deepcopy(cfg)
    for ov in overrides or []:
        if "=" not in ov:
            raise ValueError(f"override must be key=value, got: {ov!r}")
        key, _, raw = ov.partition("=")
        parts = key.strip().split(".")
        node = out
        for p in parts[:-1]:
            node = node.setdefault(p, {})
            if not isinstance(node, dict):
                raise ValueError(f"override path {key} crosses a non-dict node")
        node[parts[-1]] = _parse_value(raw)
    return out


def load_config(
    yaml_path: Optional[Union[str, Path]] = None,
    overrides: Optional[Sequence[str]] = None,
    base: Optional[Dict[str, Any]] = None,
) -> GalvatronConfig:
    """Load a GalvatronConfig from YAML (optional) + dotted overrides.

    If the merged config names a known model preset (``model.model_name``) and
    leaves architecture fields unset, the preset fills them in
    (reference: utils/hf_config_adapter.py:285 resolve_model_config).
    """
    raw: Dict[str, Any] = dict(base or {})
    if yaml_path is not None:
        with open(yaml_path) as f:
            file_cfg = yaml.safe_load(f) or {}
        raw = _deep_merge(raw, file_cfg)
    raw = apply_overrides(raw, overrides or [])
    raw = resolve_model_config(raw)
    return GalvatronConfig(**raw)


def config_from_cli(argv: Optional[List[str]] = None) -> GalvatronConfig:
    """Entry-point helper: ``prog [cfg.yaml] [a.b=c ...]``."""
    import sys

    args = list(sys.argv[1:] if argv is None else argv)
    yaml_path = None
    if args and not ("=" in args[0]) and args[0].endswith((".yaml", ".yml")):
        yaml_path = args.pop(0)
    return load_config(yaml_path, args)
