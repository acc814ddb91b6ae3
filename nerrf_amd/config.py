"""YAML + dotted-CLI-override configuration.

The reference has no config system (one env var; constants in scripts —
SURVEY.md §5), so this is new surface: dataclass-backed config tree, YAML
file load, and `--set a.b.c=value` overrides.
"""
from __future__ import annotations

import dataclasses
from dataclasses import dataclass, field
from pathlib import Path
from typing import Any, List, Optional

import yaml

from .models.joint import JointConfig


@dataclass
class DataConfig:
    n_scenarios: int = 8
    duration_s: float = 120.0
    benign_rate_hz: float = 800.0
    window_s: float = 30.0
    stride_s: float = 15.0
    attack_fraction: float = 0.6
    scenario_kinds: tuple = ("lockbit", "supply_chain")
    config_jitter: bool = False  # vary duration/rate per scenario (see dataset)
    fanout: int = 16
    seq_len: int = 100
    seed: int = 0


@dataclass
class OptimConfig:
    lr: float = 1e-3
    weight_decay: float = 1e-4
    epochs: int = 3
    grad_clip: float = 1.0
    dtype: str = "float32"  # float32 | bfloat16


@dataclass
class RunConfig:
    device: str = "auto"  # auto | cpu | cuda
    checkpoint_dir: str = "checkpoints/run"
    save_every_epochs: int = 1
    log_every: int = 20
    eval_holdout: int = 4  # scenarios held out for eval


@dataclass
class TrainConfig:
    model: JointConfig = field(default_factory=JointConfig)
    data: DataConfig = field(default_factory=DataConfig)
    optim: OptimConfig = field(default_factory=OptimConfig)
    run: RunConfig = field(default_factory=RunConfig)


def _from_dict(cls, d: Any):
    if d is None:
        return cls()
    if not dataclasses.is_dataclass(cls):
        return d
    kwargs = {}
    fields = {f.name: f for f in dataclasses.fields(cls)}
    for k, v in (d or {}).items():
        if k not in fields:
            raise KeyError(f"unknown config key '{k}' for {cls.__name__}")
        ftype = fields[k].type
        sub = _nested_dataclass(cls, k)
        kwargs[k] = _from_dict(sub, v) if sub else v
    return cls(**kwargs)


def _nested_dataclass(cls, name: str):
    for f in dataclasses.fields(cls):
        if f.name == name:
            default = f.default_factory() if f.default_factory is not dataclasses.MISSING else None  # type: ignore[misc]
            if dataclasses.is_dataclass(default):
                return type(default)
    return None


def load_config(path: Optional[str | Path] = None, overrides: Optional[List[str]] = None) -> TrainConfig:
    d: dict = {}
    if path is not None:
        with open(path) as fh:
            d = yaml.safe_load(fh) or {}
    cfg = _from_dict(TrainConfig, d)
    for ov in overrides or []:
        if "=" not in ov:
            raise ValueError(f"override must be key.path=value, got '{ov}'")
        key, val = ov.split("=", 1)
        _apply_override(cfg, key.strip(), val.strip())
    return cfg


def _apply_override(cfg: Any, dotted: str, raw: str) -> None:
    parts = dotted.split(".")
    obj = cfg
    for p in parts[:-1]:
        if not hasattr(obj, p):
            raise KeyError(f"unknown config path '{dotted}' (at '{p}')")
        obj = getattr(obj, p)
    leaf = parts[-1]
    if not hasattr(obj, leaf):
        raise KeyError(f"unknown config leaf '{dotted}'")
    cur = getattr(obj, leaf)
    val: Any = raw
    try:
        val = yaml.safe_load(raw)
    except yaml.YAMLError:
        pass
    if isinstance(cur, (tuple, list)) and val is not None:
        # sequence-typed leaf: accept "(a,b,c)", "a,b,c" or YAML "[a, b]"
        # (a bare type(cur)(str) would split the string into characters)
        if isinstance(val, str):
            val = type(cur)(
                x.strip() for x in val.strip("()[]").split(",") if x.strip()
            )
        else:
            val = type(cur)(val if isinstance(val, (list, tuple)) else [val])
    elif cur is not None and val is not None and not isinstance(val, type(cur)):
        try:
            val = type(cur)(val)
        except (TypeError, ValueError):
            pass
    setattr(obj, leaf, val)


def to_dict(cfg: Any) -> Any:
    if dataclasses.is_dataclass(cfg):
        return {f.name: to_dict(getattr(cfg, f.name)) for f in dataclasses.fields(cfg)}
    return cfg
