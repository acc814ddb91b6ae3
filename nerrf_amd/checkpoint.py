"""Checkpoint layout.

The reference defines no checkpoint format (no training code exists); this
layout is the one BASELINE.json requires the new build to define:

    <dir>/
      checkpoint.json        # manifest: step, epoch, config, metrics, version
      model.safetensors      # joint model weights (safetensors, zero-copy mmap)
      optimizer.pt           # optimizer + scheduler state (torch.save)

Rank 0 writes; all ranks can load.  `save_checkpoint` is atomic (tmp + rename)
so a killed run never leaves a half-written checkpoint behind.
"""
from __future__ import annotations

import dataclasses
import json
import os
import tempfile
from pathlib import Path
from typing import Any, Dict, Optional, Tuple

import torch

try:
    from safetensors.torch import load_file as _st_load
    from safetensors.torch import save_file as _st_save

    _HAVE_ST = True
except ImportError:  # pragma: no cover
    _HAVE_ST = False

FORMAT_VERSION = 1


def _config_to_dict(cfg: Any) -> Any:
    if dataclasses.is_dataclass(cfg):
        return {f.name: _config_to_dict(getattr(cfg, f.name)) for f in dataclasses.fields(cfg)}
    return cfg


def save_checkpoint(
    dirpath: str | Path,
    model: torch.nn.Module,
    optimizer: Optional[torch.optim.Optimizer] = None,
    step: int = 0,
    epoch: int = 0,
    metrics: Optional[Dict[str, float]] = None,
    config: Any = None,
) -> Path:
    dirpath = Path(dirpath)
    dirpath.mkdir(parents=True, exist_ok=True)
    state = {k: v.detach().cpu().contiguous() for k, v in model.state_dict().items()}
    if _HAVE_ST:
        fd, tmp = tempfile.mkstemp(dir=dirpath, suffix=".st.tmp")
        os.close(fd)
        _st_save(state, tmp)
        os.replace(tmp, dirpath / "model.safetensors")
    else:
        fd, tmp = tempfile.mkstemp(dir=dirpath, suffix=".pt.tmp")
        os.close(fd)
        torch.save(state, tmp)
        os.replace(tmp, dirpath / "model.pt")
    if optimizer is not None:
        fd, tmp = tempfile.mkstemp(dir=dirpath, suffix=".opt.tmp")
        os.close(fd)
        torch.save({"optimizer": optimizer.state_dict()}, tmp)
        os.replace(tmp, dirpath / "optimizer.pt")
    manifest = {
        "format_version": FORMAT_VERSION,
        "step": step,
        "epoch": epoch,
        "metrics": metrics or {},
        "config": _config_to_dict(config) if config is not None else None,
        "weights_file": "model.safetensors" if _HAVE_ST else "model.pt",
    }
    fd, tmp = tempfile.mkstemp(dir=dirpath, suffix=".json.tmp")
    with os.fdopen(fd, "w") as fh:
        json.dump(manifest, fh, indent=2)
    os.replace(tmp, dirpath / "checkpoint.json")
    return dirpath


def load_checkpoint(
    dirpath: str | Path,
    model: torch.nn.Module,
    optimizer: Optional[torch.optim.Optimizer] = None,
    map_location: str = "cpu",
) -> Tuple[Dict[str, Any], torch.nn.Module]:
    dirpath = Path(dirpath)
    with open(dirpath / "checkpoint.json") as fh:
        manifest = json.load(fh)
    wf = dirpath / manifest["weights_file"]
    if wf.suffix == ".safetensors":
        state = _st_load(str(wf))
    else:
        state = torch.load(wf, map_location=map_location, weights_only=True)
    model.load_state_dict(state)
    if optimizer is not None and (dirpath / "optimizer.pt").exists():
        opt_state = torch.load(dirpath / "optimizer.pt", map_location=map_location, weights_only=False)
        optimizer.load_state_dict(opt_state["optimizer"])
    return manifest, model
