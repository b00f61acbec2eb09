"""Checkpoint layout for robust-training state.

The reference has no checkpoint subsystem; this defines the real layout:
a directory with
  meta.json          — round counter, world size, layout version, shapes
  model.safetensors  — model state dict (or flat parameter vector)
  state.safetensors  — aggregator state (e.g. iterative centers), optional
Sharded (multi-GPU) checkpoints write one shard file per rank
(``model.rank{r}.safetensors``) so 288 GB-scale states never funnel
through one process.
"""
from __future__ import annotations

import json
from pathlib import Path
from typing import Any, Dict, Optional

import torch

LAYOUT_VERSION = 1


def _save_tensors(path: Path, tensors: Dict[str, torch.Tensor]) -> None:
    try:
        from safetensors.torch import save_file

        save_file({k: v.detach().cpu().contiguous() for k, v in tensors.items()}, str(path))
    except ImportError:
        torch.save({k: v.detach().cpu() for k, v in tensors.items()}, str(path))


def _load_tensors(path: Path) -> Dict[str, torch.Tensor]:
    try:
        from safetensors.torch import load_file
    except ImportError:
        # environment without safetensors: files were written by the
        # torch.save fallback in _save_tensors
        return torch.load(str(path), map_location="cpu", weights_only=True)
    # safetensors installed -> the file is safetensors-format; let a
    # corruption/truncation error propagate with its original message
    return load_file(str(path))


def save_checkpoint(
    directory: str,
    *,
    round_idx: int,
    model_state: Dict[str, torch.Tensor],
    aggregator_state: Optional[Dict[str, torch.Tensor]] = None,
    rank: int = 0,
    world_size: int = 1,
    extra_meta: Optional[dict] = None,
) -> None:
    d = Path(directory)
    d.mkdir(parents=True, exist_ok=True)
    suffix = f".rank{rank}" if world_size > 1 else ""
    _save_tensors(d / f"model{suffix}.safetensors", model_state)
    if aggregator_state:
        _save_tensors(d / f"state{suffix}.safetensors", aggregator_state)
    if rank == 0:
        meta = {
            "layout_version": LAYOUT_VERSION,
            "round": int(round_idx),
            "world_size": int(world_size),
            "has_aggregator_state": bool(aggregator_state),
        }
        meta.update(extra_meta or {})
        (d / "meta.json").write_text(json.dumps(meta, indent=2))


def load_checkpoint(
    directory: str, *, rank: int = 0
) -> Dict[str, Any]:
    d = Path(directory)
    meta = json.loads((d / "meta.json").read_text())
    world = int(meta.get("world_size", 1))
    suffix = f".rank{rank}" if world > 1 else ""
    out: Dict[str, Any] = {"meta": meta}
    out["model_state"] = _load_tensors(d / f"model{suffix}.safetensors")
    state_path = d / f"state{suffix}.safetensors"
    if state_path.exists():
        out["aggregator_state"] = _load_tensors(state_path)
    return out
