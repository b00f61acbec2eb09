"""Gradient flattening / structure restoration.

Reference parity: aggregators/coordinate_wise/_tiling.py:18-38
(flatten_gradients) and the per-file ``_to_like`` clones. MI355X design:
gradients are flattened ONCE into a resident (n, d) matrix on the target
device and all kernels operate on that; ``to_like`` restores dtype/device/
shape (or parameter-list structure) only at the boundary.
"""
from __future__ import annotations

from typing import Any, List, Tuple

import numpy as np
import torch


def flatten_one(g: Any) -> torch.Tensor:
    """Flatten one gradient — a tensor, ndarray, scalar sequence, or a
    list/tuple of tensors (per-parameter grads) — to a 1-D torch tensor."""
    if isinstance(g, torch.Tensor):
        return g.reshape(-1)
    if isinstance(g, np.ndarray):
        return torch.from_numpy(g).reshape(-1)
    if isinstance(g, (list, tuple)):
        if len(g) > 0 and isinstance(g[0], (torch.Tensor, np.ndarray, list, tuple)):
            return torch.cat([flatten_one(p) for p in g])
        return torch.as_tensor(g, dtype=torch.float32)
    if isinstance(g, dict):  # shared-tensor handle dict or param dict
        from byzpy_amd.storage.shared_store import SharedTensorHandle, open_tensor_copy

        if set(g.keys()) >= {"name", "shape", "dtype"}:
            return open_tensor_copy(
                SharedTensorHandle(g["name"], tuple(g["shape"]), g["dtype"])
            ).reshape(-1)
        return torch.cat([flatten_one(v) for v in g.values()])
    # SharedTensorHandle itself
    from byzpy_amd.storage.shared_store import SharedTensorHandle, open_tensor_copy

    if isinstance(g, SharedTensorHandle):
        return open_tensor_copy(g).reshape(-1)
    raise TypeError(f"cannot flatten gradient of type {type(g)!r}")


class LikeTemplate:
    """Remembers the structure of one input gradient so aggregates can be
    restored to it."""

    def __init__(self, g: Any) -> None:
        self.is_matrix_row = False
        self.param_shapes: List[Tuple[int, ...]] = []
        self.param_dtypes: List[torch.dtype] = []
        self.device: torch.device = torch.device("cpu")
        self.dtype: torch.dtype = torch.float32
        self.shape: Tuple[int, ...] = (-1,)
        self.is_numpy = False
        self.is_param_list = False
        if isinstance(g, torch.Tensor):
            self.device, self.dtype, self.shape = g.device, g.dtype, tuple(g.shape)
        elif isinstance(g, np.ndarray):
            self.is_numpy = True
            self.shape = tuple(g.shape)
            self.dtype = torch.from_numpy(g[:0].reshape(0)).dtype if g.size else torch.float32
            self.np_dtype = g.dtype
        elif isinstance(g, (list, tuple)) and len(g) and isinstance(
            g[0], (torch.Tensor, np.ndarray)
        ):
            self.is_param_list = True
            for p in g:
                t = p if isinstance(p, torch.Tensor) else torch.from_numpy(p)
                self.param_shapes.append(tuple(t.shape))
                self.param_dtypes.append(t.dtype)
                self.device = t.device
        else:
            self.shape = (-1,)

    def restore(self, vec: torch.Tensor) -> Any:
        if self.is_param_list:
            out, off = [], 0
            for shp, dt in zip(self.param_shapes, self.param_dtypes):
                numel = int(np.prod(shp)) if shp else 1
                out.append(vec[off : off + numel].reshape(shp).to(dt))
                off += numel
            return out
        if self.is_numpy:
            return vec.detach().cpu().numpy().reshape(self.shape).astype(self.np_dtype)
        return vec.reshape(self.shape).to(device=self.device, dtype=self.dtype)


def stack_gradients(
    gradients: Any, *, device: Any = None
) -> Tuple[torch.Tensor, LikeTemplate]:
    """Build the resident (n, d) matrix from whatever arrived. Accepts a
    2-D tensor (rows are gradients), or a sequence of per-worker gradients
    (tensors / ndarrays / param lists / shared handles)."""
    if isinstance(gradients, torch.Tensor) and gradients.dim() == 2:
        X = gradients
        like = LikeTemplate(gradients[0])
    else:
        rows = [flatten_one(g) for g in gradients]
        like = LikeTemplate(gradients[0])
        X = torch.stack([r.to(rows[0].dtype) for r in rows], dim=0)
    if device is not None:
        X = X.to(device)
    return X, like


def to_like(vec: torch.Tensor, like: LikeTemplate) -> Any:
    return like.restore(vec)
