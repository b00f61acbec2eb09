"""POSIX shared-memory tensor store for CPU process pools.

Reference parity: engine/storage/shared_store.py:11-54 and
engine/actor/ipc.py. MI355X note: this exists ONLY for the CPU
process-pool path; on-device gradients never leave HBM — workers in one
process share device pointers, and cross-process GPU movement goes through
RCCL (byzpy_amd/parallel/), not host shm.
"""
from __future__ import annotations

import contextlib
import uuid
from dataclasses import dataclass
from multiprocessing import shared_memory
from typing import Any, Iterator, Tuple

import numpy as np
import torch

_TORCH_TO_NP = {
    torch.float32: np.float32,
    torch.float64: np.float64,
    torch.float16: np.float16,
    torch.int64: np.int64,
    torch.int32: np.int32,
    torch.uint8: np.uint8,
    torch.bool: np.bool_,
}


@dataclass(frozen=True)
class SharedTensorHandle:
    name: str
    shape: Tuple[int, ...]
    dtype: str  # numpy dtype string


def register_tensor(t: Any) -> SharedTensorHandle:
    """Copy a CPU tensor/ndarray into a fresh shm segment and return its
    handle. bf16 is staged as its raw uint16 bits with dtype tag 'bfloat16'."""
    if isinstance(t, torch.Tensor):
        t = t.detach().cpu()
        if t.dtype == torch.bfloat16:
            arr = t.view(torch.uint16).numpy()
            dtype_tag = "bfloat16"
        else:
            arr = t.numpy()
            dtype_tag = str(arr.dtype)
    else:
        arr = np.asarray(t)
        dtype_tag = str(arr.dtype)
    name = f"byzamd_{uuid.uuid4().hex[:16]}"
    seg = shared_memory.SharedMemory(create=True, size=max(1, arr.nbytes), name=name)
    try:
        dst = np.ndarray(arr.shape, dtype=arr.dtype, buffer=seg.buf)
        dst[...] = arr
    finally:
        seg.close()
    return SharedTensorHandle(name=name, shape=tuple(arr.shape), dtype=dtype_tag)


@contextlib.contextmanager
def open_tensor(handle: SharedTensorHandle) -> Iterator[torch.Tensor]:
    """Map the segment and yield a zero-copy torch view (valid inside the
    context only)."""
    np_dtype = np.uint16 if handle.dtype == "bfloat16" else np.dtype(handle.dtype)
    seg = shared_memory.SharedMemory(name=handle.name)
    try:
        arr = np.ndarray(handle.shape, dtype=np_dtype, buffer=seg.buf)
        t = torch.from_numpy(arr)
        if handle.dtype == "bfloat16":
            t = t.view(torch.bfloat16)
        yield t
    finally:
        seg.close()


def open_tensor_copy(handle: SharedTensorHandle) -> torch.Tensor:
    with open_tensor(handle) as view:
        return view.clone()


def write_handle(handle: SharedTensorHandle, t: Any) -> None:
    """Overwrite an existing segment's contents in place (iterative
    aggregators re-broadcast their center each iteration — reference
    geometric_median.py:204-206)."""
    if isinstance(t, torch.Tensor):
        t = t.detach().cpu()
        arr = t.view(torch.uint16).numpy() if t.dtype == torch.bfloat16 else t.numpy()
    else:
        arr = np.asarray(t)
    seg = shared_memory.SharedMemory(name=handle.name)
    try:
        dst = np.ndarray(arr.shape, dtype=arr.dtype, buffer=seg.buf)
        dst[...] = arr
    finally:
        seg.close()


def cleanup_tensor(handle: SharedTensorHandle) -> None:
    with contextlib.suppress(FileNotFoundError):
        seg = shared_memory.SharedMemory(name=handle.name)
        seg.close()
        seg.unlink()


def resolve_matrix(ref: Any) -> Tuple[torch.Tensor, Any]:
    """Resolve a subtask matrix reference to (tensor, closer). ``ref`` is
    either a tensor (thread pool: zero-copy) or a SharedTensorHandle
    (process pool: shm view)."""
    if isinstance(ref, torch.Tensor):
        return ref, None
    if isinstance(ref, SharedTensorHandle):
        np_dtype = np.uint16 if ref.dtype == "bfloat16" else np.dtype(ref.dtype)
        seg = shared_memory.SharedMemory(name=ref.name)
        arr = np.ndarray(ref.shape, dtype=np_dtype, buffer=seg.buf)
        t = torch.from_numpy(arr)
        if ref.dtype == "bfloat16":
            t = t.view(torch.bfloat16)
        return t, seg
    raise TypeError(f"cannot resolve matrix ref of type {type(ref)!r}")


def close_ref(closer: Any) -> None:
    if closer is not None:
        closer.close()
