"""Cross-process tensor payload wrapping.

Reference parity: engine/actor/ipc.py:20-56 — tensors become shared-memory
handles tagged for transparent unwrap on the receiving side; the segment is
unlinked on first read. CUDA tensors are NEVER wrapped here: device data
stays in HBM (in-process zero copy) or moves over RCCL.
"""
from __future__ import annotations

from typing import Any

import numpy as np
import torch

from byzpy_amd.storage.shared_store import (
    SharedTensorHandle,
    cleanup_tensor,
    open_tensor,
    register_tensor,
)

_TAG = "__BYZAMD_SHARED_TENSOR__"


def wrap_payload(payload: Any) -> Any:
    if isinstance(payload, torch.Tensor):
        if payload.is_cuda:
            raise TypeError(
                "refusing to wrap a CUDA tensor for host shm transport; device "
                "tensors cross process boundaries over RCCL, not POSIX shm"
            )
        return (_TAG, register_tensor(payload))
    if isinstance(payload, np.ndarray):
        return (_TAG, register_tensor(payload))
    if isinstance(payload, list):
        return [wrap_payload(p) for p in payload]
    if isinstance(payload, tuple):
        return tuple(wrap_payload(p) for p in payload)
    if isinstance(payload, dict):
        return {k: wrap_payload(v) for k, v in payload.items()}
    return payload


def unwrap_payload(payload: Any) -> Any:
    if (
        isinstance(payload, tuple)
        and len(payload) == 2
        and payload[0] == _TAG
        and isinstance(payload[1], SharedTensorHandle)
    ):
        handle = payload[1]
        with open_tensor(handle) as view:
            out = view.clone()
        cleanup_tensor(handle)  # unlink on first read (reference ipc.py:50-56)
        return out
    if isinstance(payload, list):
        return [unwrap_payload(p) for p in payload]
    if isinstance(payload, tuple):
        return tuple(unwrap_payload(p) for p in payload)
    if isinstance(payload, dict):
        return {k: unwrap_payload(v) for k, v in payload.items()}
    return payload
