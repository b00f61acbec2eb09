from byzpy_amd.storage.shared_store import (
    SharedTensorHandle,
    cleanup_tensor,
    open_tensor,
    open_tensor_copy,
    register_tensor,
    resolve_matrix,
)

__all__ = [
    "SharedTensorHandle",
    "register_tensor",
    "open_tensor",
    "open_tensor_copy",
    "cleanup_tensor",
    "resolve_matrix",
]
