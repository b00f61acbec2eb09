"""roctx tracing markers.

``trace_range("name")`` emits roctx push/pop pairs that show up in
rocprofv3 ``--marker-trace`` timelines around kernel launches and
collectives. No-ops when libroctx is unavailable (CPU CI).
"""
from __future__ import annotations

import contextlib
import ctypes
from typing import Iterator, Optional

_lib: Optional[ctypes.CDLL] = None
_tried = False


def _roctx() -> Optional[ctypes.CDLL]:
    global _lib, _tried
    if not _tried:
        _tried = True
        # the rocprofiler-sdk roctx FIRST: rocprofv3 --marker-trace only
        # captures SDK markers (the legacy roctracer libroctx64 loads
        # fine but its ranges never reach the rocprofv3 timeline)
        for name in ("librocprofiler-sdk-roctx.so", "libroctx64.so",
                     "libroctx64.so.4"):
            try:
                _lib = ctypes.CDLL(name)
                break
            except OSError:
                continue
    return _lib


def enabled() -> bool:
    return _roctx() is not None


@contextlib.contextmanager
def trace_range(name: str) -> Iterator[None]:
    lib = _roctx()
    if lib is None:
        yield
        return
    lib.roctxRangePushA(name.encode())
    try:
        yield
    finally:
        lib.roctxRangePop()


def mark(name: str) -> None:
    lib = _roctx()
    if lib is not None:
        lib.roctxMarkA(name.encode())
