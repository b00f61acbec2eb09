"""Adaptive chunk sizing for CPU-pool subtask fan-out.

Reference parity: aggregators/_chunking.py:30-72 (targets >= 4 chunks per
worker, shrinks the requested chunk at most 8x, env-overridable). On the
GPU path chunking disappears — ops are single kernel launches.
"""
from __future__ import annotations

import os


def select_adaptive_chunk_size(total: int, workers: int, requested: int) -> int:
    min_per_worker = int(os.environ.get("BYZPY_AMD_CHUNK_MIN_PER_WORKER", "4"))
    max_shrink = int(os.environ.get("BYZPY_AMD_CHUNK_MAX_SHRINK", "8"))
    target_factor = int(os.environ.get("BYZPY_AMD_CHUNK_TARGET_FACTOR", "1"))
    requested = max(1, int(requested))
    workers = max(1, int(workers))
    target_chunks = max(1, workers * min_per_worker * target_factor)
    if total <= 0:
        return requested
    chunk = requested
    # shrink (at most max_shrink x) until we have enough chunks to go around
    while (total + chunk - 1) // chunk < target_chunks and chunk > 1:
        if requested / chunk >= max_shrink:
            break
        chunk = max(1, chunk // 2)
    return chunk


def chunk_ranges(total: int, chunk: int):
    lo = 0
    while lo < total:
        hi = min(total, lo + chunk)
        yield lo, hi
        lo = hi
