"""RCCL-over-xGMI distributed fabric (torch.distributed).

MI355X design: one process per GPU; backend "nccl" IS
RCCL on ROCm; gloo on CPU boxes so the same code paths run in CI. All
gradient movement is collectives — broadcast / all-reduce / all-gather /
reduce-scatter / all-to-all — sized for 7 p2p xGMI links per GPU.
"""
from __future__ import annotations

import datetime
import os
from typing import Optional

import torch
import torch.distributed as dist


def is_initialized() -> bool:
    return dist.is_available() and dist.is_initialized()


def get_rank() -> int:
    return dist.get_rank() if is_initialized() else 0


def get_world_size() -> int:
    return dist.get_world_size() if is_initialized() else 1


def init_from_env(timeout_s: float = 600.0) -> int:
    """Initialize from torchrun-style env vars; returns local rank. Uses
    the composite backend (gloo for CPU tensors, RCCL for CUDA tensors)
    when a GPU is visible, plain gloo otherwise — a GPU-visible host can
    still all-reduce CPU tensors (a bare "nccl" group cannot). Override
    with BYZPY_DIST_BACKEND. Safe to call twice."""
    if is_initialized():
        return int(os.environ.get("LOCAL_RANK", 0))
    if "RANK" not in os.environ:
        return 0  # single-process mode: no process group
    backend = os.environ.get(
        "BYZPY_DIST_BACKEND",
        "cpu:gloo,cuda:nccl" if torch.cuda.is_available() else "gloo",
    )
    local_rank = int(os.environ.get("LOCAL_RANK", os.environ.get("RANK", 0)))
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank % torch.cuda.device_count())
    dist.init_process_group(
        backend=backend, timeout=datetime.timedelta(seconds=timeout_s)
    )
    return local_rank


def barrier() -> None:
    if is_initialized():
        dist.barrier()


def all_reduce_(t: torch.Tensor) -> torch.Tensor:
    if is_initialized():
        dist.all_reduce(t)
    return t


def broadcast_(t: torch.Tensor, src: int = 0) -> torch.Tensor:
    if is_initialized():
        dist.broadcast(t, src)
    return t


def all_gather_rows(local: torch.Tensor) -> torch.Tensor:
    """All-gather equal-sized row blocks into one stacked matrix (PS gather
    C1: workers -> all)."""
    if not is_initialized():
        return local
    world = get_world_size()
    out = torch.empty(
        (local.shape[0] * world,) + tuple(local.shape[1:]),
        dtype=local.dtype,
        device=local.device,
    )
    dist.all_gather_into_tensor(out, local.contiguous())
    return out


def reduce_scatter_cat(full: torch.Tensor) -> torch.Tensor:
    """Reduce-scatter a replicated-layout (world*m, ...) tensor: rank r gets
    the SUM over ranks of row-block r (SURVEY.md C4 — coordinate-chunked
    partial reduction; the xGMI-friendly half of an all-reduce)."""
    if not is_initialized():
        return full
    world = get_world_size()
    assert full.shape[0] % world == 0
    out = torch.empty(
        (full.shape[0] // world,) + tuple(full.shape[1:]),
        dtype=full.dtype,
        device=full.device,
    )
    dist.reduce_scatter_tensor(out, full.contiguous())
    return out


def all_to_all_rows(local: torch.Tensor) -> torch.Tensor:
    """Row-blocked all-to-all: rank r sends row-block j to rank j and
    receives block r from everyone (n-sharded -> d-sharded transposition,
    SURVEY.md C3/C4)."""
    if not is_initialized():
        return local
    world = get_world_size()
    assert local.shape[0] % world == 0
    out = torch.empty_like(local)
    dist.all_to_all_single(out, local.contiguous())
    return out


def neighbor_exchange(
    send_to, recv_from, vec: torch.Tensor
) -> "dict[int, torch.Tensor]":
    """Batched point-to-point neighbor exchange (ring(k)
    neighbor send/recv pairs instead of a full all-gather — traffic
    proportional to the topology degree, not world size). Sends ``vec`` to
    each rank in ``send_to`` and receives one same-shaped tensor from each
    rank in ``recv_from``; returns {src_rank: tensor}.

    Uses dist.batch_isend_irecv (ncclGroupStart/End under RCCL) so the
    p2p pairs ride the xGMI links concurrently and the issue order cannot
    deadlock. Every rank in the group must post the schedule its peers
    expect (callers derive it from the shared Topology + byzantine set).
    """
    bufs = {int(j): torch.empty_like(vec) for j in recv_from}
    if not is_initialized():
        return bufs
    v = vec.contiguous()
    ops = [dist.P2POp(dist.isend, v, int(j)) for j in send_to]
    ops += [dist.P2POp(dist.irecv, bufs[int(j)], int(j)) for j in recv_from]
    if ops:
        for req in dist.batch_isend_irecv(ops):
            req.wait()
    return bufs


def all_gather_obj(obj) -> list:
    """All-gather a small picklable object (control plane, e.g. the
    byzantine-rank set at engine bring-up)."""
    if not is_initialized():
        return [obj]
    out: list = [None] * get_world_size()
    dist.all_gather_object(out, obj)
    return out


def column_shard(X: torch.Tensor, rank: Optional[int] = None) -> torch.Tensor:
    """This rank's contiguous d-shard of an (n, d) matrix."""
    world = get_world_size()
    if world == 1:
        return X
    r = get_rank() if rank is None else rank
    d = X.shape[1]
    per = (d + world - 1) // world
    return X[:, r * per : min(d, (r + 1) * per)].contiguous()
