"""RCCL parameter-server engine — one process per GPU over xGMI.

The MI355X-native replacement for the reference's actor-RPC gather
(SURVEY.md C1/C2): every rank computes its worker gradients locally, the
n-sharded gradient matrix is transposed to d-sharded with ONE
all-to-all (each GPU then owns a contiguous d/world slice of every worker
gradient), robust aggregation runs shard-local with only tiny (n,)/(n,n)
all-reduces (byzpy_amd/parallel/sharded.py), and the d-sharded update is
either consumed in place or assembled with an all-gather. xGMI note: the
all-to-all form spreads traffic over all 7 p2p links instead of
serializing a ring.

BASELINE config 3 shape: ResNet-50 grads, 8 honest + 3 SignFlip workers,
TrimmedMean, 8 GPUs.
"""
from __future__ import annotations

from typing import Callable, List

import torch

from byzpy_amd.parallel import dist as pdist
from byzpy_amd.parallel import sharded


class RcclParameterServer:
    """Symmetric (serverless) PS: every rank ends each round with the same
    d-shard of the robust aggregate (or the full vector with gather=True).

    ``local_gradient_fns``: this rank's worker-gradient callables (honest
    or byzantine), each returning a flat (d,) tensor on this rank's GPU.
    The global worker count n = sum over ranks of len(local_gradient_fns)
    and must be equal per rank (pad with byzantine/duplicate workers if
    needed).
    """

    def __init__(
        self,
        local_gradient_fns: List[Callable[[], torch.Tensor]],
        aggregate_fn: Callable[[torch.Tensor], torch.Tensor],
        *,
        gather_result: bool = False,
        overlap_chunks: int = 1,
    ) -> None:
        self.local_gradient_fns = list(local_gradient_fns)
        self.aggregate_fn = aggregate_fn
        self.gather_result = gather_result
        # >1: split the shard into chunks and pipeline — chunk i aggregates
        # on the compute stream while chunk i+1's all-to-all is in flight on
        # RCCL's stream (coordinate-chunkable aggregators only: every
        # coordinate-wise op qualifies; Krum-style global scoring does not).
        self.overlap_chunks = max(1, int(overlap_chunks))
        if self.overlap_chunks > 1 and getattr(
            aggregate_fn, "coordinate_chunkable", True
        ) is False:
            raise ValueError(
                "aggregate_fn is not coordinate-chunkable; overlap_chunks "
                "must be 1 (chunked pipelining would aggregate each chunk "
                "independently)"
            )

    def round(self) -> torch.Tensor:
        world = pdist.get_world_size()
        grads = [fn().reshape(-1) for fn in self.local_gradient_fns]
        local = torch.stack(grads)  # (n_local, d)
        n_local, d = local.shape
        if world == 1:
            out = self.aggregate_fn(local)
            return out
        # pad d to a multiple of world so shards are equal-sized
        pad = (-d) % world
        if pad:
            local = torch.nn.functional.pad(local, (0, pad))
        d_pad = d + pad
        shard = d_pad // world
        # (n_local, d) -> world blocks of (n_local, shard) -> all-to-all ->
        # this rank holds every worker's shard: (n_global, shard)
        blocks = local.reshape(n_local, world, shard).transpose(0, 1).contiguous()
        if self.overlap_chunks <= 1:
            recv = pdist.all_to_all_rows(blocks.reshape(world * n_local, shard))
            X_shard = recv.reshape(world, n_local, shard).reshape(
                world * n_local, shard
            )
            out_shard = self.aggregate_fn(X_shard)  # (shard,)
        else:
            out_shard = self._round_overlapped(blocks, world, n_local, shard)
        if not self.gather_result:
            return out_shard
        full = pdist.all_gather_rows(out_shard.reshape(1, -1)).reshape(-1)
        return full[:d]

    def _round_overlapped(
        self, blocks: torch.Tensor, world: int, n_local: int, shard: int
    ) -> torch.Tensor:
        """Chunked pipeline: issue chunk i+1's all-to-all (async, on RCCL's
        own stream) while chunk i aggregates on the compute stream
        (SURVEY.md §7 build step 4: side-stream overlap of C1 with the
        aggregation kernels)."""
        import torch.distributed as dist

        C = min(self.overlap_chunks, shard) or 1
        bounds = [(shard * c) // C for c in range(C + 1)]
        recvs, works = [], []

        def issue(c: int) -> None:
            lo, hi = bounds[c], bounds[c + 1]
            send = blocks[:, :, lo:hi].reshape(world * n_local, hi - lo).contiguous()
            recv = torch.empty_like(send)
            works.append(dist.all_to_all_single(recv, send, async_op=True))
            recvs.append(recv)

        issue(0)
        outs = []
        for c in range(C):
            if c + 1 < C:
                issue(c + 1)
            works[c].wait()  # compute stream waits on the RCCL stream event
            lo, hi = bounds[c], bounds[c + 1]
            X_c = recvs[c].reshape(world, n_local, hi - lo).reshape(
                world * n_local, hi - lo
            )
            outs.append(self.aggregate_fn(X_c))
        return torch.cat(outs)


def trimmed_mean_aggregate(f: int):
    def _agg(X_shard: torch.Tensor) -> torch.Tensor:
        return sharded.trimmed_mean(X_shard, f)

    return _agg


def median_aggregate():
    def _agg(X_shard: torch.Tensor) -> torch.Tensor:
        return sharded.median(X_shard)

    return _agg


def multi_krum_aggregate(f: int, q: int):
    """NOT coordinate-chunkable: Krum scores are a global function of the
    full shard (Gram all-reduce), so use overlap_chunks=1 with this one —
    chunked pipelining would score each chunk independently."""

    def _agg(X_shard: torch.Tensor) -> torch.Tensor:
        return sharded.multi_krum(X_shard, f, q)

    _agg.coordinate_chunkable = False
    return _agg
