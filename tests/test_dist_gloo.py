"""Multi-process distributed tests over gloo, world_size=2 (SURVEY.md §4
pattern 6b + the MI355X sharded-aggregation engine). These exercise the
exact collective call patterns the RCCL path uses on GPUs."""
import os

import pytest
import torch
import torch.multiprocessing as tmp

import byzpy_amd.ops.functional as F

WORLD = 2


def _run_workers(target, *args):
    ctx = tmp.get_context("spawn")
    port = 29531 + abs(hash(target.__name__)) % 2000
    procs = []
    for rank in range(WORLD):
        p = ctx.Process(target=_entry, args=(target, rank, port) + args)
        p.start()
        procs.append(p)
    for p in procs:
        p.join(120)
    for p in procs:
        assert p.exitcode == 0, f"worker failed with {p.exitcode}"


def _entry(target, rank, port, *args):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(WORLD)
    os.environ["LOCAL_RANK"] = str(rank)
    import torch.distributed as dist

    from byzpy_amd.parallel import dist as pdist

    pdist.init_from_env()
    try:
        target(rank, *args)
    finally:
        dist.destroy_process_group()


# -- worker bodies (module-level: spawn-picklable) --------------------------


def _body_sharded_ops(rank):
    from byzpy_amd.parallel import sharded
    from byzpy_amd.parallel.dist import column_shard

    g = torch.Generator().manual_seed(7)
    X = torch.randn(10, 64, generator=g)  # identical on both ranks
    Xs = column_shard(X, rank)
    lo, hi = rank * 32, rank * 32 + 32

    med = sharded.median(Xs)
    assert torch.allclose(med, F.median(X)[lo:hi], atol=1e-5)

    # fixed-iteration sharded geomed (poll-free: no shift all-reduce)
    gm = sharded.geometric_median(Xs, fixed_iters=40)
    gm_ref = F.geometric_median(X, tol=1e-30, max_iter=40)
    assert torch.allclose(gm, gm_ref[lo:hi], atol=1e-4)

    mk = sharded.multi_krum(Xs, 2, 3)
    assert torch.allclose(mk, F.multi_krum(X, 2, 3)[lo:hi], atol=1e-4)

    cge = sharded.cge(Xs, 2)
    assert torch.allclose(cge, F.cge(X, 2)[lo:hi], atol=1e-4)

    gm = sharded.geometric_median(Xs, tol=1e-8, max_iter=100)
    ref = F.geometric_median(X, tol=1e-8, max_iter=100)
    assert (gm - ref[lo:hi]).norm() < 1e-3

    cc = sharded.centered_clipping(Xs, c_tau=0.5, M=5)
    ref_cc = F.centered_clipping(X, c_tau=0.5, M=5)
    assert torch.allclose(cc, ref_cc[lo:hi], atol=1e-4)

    cl = sharded.clip_rows(Xs, 1.0)
    assert torch.allclose(cl, F.clip_rows(X, 1.0)[:, lo:hi], atol=1e-5)

    nn_ = sharded.nnm(Xs, 2)
    assert torch.allclose(nn_, F.nnm(X, 2)[:, lo:hi], atol=1e-4)

    tm = sharded.trimmed_mean(Xs, 2)
    assert torch.allclose(tm, F.trimmed_mean(X, 2)[lo:hi], atol=1e-5)

    mm = sharded.mean_of_medians(Xs, 2)
    assert torch.allclose(mm, F.mean_of_medians(X, 2)[lo:hi], atol=1e-5)

    kr = sharded.krum(Xs, 2)
    assert torch.allclose(kr, F.krum(X, 2)[lo:hi], atol=1e-4)

    ar = sharded.arc_clip(Xs, 2)
    assert torch.allclose(ar, F.arc_clip(X, 2)[:, lo:hi], atol=1e-4)

    bk = sharded.bucketing(Xs, 3, list(range(10)))
    assert torch.allclose(
        bk, F.bucketing(X, 3, perm=list(range(10)))[:, lo:hi], atol=1e-5
    )


def _body_rccl_ps(rank):
    from byzpy_amd.engine.parameter_server.rccl import (
        RcclParameterServer,
        trimmed_mean_aggregate,
    )

    d = 37  # odd on purpose: exercises the pad path

    def honest(v):
        return lambda: torch.full((d,), float(v))

    def byz():
        return lambda: torch.full((d,), 1.0e6)

    fns = [honest(1 + rank), byz()] if rank == 1 else [honest(1 + rank), honest(5)]
    ps = RcclParameterServer(fns, trimmed_mean_aggregate(1), gather_result=True)
    out = ps.round()
    assert out.shape == (d,)
    # n=4 grads: {1, 5, 2, 1e6}; trimmed f=1 -> mean(2, 5) = 3.5
    assert torch.allclose(out, torch.full((d,), 3.5), atol=1e-4), out[:3]
    # chunked comm/compute overlap pipeline gives the same result
    ps2 = RcclParameterServer(
        fns, trimmed_mean_aggregate(1), gather_result=True, overlap_chunks=3
    )
    out2 = ps2.round()
    assert torch.allclose(out2, out, atol=1e-5)


def _body_rccl_p2p(rank):
    from byzpy_amd.aggregators import CoordinateWiseMedian
    from byzpy_amd.engine.peer_to_peer.rccl import RcclPeerToPeer

    state = {"params": torch.full((5,), float(rank + 1))}

    def half_step():
        return state["params"] * 2

    def write(params):
        state["params"] = params

    p2p = RcclPeerToPeer(half_step, write, CoordinateWiseMedian())
    out = p2p.round()
    # vectors: rank0 -> 2, rank1 -> 4; median = 3
    assert torch.allclose(out, torch.full((5,), 3.0), atol=1e-5)
    assert torch.allclose(state["params"], torch.full((5,), 3.0), atol=1e-5)

    # with a pre-aggregator (clip to norm 1): both thetas get clipped first
    from byzpy_amd.pre_aggregators import Clipping

    state["params"] = torch.full((5,), float(rank + 1))
    p2p2 = RcclPeerToPeer(
        half_step, write, CoordinateWiseMedian(), pre_aggregator=Clipping(1.0)
    )
    out2 = p2p2.round()
    assert float(out2.norm()) <= 1.0 + 1e-4

    # byzantine rank 1 broadcasts an empire attack; rank 0 (honest, median
    # over [self, byz]) must stay bounded by its own vector's scale
    from byzpy_amd.attacks import EmpireAttack

    state["params"] = torch.full((5,), float(rank + 1))
    p2p3 = RcclPeerToPeer(
        half_step,
        write,
        CoordinateWiseMedian(),
        attack=EmpireAttack(scale=-100.0) if rank == 1 else None,
    )
    out3 = p2p3.round()
    assert torch.isfinite(out3).all()
    if rank == 0:
        # median of {own 2, attack -200} with n=2 = mean = -99: median of two
        # IS affected; just verify the attack vector actually propagated
        assert float(out3.min()) < -50.0


def _body_collectives(rank):
    from byzpy_amd.parallel import dist as pdist

    t = torch.full((4,), float(rank + 1))
    s = pdist.all_reduce_(t.clone())
    assert torch.allclose(s, torch.full((4,), 3.0))
    g = pdist.all_gather_rows(t.reshape(1, -1))
    assert g.shape == (2, 4)
    a2a = pdist.all_to_all_rows(torch.arange(4.0).reshape(2, 2) + 10 * rank)
    assert a2a.shape == (2, 2)
    # reduce-scatter: rank r gets sum over ranks of row-block r
    full = torch.arange(8.0).reshape(4, 2) * (rank + 1)
    rs = pdist.reduce_scatter_cat(full)
    lo = rank * 2
    expect = (torch.arange(8.0).reshape(4, 2) * 3)[lo : lo + 2]
    assert torch.allclose(rs, expect), rs


def test_collectives():
    _run_workers(_body_collectives)


def test_sharded_aggregation_parity():
    _run_workers(_body_sharded_ops)


def test_rccl_parameter_server():
    _run_workers(_body_rccl_ps)


def test_rccl_peer_to_peer():
    _run_workers(_body_rccl_p2p)


# -- ring(k) neighbor send/recv semantics (world_size=4) --------------------


def _run_workers_n(target, world, *args):
    ctx = tmp.get_context("spawn")
    port = 29531 + abs(hash(target.__name__ + "n")) % 2000
    procs = []
    for rank in range(world):
        p = ctx.Process(target=_entry_n, args=(target, rank, port, world) + args)
        p.start()
        procs.append(p)
    for p in procs:
        p.join(120)
    for p in procs:
        assert p.exitcode == 0, f"worker failed with {p.exitcode}"


def _entry_n(target, rank, port, world, *args):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["LOCAL_RANK"] = str(rank)
    import torch.distributed as dist

    from byzpy_amd.parallel import dist as pdist

    pdist.init_from_env()
    try:
        target(rank, *args)
    finally:
        dist.destroy_process_group()


def _body_ring_p2p(rank):
    """ring byzantine rank's Empire output must equal
    -mean(received neighbor theta-halves), and ring traffic must ride
    neighbor send/recv pairs (p2p transport), not a full all-gather."""
    from byzpy_amd.aggregators import CoordinateWiseMedian
    from byzpy_amd.attacks import EmpireAttack
    from byzpy_amd.engine.peer_to_peer.rccl import RcclPeerToPeer
    from byzpy_amd.engine.peer_to_peer.topology import Topology

    d = 6
    state = {"params": torch.full((d,), float(rank + 1))}

    def half_step():
        return state["params"].clone()

    def write(params):
        state["params"] = params

    p2p = RcclPeerToPeer(
        half_step,
        write,
        CoordinateWiseMedian(),
        topology=Topology.ring(4, 1),
        attack=EmpireAttack(scale=-1.0) if rank == 3 else None,
    )
    assert p2p.transport == "p2p", p2p.transport  # ring(4,1) is sparse
    assert p2p.byzantine_ranks == frozenset({3})
    out = p2p.round()
    if rank == 3:
        # empire over the RECEIVED honest in-neighbor vectors {1.0, 3.0}
        assert torch.allclose(out, torch.full((d,), -2.0), atol=1e-6), out
        # byzantine rank does not write params
        assert torch.allclose(state["params"], torch.full((d,), 4.0))
    else:
        expect = {0: 1.0, 1: 2.0, 2: 2.0}[rank]
        assert torch.allclose(out, torch.full((d,), expect), atol=1e-6), (
            rank,
            out,
        )
        assert torch.allclose(state["params"], out)


def _body_ring_honest_only(rank):
    """Pure-honest ring: p2p phase-1 exchange only, no byz edges."""
    from byzpy_amd.aggregators import CoordinateWiseTrimmedMean
    from byzpy_amd.engine.peer_to_peer.rccl import RcclPeerToPeer
    from byzpy_amd.engine.peer_to_peer.topology import Topology

    d = 5
    theta = torch.full((d,), float(rank * 10))
    got = {}

    p2p = RcclPeerToPeer(
        lambda: theta,
        lambda p: got.__setitem__("p", p),
        CoordinateWiseTrimmedMean(1),
        topology=Topology.ring(4, 1),
    )
    out = p2p.round()
    nbrs = sorted({(rank - 1) % 4, (rank + 1) % 4})
    vals = sorted([rank * 10.0] + [j * 10.0 for j in nbrs])
    assert torch.allclose(out, torch.full((d,), vals[1]), atol=1e-6)


def _body_config5_path(rank, tmpdir):
    """Config-5 engine path at gloo ws=2: d-sharded bucketing + centered
    clipping and the per-rank sharded checkpoint round-trip (VERDICT r01
    item 7)."""
    from byzpy_amd.parallel import sharded
    from byzpy_amd.parallel.dist import column_shard
    from byzpy_amd.utils.checkpoint import load_checkpoint, save_checkpoint

    g = torch.Generator().manual_seed(11)
    n, d = 8, 96
    X = torch.randn(n, d, generator=g)  # identical on both ranks
    Xs = column_shard(X, rank)
    perm = list(range(n))
    B = sharded.bucketing(Xs, 2, perm)
    out = sharded.centered_clipping(B, c_tau=1.0, M=5)

    # reference: full-matrix pipeline, then slice this rank's shard
    import byzpy_amd.ops.functional as FF

    ref = FF.centered_clipping(FF.bucketing(X, 2, perm), c_tau=1.0, M=5)
    lo = rank * 48
    assert torch.allclose(out, ref[lo : lo + 48], atol=1e-5)

    save_checkpoint(
        tmpdir,
        round_idx=3,
        model_state={"aggregate_shard": out},
        rank=rank,
        world_size=2,
        extra_meta={"d_local": 48},
    )
    import torch.distributed as dist

    dist.barrier()
    back = load_checkpoint(tmpdir, rank=rank)
    assert back["meta"]["round"] == 3
    assert back["meta"]["world_size"] == 2
    assert torch.equal(back["model_state"]["aggregate_shard"], out)


def test_config5_sharded_path_and_checkpoint(tmp_path):
    _run_workers_n(_body_config5_path, 2, str(tmp_path))


def _body_bench_path_ws8(rank):
    """The exact bench step (sharded median_and_multi_krum) at world
    size 8 — the driver's largest SCALE point — against the full-matrix
    single-process reference."""
    from byzpy_amd.hip import dispatch as D
    from byzpy_amd.parallel import sharded
    from byzpy_amd.parallel.dist import column_shard

    g = torch.Generator().manual_seed(42)
    n, d = 16, 320
    X = torch.randn(n, d, generator=g)  # identical on all ranks
    Xs = column_shard(X, rank)
    med, krum = sharded.median_and_multi_krum(Xs, 3, 4)
    lo = rank * 40
    ref_med, ref_G = D.median_and_gram(X)
    assert torch.allclose(med, ref_med[lo : lo + 40], atol=1e-5)
    ref_krum = D.multi_krum(X, 3, 4)
    assert torch.allclose(krum, ref_krum[lo : lo + 40], atol=1e-4)


def test_bench_path_world8():
    _run_workers_n(_body_bench_path_ws8, 8)


def test_ring_p2p_byzantine_context():
    _run_workers_n(_body_ring_p2p, 4)


def test_ring_p2p_honest():
    _run_workers_n(_body_ring_honest_only, 4)


def _body_rccl_ps_census(rank):
    """50 rounds through the RCCL PS: tensor census must be flat (the
    collective path must not retain per-round buffers)."""
    import gc

    from byzpy_amd.engine.parameter_server.rccl import (
        RcclParameterServer,
        trimmed_mean_aggregate,
    )

    d = 256
    fns = [lambda: torch.randn(d, generator=torch.Generator().manual_seed(1)),
           lambda: torch.full((d,), float(rank))]
    ps = RcclParameterServer(fns, trimmed_mean_aggregate(1), gather_result=True)
    for _ in range(5):
        ps.round()
    gc.collect()
    c0 = sum(1 for o in gc.get_objects() if torch.is_tensor(o))
    for _ in range(45):
        ps.round()
    gc.collect()
    c1 = sum(1 for o in gc.get_objects() if torch.is_tensor(o))
    assert c1 <= c0 + 4, (c0, c1)


def test_rccl_ps_no_tensor_retention():
    _run_workers(_body_rccl_ps_census)
