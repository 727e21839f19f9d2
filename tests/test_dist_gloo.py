"""Multi-process coverage of the distributed path on CPU (gloo backend,
world_size=2): the same comm code (PamComm over torch.distributed) and the
same data-movement orchestration (to_dist / asarray / ghost cells / the
reshaped rebalance) that the RCCL path uses on the GPU box, checked against
the oracle's rank-simulated semantics."""
import os
import sys

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
WORLD = 2


def _worker(rank: int, port: int, fn_name: str):
    sys.path.insert(0, ROOT)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(WORLD)
    import torch.distributed as dist
    from pylops_mpi_amd import comm as pam_comm
    c = pam_comm.init_default_comm(device=torch.device("cpu"))
    try:
        globals()[fn_name](c)
    finally:
        dist.barrier()
        dist.destroy_process_group()


def _spawn(fn_name: str):
    port = 29500 + (hash(fn_name) % 1000)
    mp.spawn(_worker, args=(port, fn_name), nprocs=WORLD, join=True)


# ------------------------------------------------------------- worker bodies
def body_allreduce(c):
    t = torch.tensor([float(c.rank + 1)])
    c.allreduce_(t, "sum")
    assert t.item() == 3.0
    t = torch.tensor([float(c.rank + 1)])
    c.allreduce_(t, "max")
    assert t.item() == 2.0


def body_allgather_obj(c):
    out = c.allgather_obj(("shape", c.rank))
    assert out == [("shape", 0), ("shape", 1)]


def body_allgather_tensors(c):
    shapes = [(3,), (2,)]
    t = torch.arange(shapes[c.rank][0], dtype=torch.float64) + 10 * c.rank
    got = c.allgather_tensors(t, shapes)
    assert torch.equal(got[0], torch.arange(3, dtype=torch.float64))
    assert torch.equal(got[1], torch.arange(2, dtype=torch.float64) + 10)


def body_to_dist_asarray(c):
    import pylops_mpi_amd as pm
    import oracle
    x = torch.arange(26, dtype=torch.float64).reshape(13, 2)
    d = pm.DistributedArray.to_dist(x, base_comm=c)
    sim = oracle.to_dist(x.numpy(), WORLD)
    assert np.array_equal(d.local_array.numpy(), sim.locals[c.rank])
    full = d.asarray()
    assert torch.equal(full, x)


def body_ghost_cells(c):
    import pylops_mpi_amd as pm
    import oracle
    x = torch.arange(24, dtype=torch.float64).reshape(12, 2)
    d = pm.DistributedArray.to_dist(x, base_comm=c)
    sim = oracle.to_dist(x.numpy(), WORLD)
    for cf, cb in ((1, 1), (2, None), (None, 2), (2, 2)):
        g = d.add_ghost_cells(cells_front=cf, cells_back=cb)
        sg = oracle.add_ghost_cells(
            sim.locals,
            None if cf is None else [cf] * WORLD,
            None if cb is None else [cb] * WORLD)
        assert np.array_equal(g.numpy(), sg[c.rank]), (cf, cb)
    # halo_exchange carries the same values without concatenation
    gf, gb = d.halo_exchange(2)
    if c.rank == 0:
        assert gf is None
        assert np.array_equal(gb.numpy(), sim.locals[1][:2])
    else:
        assert gb is None
        assert np.array_equal(gf.numpy(), sim.locals[0][-2:])


def body_rebalance(c):
    """The reshaped rebalance arithmetic (ref decorators.py:66-77) against
    the oracle's restatement, on an element-balanced 1-D input that does
    NOT align with the plane split."""
    import pylops_mpi_amd as pm
    from oracle.ranksim import reshaped_apply, to_dist as sim_to_dist
    dims = (13, 3)  # 13 rows over 2 ranks -> planes (7,3),(6,3) = 21/18
    n = int(np.prod(dims))
    rng = np.random.default_rng(42)
    xg = rng.standard_normal(n)
    op = pm.MPIFirstDerivative(dims, base_comm=c)
    x = pm.DistributedArray.to_dist(torch.as_tensor(xg), base_comm=c)
    shapes, counts = op._plane_counts()
    flat = op._rebalance(x, counts)
    # oracle: capture what the reshaped wrapper hands to the body
    captured = {}

    def body(arr_locals):
        captured["arr"] = [a.copy() for a in arr_locals]
        return arr_locals

    reshaped_apply(body, dims, sim_to_dist(xg, WORLD))
    want = captured["arr"][c.rank].reshape(-1)
    assert np.allclose(flat.numpy(), want, rtol=0, atol=0)


def body_sendrecv(c):
    t = torch.full((4,), float(c.rank), dtype=torch.float64)
    r = torch.empty(4, dtype=torch.float64)
    other = 1 - c.rank
    c.sendrecv(t, other, r, other)
    assert torch.all(r == float(other))


def body_blockdiag(c):
    """MPIBlockDiag distribution logic (stacking rebalance + per-op
    slicing + output split) vs the oracle, with uneven per-rank blocks.
    Local compute is a CallableLocal torch matmul (CPU test adapter —
    the product DenseLocal path is GPU-only and covered in the gpu
    suite)."""
    import numpy as np
    import pylops_mpi_amd as pm
    from oracle import SimBlockDiag, to_dist as sim_to_dist
    rng = np.random.default_rng(50)
    mats_all = [[rng.standard_normal((4, 6)), rng.standard_normal((3, 2))],
                [rng.standard_normal((5, 5))]]
    mine = mats_all[c.rank]
    ops = []
    for A in mine:
        At = torch.as_tensor(A)
        ops.append(pm.CallableLocal(
            (A.shape[0], A.shape[1]),
            lambda v, At=At: At @ v,
            lambda v, At=At: At.T @ v))
    op = pm.MPIBlockDiag(ops, base_comm=c)
    sop = SimBlockDiag(mats_all)
    assert op.shape == sop.shape
    n, m = op.shape
    xg = rng.standard_normal(m)
    yg = rng.standard_normal(n)
    x = pm.DistributedArray.to_dist(torch.as_tensor(xg), base_comm=c)
    y = pm.DistributedArray.to_dist(torch.as_tensor(yg), base_comm=c)
    got_mv = op.matvec(x).asarray().numpy()
    got_rmv = op.rmatvec(y).asarray().numpy()
    np.testing.assert_allclose(got_mv, sop.matvec(sim_to_dist(xg, 2)).asarray(),
                               rtol=1e-13)
    np.testing.assert_allclose(got_rmv, sop.rmatvec(sim_to_dist(yg, 2)).asarray(),
                               rtol=1e-13)


# ------------------------------------------------------------------- drivers
@pytest.mark.parametrize("body", [
    "body_allreduce", "body_allgather_obj", "body_allgather_tensors",
    "body_to_dist_asarray", "body_ghost_cells", "body_rebalance",
    "body_sendrecv", "body_blockdiag", "body_fredholm", "body_vstack",
    "body_post_neighbors_overlap", "body_mask_subcomm", "body_nonstatconv",
    "body_proximal_call_reduction", "body_redistribute", "body_fftnd",
    "body_norm_axis", "body_broadcast_setitem",
    "body_halo_guard_deterministic", "body_subcomm_split",
])
def test_gloo_world2(body):
    _spawn(body)


def body_fredholm(c):
    """MPIFredholm1 slice/allgather logic across 2 ranks vs the oracle
    (torch-matmul test adapter for the GPU-only batched kernel)."""
    import numpy as np
    import pylops_mpi_amd as pm
    from oracle.fredholm import SimFredholm1
    from oracle.ranksim import Partition as SP, SimArray
    rng = np.random.default_rng(60)
    nsl, nx, ny, nz = 6, 3, 4, 2
    G = (rng.standard_normal((nsl, nx, ny))
         + 1j * rng.standard_normal((nsl, nx, ny)))
    blocks = [G[:3].copy(), G[3:].copy()]
    op = pm.MPIFredholm1(torch.as_tensor(blocks[c.rank]), nz=nz,
                         saveGt=False, base_comm=c, dtype="complex128")
    op._batched = lambda A, X, opa: (
        A.conj().transpose(1, 2) @ X if opa else A @ X)
    sop = SimFredholm1(blocks, nz=nz, saveGt=False)
    assert op.shape == sop.shape
    x = rng.standard_normal(op.shape[1]) + 0j
    y = rng.standard_normal(op.shape[0]) + 0j
    xd = pm.DistributedArray.to_dist(torch.as_tensor(x), base_comm=c,
                                     partition=pm.Partition.BROADCAST)
    yd = pm.DistributedArray.to_dist(torch.as_tensor(y), base_comm=c,
                                     partition=pm.Partition.BROADCAST)
    bx = SimArray([x.copy(), x.copy()], x.shape, partition=SP.BROADCAST)
    by = SimArray([y.copy(), y.copy()], y.shape, partition=SP.BROADCAST)
    np.testing.assert_allclose(op.matvec(xd).local_array.numpy(),
                               sop.matvec(bx).locals[c.rank], rtol=1e-12)
    np.testing.assert_allclose(op.rmatvec(yd).local_array.numpy(),
                               sop.rmatvec(by).locals[c.rank], rtol=1e-12)


def body_vstack(c):
    """MPIVStack forward/adjoint distribution logic across 2 ranks vs the
    dense vertical stack (torch-matmul test adapter)."""
    import numpy as np
    import pylops_mpi_amd as pm
    rng = np.random.default_rng(70)
    mats = [rng.standard_normal((4, 6)), rng.standard_normal((3, 6))]
    A = torch.as_tensor(mats[c.rank])
    op = pm.MPIVStack([pm.CallableLocal(
        (A.shape[0], A.shape[1]),
        lambda v, A=A: A @ v, lambda v, A=A: A.T @ v)], base_comm=c)
    D = np.vstack(mats)
    assert op.shape == D.shape
    x = rng.standard_normal(6)
    y = rng.standard_normal(7)
    xd = pm.DistributedArray.to_dist(torch.as_tensor(x), base_comm=c,
                                     partition=pm.Partition.BROADCAST)
    yd = pm.DistributedArray.to_dist(torch.as_tensor(y), base_comm=c)
    got = op.matvec(xd)
    np.testing.assert_allclose(got.asarray().numpy(), D @ x, rtol=1e-12)
    gotr = op.rmatvec(yd)
    np.testing.assert_allclose(gotr.local_array.numpy(), D.T @ y,
                               rtol=1e-12)


def body_post_neighbors_overlap(c):
    """post_neighbors returns live Work handles; values land after wait
    (the derivative overlap choreography on the comm layer)."""
    import numpy as np
    t = torch.full((3, 4), float(c.rank + 1), dtype=torch.float64)
    gf = torch.empty((1, 4), dtype=torch.float64) if c.rank > 0 else None
    gb = torch.empty((1, 4), dtype=torch.float64) if c.rank < 1 else None
    works = c.post_neighbors(t[:1].contiguous() if c.rank > 0 else None,
                             t[-1:].contiguous() if c.rank < 1 else None,
                             gf, gb)
    acc = t.sum()  # "interior compute" between post and wait
    for w in works:
        w.wait()
    if c.rank == 0:
        assert torch.all(gb == 2.0)
    else:
        assert torch.all(gf == 1.0)
    assert float(acc) == 12.0 * (c.rank + 1)


def body_mask_subcomm(c):
    """Masked sub-communicator semantics (ref DistributedArray.py:74-100):
    world 2 with mask [0, 1] -> each rank reduces/gathers alone."""
    import numpy as np
    import pylops_mpi_amd as pm
    d = pm.DistributedArray((4,), c, mask=[0, 1])
    assert d.sub_comm.size == 1 and d.mask == [0, 1]
    d[:] = float(c.rank + 1)
    # masked asarray sees only this group's piece
    ma = d.asarray(masked=True)
    assert ma.numel() == 2 and torch.all(ma == float(c.rank + 1))
    # same-mask group of 2: reductions span both ranks
    d2 = pm.DistributedArray((4,), c, mask=[0, 0])
    assert d2.sub_comm.size == 2
    t = torch.tensor([float(c.rank + 1)])
    d2.sub_comm.allreduce_(t, "sum")
    assert t.item() == 3.0
    # mask mismatch raises like ref :581-585
    import pytest as _pt
    with _pt.raises(ValueError, match="Mask of both the arrays"):
        d._check_mask(d2)
    # reshape round trip (ref :899-944)
    r = d2.reshape((2, 1))
    assert r.global_shape == (4, 1) and r.local_shape == (2, 1)


def body_nonstatconv(c):
    """MPINonStationaryConvolve1D (halo sandwich + per-rank filter subset)
    vs the GLOBAL serial restatement — the wrapper's real parity claim
    (ref NonStatConvolve1d.py:101-168).  CPU test adapter local op."""
    import numpy as np
    import pylops_mpi_amd as pm
    from oracle import serial_nsconv_mv, serial_nsconv_rmv

    class CpuNsLocal(pm.CallableLocal):
        def __init__(self, dims, hs, ih, axis=-1, dtype=np.float64):
            hs_np = hs if isinstance(hs, np.ndarray) else hs.numpy()
            n = int(np.prod(dims))
            super().__init__(
                (n, n),
                lambda v: torch.as_tensor(
                    serial_nsconv_mv(v.numpy(), dims, hs_np, ih, axis)),
                lambda v: torch.as_tensor(
                    serial_nsconv_rmv(v.numpy(), dims, hs_np, ih, axis)),
                dtype=dtype)

    rng = np.random.default_rng(16)
    dims, hsize = (16,), 5
    ih = np.array([2, 6, 10, 14])
    hs = rng.standard_normal((len(ih), hsize))
    op = pm.MPINonStationaryConvolve1D(dims, hs, ih, axis=0, base_comm=c,
                                       _local_factory=CpuNsLocal)
    xg = rng.standard_normal(16)
    x = pm.DistributedArray.to_dist(torch.as_tensor(xg), base_comm=c)
    got = op.matvec(x).asarray().numpy()
    np.testing.assert_allclose(got, serial_nsconv_mv(xg, dims, hs, ih, 0),
                               rtol=1e-12, atol=1e-13)
    yg = rng.standard_normal(16)
    y = pm.DistributedArray.to_dist(torch.as_tensor(yg), base_comm=c)
    gotr = op.rmatvec(y).asarray().numpy()
    np.testing.assert_allclose(gotr, serial_nsconv_rmv(yg, dims, hs, ih, 0),
                               rtol=1e-12, atol=1e-13)


def body_proximal_call_reduction(c):
    """MPIProxOperator.__call__ reduction semantics across 2 ranks
    (ref proximal/ProxOperator.py:56-110: Box -> LAND, L0/L1 -> SUM).
    The functional evaluations are torch ops, so this runs on gloo/CPU;
    the prox kernels themselves are GPU-only (test_gpu_proximal)."""
    import numpy as np
    from pylops_mpi_amd import DistributedArray
    from pylops_mpi_amd.proximal import Box, L1, MPIProxOperator
    rng = np.random.default_rng(80 + c.rank)
    nloc = 40
    xl = rng.standard_normal(nloc)
    x = DistributedArray((2 * nloc,), c)
    x[:] = torch.from_numpy(xl)
    # L1: global sum of per-rank sums
    l1d = MPIProxOperator(L1(sigma=0.5))
    want = 0.0
    for r in range(2):
        want += 0.5 * np.abs(
            np.random.default_rng(80 + r).standard_normal(nloc)).sum()
    got = l1d(x)
    np.testing.assert_allclose(got, want, rtol=1e-12)
    # Box: logical AND across ranks — make it fail on rank 1 only
    y = DistributedArray((2 * nloc,), c)
    y[:] = torch.zeros(nloc, dtype=torch.float64) + (5.0 if c.rank else 0.5)
    boxd = MPIProxOperator(Box(lower=0.0, upper=1.0))
    assert boxd(y) is False
    z = DistributedArray((2 * nloc,), c)
    z[:] = torch.full((nloc,), 0.5, dtype=torch.float64)
    assert boxd(z) is True


def body_redistribute(c):
    """redistribute(axis=new): all-to-all realignment vs direct slicing
    of the global array (ref DistributedArray.py:493-552; balanced
    output along the new axis)."""
    import numpy as np
    from pylops_mpi_amd import DistributedArray
    from pylops_mpi_amd.distributedarray import local_split, Partition
    g = np.arange(5 * 6 * 3, dtype=np.float64).reshape(5, 6, 3)
    x = DistributedArray.to_dist(torch.from_numpy(g), c)
    y = x.redistribute(axis=1)
    # expected: balanced split of axis 1 with the remainder rule
    counts = [local_split((5, 6, 3), c.size, r, Partition.SCATTER, 1)[1]
              for r in range(c.size)]
    start = sum(counts[:c.rank])
    np.testing.assert_array_equal(
        y.local_array.numpy(),
        g[:, start:start + counts[c.rank], :])
    assert y.global_shape == (5, 6, 3) and y.axis == 1
    # round trip back to axis 0 returns the original locals
    z = y.redistribute(axis=0)
    np.testing.assert_array_equal(z.local_array.numpy(),
                                  x.local_array.numpy())


def body_fftnd(c):
    """MPIFFTND across 2 ranks vs the serial oracle (the pencil
    transposes + reshaped I/O are the distributed content; torch.fft
    runs on CPU here, rocFFT on the GPU box)."""
    import numpy as np
    import oracle
    import pylops_mpi_amd as pm
    from pylops_mpi_amd.distributedarray import local_split, Partition
    rng = np.random.default_rng(90)
    for par in (
        dict(dims=(9, 8), axes=(0, 1), real=True, norm="none",
             dtype=np.float64, imag=0),
        dict(dims=(8, 9), axes=(1, 0), real=False, norm="1/n",
             dtype=np.complex128, imag=1j),
        dict(dims=(5, 6, 4), axes=(2, 0, 1), real=True, norm="1/n",
             dtype=np.float64, imag=0),
        # 0 not in axes: no pencil transpose at all
        dict(dims=(6, 5, 4), axes=(1, 2), real=True, norm="none",
             dtype=np.float64, imag=0),
    ):
        op = pm.MPIFFTND(dims=par["dims"], axes=par["axes"],
                         norm=par["norm"], real=par["real"],
                         dtype=par["dtype"], base_comm=c)
        n = int(np.prod(par["dims"]))
        xg = rng.standard_normal(n)
        if par["imag"]:
            xg = xg + 1j * rng.standard_normal(n)
        xg = xg.astype(par["dtype"])
        x = pm.DistributedArray.to_dist(torch.from_numpy(xg), c)
        y = op.matvec(x)
        y_ref = oracle.serial_fftnd_mv(xg, par["dims"], par["axes"],
                                       norm=par["norm"], real=par["real"])
        np.testing.assert_allclose(y.asarray().numpy(), y_ref,
                                   rtol=1e-10, atol=1e-11)
        yg = rng.standard_normal(op.shape[0]) \
            + 1j * rng.standard_normal(op.shape[0])
        yd = pm.DistributedArray.to_dist(torch.from_numpy(yg), c)
        z = op.rmatvec(yd)
        z_ref = oracle.serial_fftnd_rmv(yg, par["dims"], par["axes"],
                                        norm=par["norm"], real=par["real"])
        np.testing.assert_allclose(z.asarray().numpy(), z_ref,
                                   rtol=1e-10, atol=1e-11)


def body_norm_axis(c):
    """norm(ord, axis) across 2 ranks vs np.linalg.norm on the global
    array (ref DistributedArray.py:719-771,828-838), both when the
    reduced axis is local (allgather+concat) and when it is the
    distributed axis (elementwise allreduce)."""
    import numpy as np
    from pylops_mpi_amd import DistributedArray
    rng = np.random.default_rng(70)
    g = rng.standard_normal((7, 5, 4))
    x = DistributedArray.to_dist(torch.from_numpy(g), c)
    for ordv in (1, 2, np.inf, 0):
        for ax in (0, 1, 2):
            got = x.norm(ord=ordv, axis=ax).numpy()
            want = np.linalg.norm(g, ord=ordv, axis=ax)
            np.testing.assert_allclose(got, want, rtol=1e-13,
                                       err_msg=f"ord={ordv} axis={ax}")


def body_broadcast_setitem(c):
    """__setitem__ partition semantics across 2 ranks (ref
    DistributedArray.py:217-252): BROADCAST re-broadcasts rank 0's
    assignment; UNSAFE_BROADCAST keeps each rank's local value."""
    import numpy as np
    from pylops_mpi_amd import DistributedArray, Partition
    b = DistributedArray((6,), c, Partition.BROADCAST)
    b[:] = torch.full((6,), float(c.rank + 1), dtype=torch.float64)
    np.testing.assert_array_equal(b.local_array.numpy(), np.ones(6))
    u = DistributedArray((6,), c, Partition.UNSAFE_BROADCAST)
    u[:] = torch.full((6,), float(c.rank + 1), dtype=torch.float64)
    np.testing.assert_array_equal(u.local_array.numpy(),
                                  np.full(6, c.rank + 1.0))


def body_halo_guard_deterministic(c):
    """Undersized halo sources raise ValueError on EVERY rank (r01
    advice: the old per-rank guard exempted edge ranks, which would post
    a mismatched irecv and hang instead of raising)."""
    import pylops_mpi_amd as pm
    shapes = [(1, 4), (2, 4)]
    d = pm.DistributedArray(
        (3, 4), c, pm.Partition.SCATTER, 0,
        local_array=torch.zeros(shapes[c.rank], dtype=torch.float64),
        local_shapes=shapes, dtype=np.float64)
    # rank 0 (an edge rank, and a sender for cells_front) holds only 1
    # plane < 2 requested -> both ranks must raise, deterministically
    with pytest.raises(ValueError, match="Local Shape at rank=0"):
        d.add_ghost_cells(cells_front=2)
    with pytest.raises(ValueError, match="Local Shape at rank=0"):
        d.halo_exchange(2)


def body_subcomm_split(c):
    """split_by on a SUBcommunicator whose members are not the identity
    prefix (r01 advice, medium): group-rank colors translate through
    comm.ranks, and ranks outside the subcomm need not participate."""
    sub = c.split_by([0, 1])     # rank 0 alone; rank 1 alone
    assert sub.size == 1 and sub.ranks == [c.rank]
    sub2 = c.split_by([5, 5])    # both in one group
    assert sub2.size == 2 and sub2.ranks == [0, 1]
    # nested split on the subcomm: rank 0 does NOT call it (the old
    # world-collective new_group would deadlock here)
    if c.rank == 1:
        nested = sub.split_by([0])
        assert nested.size == 1 and nested.ranks == [1]
    t = torch.tensor([float(c.rank + 1)])
    sub2.allreduce_(t, "sum")
    assert t.item() == 3.0
