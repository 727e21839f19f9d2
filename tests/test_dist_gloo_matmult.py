"""MPIMatrixMult distributed logic on a 2x2 grid, CPU/gloo, world_size=4:
grid split_by, block allgather/allreduce, SUMMA bcast loops and the
adjoint A^T exchange — checked against the reference's own pin
(A_glob @ X_glob, ref tests/test_matrixmult.py:37-45).

The local GEMM/transpose are overridden with torch matmul (test adapter;
the MFMA pam_gemm path is GPU-only and covered by the gpu suite)."""
import os
import sys

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
WORLD = 4


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
    port = 28500 + (hash(fn_name) % 1000)
    mp.spawn(_worker, args=(port, fn_name), nprocs=WORLD, join=True)


def _patch_cpu_gemm(op):
    """Test adapter: torch matmul instead of the GPU-only pam_gemm."""
    def gemm(A, B, C=None, accumulate=False):
        out = A @ B
        if C is None:
            return out
        if accumulate:
            C += out
        else:
            C.copy_(out)
        return C

    op._local_gemm = gemm
    op._local_transpose = lambda A: A.t().contiguous()
    return op


def _mk(c, shapes):
    import pylops_mpi_amd as pm
    N, K, M = shapes
    rng = np.random.default_rng(77)
    A = rng.standard_normal((N, K))
    X = rng.standard_normal((K, M))
    Y = rng.standard_normal((N, M))
    return pm, A, X, Y


def _dist_from_locals(pm, c, locals_per_rank):
    counts = [v.size for v in locals_per_rank]
    n = int(sum(counts))
    return pm.DistributedArray(
        (n,), c, pm.Partition.SCATTER, 0,
        local_array=torch.as_tensor(locals_per_rank[c.rank]),
        local_shapes=[(int(v),) for v in counts], dtype=np.float64)


def _check_block(c, shapes):
    from oracle import matmult as om
    pm, A, X, Y = _mk(c, shapes)
    N, K, M = shapes
    inputs = om.block_inputs(A, X, WORLD)
    op = pm.matmult.MPIMatrixMult(
        torch.as_tensor(inputs[c.rank][0]), M, kind="block", base_comm=c)
    _patch_cpu_gemm(op)
    assert op.N == N and op.K == K
    x = _dist_from_locals(pm, c, [v for _, v in inputs])
    got = op.matvec(x)
    want = om.block_expected_mv(A, X, WORLD)
    np.testing.assert_allclose(got.local_array.numpy(), want[c.rank],
                               rtol=1e-12, atol=1e-12)
    # adjoint: input is the (N, M_loc) layout
    ylocals = [Y[:, om.split_slice(M, 2, q // 2)].ravel()
               for q in range(WORLD)]
    yv = _dist_from_locals(pm, c, ylocals)
    gotr = op.rmatvec(yv)
    wantr = om.block_expected_rmv(A, Y, WORLD)
    np.testing.assert_allclose(gotr.local_array.numpy(), wantr[c.rank],
                               rtol=1e-12, atol=1e-12)


def _check_summa(c, shapes, saveAt=False):
    from oracle import matmult as om
    pm, A, X, Y = _mk(c, shapes)
    N, K, M = shapes
    inputs = om.summa_inputs(A, X, WORLD)
    op = pm.matmult.MPIMatrixMult(
        torch.as_tensor(inputs[c.rank][0]), M, kind="summa", base_comm=c,
        saveAt=False)
    _patch_cpu_gemm(op)
    if saveAt:
        op.At = op._local_transpose(op.A)
    assert op.N == N and op.K == K
    x = _dist_from_locals(pm, c, [v for _, v in inputs])
    got = op.matvec(x)
    want = om.summa_expected_mv(A, X, WORLD)
    np.testing.assert_allclose(got.local_array.numpy(), want[c.rank],
                               rtol=1e-12, atol=1e-12)
    yt = [om.summa_tile(Y, WORLD, q).ravel() for q in range(WORLD)]
    yv = _dist_from_locals(pm, c, yt)
    gotr = op.rmatvec(yv)
    wantr = om.summa_expected_rmv(A, Y, WORLD)
    np.testing.assert_allclose(gotr.local_array.numpy(), wantr[c.rank],
                               rtol=1e-12, atol=1e-12)


# ------------------------------------------------------------- worker bodies
def body_block_even(c):
    _check_block(c, (8, 6, 10))


def body_block_uneven(c):
    _check_block(c, (7, 5, 9))


def body_summa_even(c):
    _check_summa(c, (8, 6, 10))


def body_summa_uneven(c):
    _check_summa(c, (7, 5, 9))


def body_summa_saveat(c):
    _check_summa(c, (9, 11, 6), saveAt=True)


def body_active_grid(c):
    import pylops_mpi_amd as pm
    new_comm, new_rank, row, col, act = pm.matmult.active_grid_comm(c, 8, 8)
    # world 4 = 2x2, all active
    assert act and new_comm.size == 4
    assert (row, col) == divmod(c.rank, 2)
    # tiny N: only a 1x1 grid stays active
    new_comm2, _, _, _, act2 = pm.matmult.active_grid_comm(c, 1, 8)
    assert act2 == (c.rank == 0)


def body_block_gather(c):
    import pylops_mpi_amd as pm
    from oracle import matmult as om
    rng = np.random.default_rng(9)
    G = rng.standard_normal((7, 9))
    tile = om.summa_tile(G, WORLD, c.rank)
    counts = [om.summa_tile(G, WORLD, q).size for q in range(WORLD)]
    d = pm.DistributedArray((int(sum(counts)),), c, pm.Partition.SCATTER, 0,
                            local_array=torch.as_tensor(tile.ravel()),
                            local_shapes=[(int(v),) for v in counts],
                            dtype=np.float64)
    C = pm.matmult.block_gather(d, (7, 9), c)
    np.testing.assert_allclose(C.numpy(), G, rtol=0, atol=0)


@pytest.mark.parametrize("body", [
    "body_block_even", "body_block_uneven", "body_summa_even",
    "body_summa_uneven", "body_summa_saveat", "body_active_grid",
    "body_block_gather",
])
def test_gloo_world4_matmult(body):
    _spawn(body)


def _halo_expected(G, grid, rank, halo_spec, op):
    """Independent pin: after the exchange, the extended block equals the
    zero-padded GLOBAL window around this rank's Cartesian block."""
    import math as _m
    import numpy as np
    nd = G.ndim
    coords = np.unravel_index(rank, grid)
    halo = op._parse_halo(halo_spec, rank)
    starts, ends = [], []
    for gdim, c, p in zip(G.shape, coords, grid):
        blk = _m.ceil(gdim / p)
        starts.append(c * blk)
        ends.append(min(c * blk + blk, gdim))
    ext_shape = tuple((ends[a] - starts[a]) + halo[2 * a] + halo[2 * a + 1]
                      for a in range(nd))
    out = np.zeros(ext_shape, dtype=G.dtype)
    src, dst = [], []
    for a in range(nd):
        lo = starts[a] - halo[2 * a]
        hi = ends[a] + halo[2 * a + 1]
        s0, s1 = max(lo, 0), min(hi, G.shape[a])
        src.append(slice(s0, s1))
        dst.append(slice(s0 - lo, (s0 - lo) + (s1 - s0)))
    out[tuple(dst)] = G[tuple(src)]
    return out


def _check_halo(c, dims, grid, halo_spec):
    import numpy as np
    import pylops_mpi_amd as pm
    rng = np.random.default_rng(90)
    G = rng.standard_normal(dims)
    op = pm.MPIHalo(dims, halo_spec, proc_grid_shape=grid, comm=c)
    blocks = []
    import math as _m
    for q in range(WORLD):
        coords = np.unravel_index(q, grid)
        sl = []
        for gdim, cc, p in zip(dims, coords, grid):
            blk = _m.ceil(gdim / p)
            sl.append(slice(cc * blk, min(cc * blk + blk, gdim)))
        blocks.append(G[tuple(sl)].copy())
    counts = [b.size for b in blocks]
    x = pm.DistributedArray(
        (int(np.prod(dims)),), c, pm.Partition.SCATTER, 0,
        local_array=torch.as_tensor(blocks[c.rank].ravel()),
        local_shapes=[(int(v),) for v in counts], dtype=np.float64)
    y = op.matvec(x)
    want = _halo_expected(G, grid, c.rank, halo_spec, op)
    np.testing.assert_allclose(
        y.local_array.numpy().reshape(want.shape), want, rtol=0, atol=0)
    # adjoint strips back to the original block
    back = op.rmatvec(y)
    np.testing.assert_allclose(back.local_array.numpy(),
                               blocks[c.rank].ravel(), rtol=0, atol=0)


def body_halo_1d(c):
    _check_halo(c, (13, 8), (1, 4), 2)
    # oversized halo rejected like ref Halo.py:309-313
    import pytest as _pt
    import pylops_mpi_amd as pm
    with _pt.raises(ValueError, match="exceeds local block size"):
        pm.MPIHalo((13, 5), 2, proc_grid_shape=(1, 4), comm=c)


def body_halo_2d(c):
    _check_halo(c, (9, 11), (2, 2), 1)


def body_halo_2d_tuple(c):
    _check_halo(c, (12, 10), (2, 2), (2, 1))


@pytest.mark.parametrize("body", ["body_halo_1d", "body_halo_2d",
                                  "body_halo_2d_tuple"])
def test_gloo_world4_halo(body):
    _spawn(body)


def body_fftnd_world4(c):
    """MPIFFTND across 4 ranks: the multi-transpose path (axes[-1] == 0
    forces the input realignment, ref FFTND.py:190-196) and a 3-D real
    case, vs the serial oracle."""
    import numpy as np
    import torch
    import oracle
    import pylops_mpi_amd as pm
    rng = np.random.default_rng(91)
    for par in (
        dict(dims=(12, 9), axes=(1, 0), real=False, norm="none",
             dtype=np.complex128, imag=1j),
        dict(dims=(11, 8, 6), axes=(0, 1, 2), real=True, norm="1/n",
             dtype=np.float64, imag=0),
        dict(dims=(8, 11, 6), axes=(2, 1, 0), real=True, norm="none",
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
        yg = (rng.standard_normal(op.shape[0])
              + 1j * rng.standard_normal(op.shape[0]))
        yd = pm.DistributedArray.to_dist(torch.from_numpy(yg), c)
        z = op.rmatvec(yd)
        z_ref = oracle.serial_fftnd_rmv(yg, par["dims"], par["axes"],
                                        norm=par["norm"], real=par["real"])
        np.testing.assert_allclose(z.asarray().numpy(), z_ref,
                                   rtol=1e-10, atol=1e-11)


@pytest.mark.parametrize("body", ["body_fftnd_world4"])
def test_gloo_world4_fftnd(body):
    _spawn(body)
