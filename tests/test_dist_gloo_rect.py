"""Rectangular-grid SUMMA on CPU/gloo, world_size=8 — the BASELINE cfg4
layout (2x4 process grid) the reference's square-only SUMMA cannot run
(ref MatrixMult.py:564-566; its active_grid_comm would idle 4 of the 8
ranks).  Covers:

  * small uneven shapes on 2x4 AND 4x2 grids vs dense A@X / A^H y;
  * the cfg4-shaped 32768x32768 fp32 apply with all 8 ranks active,
    verified against the closed form of a rank-1-structured A
    (A = u v^T, so y = u (v^T X) — O(N) to check without ever
    materializing the 4 GiB global A).

The local GEMM/transpose are overridden with torch matmul (CPU test
adapter; the MFMA pam_gemm path is GPU-only, covered by the gpu suite
— the rect orchestration layer is identical on both)."""
import os
import sys

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
WORLD = 8


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
    port = 27500 + (hash(fn_name) % 1000)
    mp.spawn(_worker, args=(port, fn_name), nprocs=WORLD, join=True)


def _patch_cpu_gemm(op):
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
    op._local_transpose = lambda A: A.t().conj().contiguous()
    return op


def _dist_from_locals(pm, c, locals_per_rank, dtype=np.float64):
    counts = [int(v.size) for v in locals_per_rank]
    n = int(sum(counts))
    return pm.DistributedArray(
        (n,), c, pm.Partition.SCATTER, 0,
        local_array=torch.as_tensor(np.ascontiguousarray(
            locals_per_rank[c.rank])),
        local_shapes=[(v,) for v in counts], dtype=dtype)


def _check_rect(c, shapes, grid, rtol=1e-12):
    import pylops_mpi_amd as pm
    from pylops_mpi_amd.matmult import summa_grid_splits
    N, K, M = shapes
    Pr, Pc = grid
    rng = np.random.default_rng(123)
    A = rng.standard_normal((N, K))
    X = rng.standard_normal((K, M))
    Y = rng.standard_normal((N, M))
    a_sl, x_sl, y_sl = summa_grid_splits(N, K, M, Pr, Pc)
    op = pm.matmult.MPIMatrixMult(
        torch.as_tensor(A[a_sl[c.rank]].copy()), M, kind="summa",
        base_comm=c, grid=grid)
    _patch_cpu_gemm(op)
    assert op.N == N and op.K == K
    x = _dist_from_locals(pm, c, [X[s].ravel() for s in x_sl])
    got = op.matvec(x)
    want = (A @ X)[y_sl[c.rank]]
    np.testing.assert_allclose(got.local_array.numpy(),
                               want.ravel(), rtol=rtol, atol=1e-10)
    yv = _dist_from_locals(pm, c, [Y[s].ravel() for s in y_sl])
    gotr = op.rmatvec(yv)
    wantr = (A.conj().T @ Y)[x_sl[c.rank]]
    np.testing.assert_allclose(gotr.local_array.numpy(),
                               wantr.ravel(), rtol=rtol, atol=1e-10)


def body_rect_2x4(c):
    _check_rect(c, (7, 13, 10), (2, 4))


def body_rect_4x2(c):
    _check_rect(c, (11, 13, 9), (4, 2))


def body_rect_2x4_even(c):
    _check_rect(c, (16, 16, 8), (2, 4))


def body_cfg4_32768(c):
    """cfg4 shape: 32768^2 fp32 on the 2x4 grid, all 8 ranks active.
    A is rank-1 (u v^T) so the check is closed-form O(N); the SUMMA
    data movement and panel GEMM sizes are exactly cfg4's."""
    import pylops_mpi_amd as pm
    from pylops_mpi_amd.matmult import summa_grid_splits
    N = K = 32768
    M = 8
    grid = (2, 4)
    Pr, Pc = grid
    rng = np.random.default_rng(5)
    u = rng.standard_normal(N).astype(np.float32)
    v = rng.standard_normal(K).astype(np.float32)
    X = rng.standard_normal((K, M)).astype(np.float32)
    a_sl, x_sl, y_sl = summa_grid_splits(N, K, M, Pr, Pc)
    rs, cs = a_sl[c.rank]
    A_local = np.outer(u[rs], v[cs])      # 512 MiB fp32, never global
    op = pm.matmult.MPIMatrixMult(
        torch.as_tensor(A_local), M, kind="summa", base_comm=c,
        grid=grid, dtype="float32")
    _patch_cpu_gemm(op)
    assert op.N == N and op.K == K
    del A_local
    # every rank holds a non-empty tile (the whole point vs the
    # reference's square-only active grid)
    counts = op._rect_counts("n")
    assert all(ct > 0 for ct in counts), counts
    x = _dist_from_locals(pm, c, [X[s].astype(np.float32).ravel()
                                  for s in x_sl], dtype=np.float32)
    got = op.matvec(x)
    vtX = v.astype(np.float64) @ X.astype(np.float64)    # (M,)
    yrs, ycs = y_sl[c.rank]
    want = np.outer(u[yrs].astype(np.float64), vtX[ycs])
    np.testing.assert_allclose(got.local_array.numpy().astype(np.float64),
                               want.ravel(), rtol=2e-4, atol=1e-2)
    # adjoint at the same scale: z = A^H y = v (u^T y)
    Yin = rng.standard_normal((N, M)).astype(np.float32)
    yv = _dist_from_locals(pm, c, [Yin[s].ravel() for s in y_sl],
                           dtype=np.float32)
    gotr = op.rmatvec(yv)
    utY = u.astype(np.float64) @ Yin.astype(np.float64)
    zrs, zcs = x_sl[c.rank]
    wantr = np.outer(v[zrs].astype(np.float64), utY[zcs])
    np.testing.assert_allclose(gotr.local_array.numpy().astype(np.float64),
                               wantr.ravel(), rtol=2e-4, atol=1e-2)


@pytest.mark.parametrize("body", [
    "body_rect_2x4", "body_rect_4x2", "body_rect_2x4_even",
    "body_cfg4_32768",
])
def test_gloo_world8_rect_summa(body):
    _spawn(body)
