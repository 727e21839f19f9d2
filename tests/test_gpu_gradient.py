"""GPU parity: serial per-axis derivative local ops (fd_serial kernel) and
the MPIGradient / MPILaplacian compositions (world size 1; the multi-rank
pieces — MPIFirstDerivative halo, BlockDiag, StackedVStack — have their
own multi-rank coverage)."""
import numpy as np
import pytest
import torch
from numpy.testing import assert_allclose

import oracle
import pylops_mpi_amd as pm

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _init():
    from pylops_mpi_amd.comm import init_default_comm
    init_default_comm(torch.device("cuda:0"))


def dev(a):
    return torch.as_tensor(a, device="cuda:0")


def host(t):
    return t.cpu().numpy()


def serial_axis_fd1(x, dims, axis, sampling, kind, edge, order):
    xg = np.moveaxis(x.reshape(dims), axis, 0)
    y = oracle.serial_fd1_matvec(xg, sampling, kind, edge, order)
    return np.moveaxis(y, 0, axis).ravel()


def serial_axis_fd2(x, dims, axis, sampling, kind, edge):
    xg = np.moveaxis(x.reshape(dims), axis, 0)
    y = oracle.serial_fd2_matvec(xg, sampling, kind, edge)
    return np.moveaxis(y, 0, axis).ravel()


@pytest.mark.parametrize("axis", [0, 1, 2, -1])
@pytest.mark.parametrize("kind,order,edge", [
    ("centered", 3, False), ("centered", 3, True),
    ("centered", 5, True), ("forward", 3, False), ("backward", 3, False),
])
def test_fd1_local_vs_serial(axis, kind, order, edge):
    dims = (9, 8, 7)
    n = int(np.prod(dims))
    rng = np.random.default_rng(1)
    xg = rng.standard_normal(n)
    op = pm.FirstDerivativeLocal(dims, axis=axis, sampling=1.4, kind=kind,
                                 edge=edge, order=order)
    got = host(op.matvec(dev(xg)))
    want = serial_axis_fd1(xg, dims, axis % 3, 1.4, kind, edge, order)
    assert_allclose(got, want, rtol=1e-13, atol=1e-14)
    # adjoint vs dense transpose
    A = oracle.dense_matrix_from_matvec(
        lambda v: serial_axis_fd1(v, dims, axis % 3, 1.4, kind, edge, order),
        n)
    gotr = host(op.rmatvec(dev(xg)))
    assert_allclose(gotr, A.T @ xg, rtol=1e-12, atol=1e-12)


@pytest.mark.parametrize("axis", [1, 2])
@pytest.mark.parametrize("kind,edge", [("centered", False),
                                       ("centered", True),
                                       ("forward", False)])
def test_fd2_local_vs_serial(axis, kind, edge):
    dims = (6, 9, 8)
    n = int(np.prod(dims))
    rng = np.random.default_rng(2)
    xg = rng.standard_normal(n)
    op = pm.SecondDerivativeLocal(dims, axis=axis, sampling=0.9, kind=kind,
                                  edge=edge)
    got = host(op.matvec(dev(xg)))
    want = serial_axis_fd2(xg, dims, axis, 0.9, kind, edge)
    assert_allclose(got, want, rtol=1e-13, atol=1e-14)


def test_fd1_local_complex():
    dims = (8, 6)
    n = int(np.prod(dims))
    rng = np.random.default_rng(3)
    xg = rng.standard_normal(n) + 1j * rng.standard_normal(n)
    op = pm.FirstDerivativeLocal(dims, axis=1, sampling=1.1,
                                 dtype=np.complex128)
    got = host(op.matvec(dev(xg)))
    want = serial_axis_fd1(xg, dims, 1, 1.1, "centered", False, 3)
    assert_allclose(got, want, rtol=1e-13, atol=1e-14)


def test_gradient_vs_serial():
    # ref Gradient.py:101-118 composition; expected = per-axis serial FDs
    dims = (12, 7, 5)
    n = int(np.prod(dims))
    rng = np.random.default_rng(4)
    xg = rng.standard_normal(n)
    samp = (1.0, 2.0, 0.5)
    op = pm.MPIGradient(dims, sampling=samp, edge=True, kind="centered")
    x = pm.DistributedArray.to_dist(dev(xg))
    y = op.matvec(x)
    assert y.narrays == 3
    for iax in range(3):
        want = serial_axis_fd1(xg, dims, iax, samp[iax], "centered", True, 3)
        assert_allclose(host(y[iax].asarray()), want, rtol=1e-12,
                        atol=1e-13)
    # adjoint consistency (dottest through the stack)
    vg = [rng.standard_normal(n) for _ in range(3)]
    v = pm.StackedDistributedArray(
        [pm.DistributedArray.to_dist(dev(g)) for g in vg])
    u = pm.DistributedArray.to_dist(dev(rng.standard_normal(n)))
    yy = float(op.matvec(u).dot(v))
    xx = float(u.dot(op.rmatvec(v)))
    assert_allclose(yy, xx, rtol=1e-10)


def test_laplacian_vs_serial():
    # ref Laplacian.py:97-126; expected = weighted sum of serial fd2 per axis
    dims = (10, 8, 6)
    n = int(np.prod(dims))
    rng = np.random.default_rng(5)
    xg = rng.standard_normal(n)
    axes, weights, samp = (0, 1, 2), (1.0, 2.0, 0.5), (1.0, 1.5, 0.7)
    op = pm.MPILaplacian(dims, axes=axes, weights=weights, sampling=samp,
                         edge=True, kind="centered")
    x = pm.DistributedArray.to_dist(dev(xg))
    y = op.matvec(x)
    want = sum(w * serial_axis_fd2(xg, dims, ax, s, "centered", True)
               for ax, w, s in zip(axes, weights, samp))
    assert_allclose(host(y.asarray()), want, rtol=1e-12, atol=1e-13)
    u = pm.DistributedArray.to_dist(dev(rng.standard_normal(n)))
    v = pm.DistributedArray.to_dist(dev(rng.standard_normal(n)))
    assert pm.dottest(op, u, v, rtol=1e-10)


# ------------------------------------------------- non-stationary convolution
@pytest.mark.parametrize("dims,axis", [((24,), 0), ((3, 24), 1),
                                       ((24, 4), 0)])
def test_nsconv_local_vs_serial(dims, axis):
    from oracle import serial_nsconv_mv, serial_nsconv_rmv
    rng = np.random.default_rng(20)
    hsize = 7
    ih = np.array([3, 9, 15, 21])
    hs = rng.standard_normal((len(ih), hsize))
    n = int(np.prod(dims))
    xg = rng.standard_normal(n)
    op = pm.NonStationaryConvolve1DLocal(dims, dev(hs), ih, axis=axis)
    assert_allclose(host(op.matvec(dev(xg))),
                    serial_nsconv_mv(xg, dims, hs, ih, axis),
                    rtol=1e-12, atol=1e-13)
    assert_allclose(host(op.rmatvec(dev(xg))),
                    serial_nsconv_rmv(xg, dims, hs, ih, axis),
                    rtol=1e-12, atol=1e-13)


def test_mpinonstatconv_world1():
    from oracle import serial_nsconv_mv, serial_nsconv_rmv
    rng = np.random.default_rng(21)
    dims = (32,)
    ih = np.array([4, 12, 20, 28])
    hs = rng.standard_normal((len(ih), 5))
    op = pm.MPINonStationaryConvolve1D(dims, dev(hs), ih, axis=0)
    xg = rng.standard_normal(32)
    x = pm.DistributedArray.to_dist(dev(xg))
    assert_allclose(host(op.matvec(x).asarray()),
                    serial_nsconv_mv(xg, dims, hs, ih, 0),
                    rtol=1e-12, atol=1e-13)
    u = pm.DistributedArray.to_dist(dev(rng.standard_normal(32)))
    v = pm.DistributedArray.to_dist(dev(rng.standard_normal(32)))
    assert pm.dottest(op, u, v, rtol=1e-10)


def test_gradient_laplacian_vs_reference_fixtures():
    """Direct product <-> REFERENCE pin for the composed operators: the
    P=1 grad_/lap_ fixtures in golden_ref.npz were generated by
    executing /root/reference/pylops_mpi (tests/golden/refgen.py)."""
    import os
    import sys
    sys.path.insert(0, os.path.join(os.path.dirname(__file__), "golden"))
    import refgen
    g = np.load(refgen.GOLDEN_PATH)
    ng = int(np.prod(refgen.GRAD_DIMS))
    xg = refgen.make_global_x(ng, 1)
    gop = pm.MPIGradient(dims=refgen.GRAD_DIMS, sampling=refgen.GRAD_SAMP,
                         edge=False, kind="centered")
    xd = pm.DistributedArray.to_dist(dev(xg))
    ym = gop.matvec(xd)
    for i in range(3):
        assert_allclose(host(ym.distarrays[i].asarray()),
                        g[f"grad_P1_mv{i}"], rtol=1e-12, atol=1e-12)
    ys = [refgen.make_global_x(ng, 1, seed_shift=2 + i) for i in range(3)]
    for i, di in enumerate(ym.distarrays):
        di[:] = dev(ys[i]).reshape(di.local_array.shape)
    assert_allclose(host(gop.rmatvec(ym).asarray()), g["grad_P1_rmv"],
                    rtol=1e-12, atol=1e-12)
    lop = pm.MPILaplacian(dims=refgen.GRAD_DIMS, axes=refgen.LAP_AXES,
                          weights=refgen.LAP_W, sampling=refgen.LAP_SAMP,
                          edge=False, kind="centered")
    assert_allclose(host(lop.matvec(xd).asarray()), g["lap_P1_mv"],
                    rtol=1e-12, atol=1e-12)
    yl = refgen.make_global_x(ng, 1, seed_shift=5)
    assert_allclose(host(lop.rmatvec(
        pm.DistributedArray.to_dist(dev(yl))).asarray()),
        g["lap_P1_rmv"], rtol=1e-12, atol=1e-12)
