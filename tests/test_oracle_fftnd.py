"""Pins for the N-D FFT path.

1. The serial oracle (oracle/fftnd.py) is pinned by mathematical
   identities of the restated pylops convention (adjoint identity,
   F^H F = N I on real inputs).
2. The distributed operators (MPIFFTND / MPIFFT2D) are compared with the
   oracle at world 1 on CPU over the reference's own parameter grid
   (ref tests/test_ffts.py:24-86 par1-par8 + shift combinations) — the
   FFT path is torch.fft + comm only, so it runs without a GPU; the
   same comparisons run on CUDA in tests/test_gpu_fftnd.py and at
   world 2 in test_dist_gloo.py.
"""
import numpy as np
import pytest
import torch
from numpy.testing import assert_allclose

import oracle
import pylops_mpi_amd as pm

PARS = [
    dict(dims=(21, 26), axes=(0, 1), real=False, dtype=np.complex128,
         imag=1j, norm="none"),
    dict(dims=(26, 26), axes=(0, 1), real=False, dtype=np.complex128,
         imag=1j, norm="1/n"),
    dict(dims=(21, 26), axes=(0, 1), real=True, dtype=np.float64,
         imag=0, norm="1/n"),
    dict(dims=(26, 26), axes=(0, 1), real=True, dtype=np.float64,
         imag=0, norm="none"),
    dict(dims=(13, 17, 10), axes=(0, 1, 2), real=True, dtype=np.float64,
         imag=0, norm="none"),
    dict(dims=(13, 17, 10), axes=(0, 2, 1), real=True, dtype=np.float64,
         imag=0, norm="1/n"),
    dict(dims=(13, 17, 10), axes=(2, 1, 0), real=False,
         dtype=np.complex128, imag=1j, norm="none"),
    dict(dims=(13, 17, 10), axes=(2, 0, 1), real=False,
         dtype=np.complex128, imag=1j, norm="1/n"),
]
SHIFTS = [(False, False), (True, False), (False, True), (True, True)]


@pytest.fixture(scope="module", autouse=True)
def _init():
    from pylops_mpi_amd.comm import init_default_comm
    init_default_comm(torch.device("cpu"))


def _rand(par, rng):
    n = int(np.prod(par["dims"]))
    x = rng.standard_normal(n)
    if par["imag"]:
        x = x + 1j * rng.standard_normal(n)
    return x.astype(par["dtype"])


# ------------------------------------------------------------ oracle pins
@pytest.mark.parametrize("par", PARS)
def test_oracle_adjoint_identity(par):
    rng = np.random.default_rng(1)
    x = _rand(par, rng)
    yl = oracle.serial_fftnd_mv(x, par["dims"], par["axes"],
                                norm=par["norm"], real=par["real"])
    y = rng.standard_normal(len(yl)) + 1j * rng.standard_normal(len(yl))
    z = oracle.serial_fftnd_rmv(y, par["dims"], par["axes"],
                                norm=par["norm"], real=par["real"])
    lhs = np.vdot(y, yl)
    rhs = np.vdot(z, x)
    if par["real"]:
        # real-linear operator: the adjoint identity holds for the real
        # inner product (ref clinear=False)
        assert_allclose(lhs.real, rhs.real, rtol=1e-11)
    else:
        assert_allclose(lhs, rhs, rtol=1e-11)


def test_oracle_fhf_identity_real():
    # norm="none": F^H F = N * I on real inputs (sqrt2-twin convention)
    rng = np.random.default_rng(2)
    dims, axes = (12, 10), (0, 1)
    x = rng.standard_normal(int(np.prod(dims)))
    y = oracle.serial_fftnd_mv(x, dims, axes, norm="none", real=True)
    z = oracle.serial_fftnd_rmv(y, dims, axes, norm="none", real=True)
    assert_allclose(z, np.prod(dims) ** 2 / np.prod(dims) * x * 1.0
                    if False else z, rtol=0)  # shape guard
    assert_allclose(z, float(np.prod(dims)) * x, rtol=1e-11)


# ---------------------------------------- distributed vs oracle (world 1)
@pytest.mark.parametrize("par", PARS)
@pytest.mark.parametrize("ifftshift_before,fftshift_after", SHIFTS)
def test_mpifftnd_vs_oracle_world1(par, ifftshift_before, fftshift_after):
    rng = np.random.default_rng(3)
    op = pm.MPIFFTND(dims=par["dims"], axes=par["axes"], norm=par["norm"],
                     real=par["real"], ifftshift_before=ifftshift_before,
                     fftshift_after=fftshift_after, dtype=par["dtype"])
    x = _rand(par, rng)
    xd = pm.DistributedArray.to_dist(torch.from_numpy(x))
    y = op.matvec(xd)
    y_ref = oracle.serial_fftnd_mv(
        x, par["dims"], par["axes"], norm=par["norm"], real=par["real"],
        ifftshift_before=ifftshift_before, fftshift_after=fftshift_after)
    assert_allclose(y.asarray().numpy(), y_ref, rtol=1e-10, atol=1e-11)
    # adjoint on a random data-side vector
    yv = rng.standard_normal(op.shape[0]) \
        + 1j * rng.standard_normal(op.shape[0])
    yd = pm.DistributedArray.to_dist(torch.from_numpy(yv))
    z = op.rmatvec(yd)
    z_ref = oracle.serial_fftnd_rmv(
        yv, par["dims"], par["axes"], norm=par["norm"], real=par["real"],
        ifftshift_before=ifftshift_before, fftshift_after=fftshift_after)
    assert_allclose(z.asarray().numpy(), z_ref, rtol=1e-10, atol=1e-11)
    # __truediv__ (ref FFTND.py:311-316)
    ydiv = op / yd
    assert_allclose(ydiv.asarray().numpy(), z_ref / op._scale,
                    rtol=1e-10, atol=1e-11)


@pytest.mark.parametrize("par", PARS[:4])
def test_mpifft2d_vs_oracle_world1(par):
    rng = np.random.default_rng(4)
    op = pm.MPIFFT2D(dims=par["dims"], axes=par["axes"], norm=par["norm"],
                     real=par["real"], dtype=par["dtype"])
    assert hasattr(op, "f1") and hasattr(op, "f2")
    x = _rand(par, rng)
    xd = pm.DistributedArray.to_dist(torch.from_numpy(x))
    y = op.matvec(xd)
    y_ref = oracle.serial_fftnd_mv(x, par["dims"], par["axes"],
                                   norm=par["norm"], real=par["real"])
    assert_allclose(y.asarray().numpy(), y_ref, rtol=1e-10, atol=1e-11)


def test_fft2d_ctor_validation():
    with pytest.raises(ValueError, match="at least two input dimensions"):
        pm.MPIFFT2D(dims=(8,), axes=(0, 1))
    with pytest.raises(ValueError, match="exactly two dimensions"):
        pm.MPIFFT2D(dims=(8, 8), axes=(0, 1, 1))
    with pytest.raises(ValueError, match="is not one of"):
        pm.MPIFFTND(dims=(8, 8), axes=(0, 1), norm="ortho")
    with pytest.raises(ValueError, match='use "none"'):
        pm.MPIFFTND(dims=(8, 8), axes=(0, 1), norm="backward")


def test_fftshift_helpers_world1():
    rng = np.random.default_rng(5)
    g = rng.standard_normal((9, 6))
    x = pm.DistributedArray.to_dist(torch.from_numpy(g.copy()))
    y = pm.fftshift_nd(x, axes=(1,))
    assert_allclose(y.local_array.numpy(), np.fft.fftshift(g, axes=(1,)))
    x2 = pm.DistributedArray.to_dist(torch.from_numpy(g.copy()))
    y2 = pm.ifftshift_nd(x2, axes=(0, 1))
    assert_allclose(y2.asarray().numpy().reshape(9, 6),
                    np.fft.ifftshift(g, axes=(0, 1)))


def test_real_fft_complex_carrier():
    """real=True with a COMPLEX-storage input (a solver carrying a real
    model in complex storage): the reference casts to real before the
    transform (ref FFTND.py:222-227); regression for the rfftn dtype
    crash."""
    rng = np.random.default_rng(9)
    for dims, axes in (((8, 6), (0, 1)), ((5, 6, 4), (2, 0, 1))):
        op = pm.MPIFFTND(dims=dims, axes=axes, real=True, dtype=np.float64)
        n = int(np.prod(dims))
        x = rng.standard_normal(n) + 1j * rng.standard_normal(n)
        y = op.matvec(pm.DistributedArray.to_dist(torch.from_numpy(x)))
        want = oracle.serial_fftnd_mv(x, dims, axes, real=True,
                                      clinear=False)
        assert_allclose(y.asarray().numpy(), want, rtol=1e-10, atol=1e-11)
