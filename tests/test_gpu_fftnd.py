"""GPU (rocFFT-backed) parity for MPIFFTND/MPIFFT2D vs the serial
oracle — the same grid as tests/test_oracle_fftnd.py (which runs the
torch-CPU path), plus dottest through the operator interface."""
import numpy as np
import pytest
import torch
from numpy.testing import assert_allclose

import oracle
import pylops_mpi_amd as pm
from test_oracle_fftnd import PARS, SHIFTS, _rand

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def _init():
    from pylops_mpi_amd.comm import init_default_comm
    init_default_comm(torch.device("cuda:0"))


@pytest.mark.parametrize("par", PARS)
@pytest.mark.parametrize("ifftshift_before,fftshift_after", SHIFTS)
def test_mpifftnd_gpu_vs_oracle(par, ifftshift_before, fftshift_after):
    rng = np.random.default_rng(7)
    op = pm.MPIFFTND(dims=par["dims"], axes=par["axes"], norm=par["norm"],
                     real=par["real"], ifftshift_before=ifftshift_before,
                     fftshift_after=fftshift_after, dtype=par["dtype"])
    x = _rand(par, rng)
    xd = pm.DistributedArray.to_dist(torch.as_tensor(x, device="cuda"))
    y = op.matvec(xd)
    y_ref = oracle.serial_fftnd_mv(
        x, par["dims"], par["axes"], norm=par["norm"], real=par["real"],
        ifftshift_before=ifftshift_before, fftshift_after=fftshift_after)
    assert_allclose(y.asarray().cpu().numpy(), y_ref,
                    rtol=1e-10, atol=1e-11)
    yv = rng.standard_normal(op.shape[0]) \
        + 1j * rng.standard_normal(op.shape[0])
    yd = pm.DistributedArray.to_dist(torch.as_tensor(yv, device="cuda"))
    z = op.rmatvec(yd)
    z_ref = oracle.serial_fftnd_rmv(
        yv, par["dims"], par["axes"], norm=par["norm"], real=par["real"],
        ifftshift_before=ifftshift_before, fftshift_after=fftshift_after)
    assert_allclose(z.asarray().cpu().numpy(), z_ref,
                    rtol=1e-10, atol=1e-11)


def test_mpifft2d_gpu_dottest():
    # complex 2-D: exact adjoint identity through the operator surface
    op = pm.MPIFFT2D(dims=(24, 18), axes=(0, 1), dtype=np.complex128)
    rng = np.random.default_rng(8)
    u = pm.DistributedArray.to_dist(torch.as_tensor(
        rng.standard_normal(op.shape[1])
        + 1j * rng.standard_normal(op.shape[1]), device="cuda"))
    v = pm.DistributedArray.to_dist(torch.as_tensor(
        rng.standard_normal(op.shape[0])
        + 1j * rng.standard_normal(op.shape[0]), device="cuda"))
    assert pm.dottest(op, u, v, rtol=1e-10)
